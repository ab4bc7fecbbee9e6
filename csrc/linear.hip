// Fused logistic-regression loss+gradient for gfx950 (CDNA4).
//
// Replaces the 3-pass eager path of LogisticRegression._fit's
// eval_loss_grad (X @ W -> log_softmax -> X^T @ G; reference semantics:
// the Breeze LBFGS objective the reference drives through MLlib /
// RDDLossFunction — see reference GBMClassifier.scala:423-431 for the
// aggregator shape) with ONE pass over X: per row the margins, the
// softmax loss and the gradient outer-product are computed in registers,
// so HBM traffic drops from ~3x N*F*4 bytes to ~1x (X is read once; the
// [N,K] margin/G intermediates never exist).
//
// Shape notes: this op is BANDWIDTH-bound, not MFMA-bound — at K <= 8
// classes the two GEMMs are rank-K row-dots / outer-products
// (2*N*F*K flops over N*F*4 bytes = K/2 flop/byte, far under the ~600
// flop/byte MI355X f32 roofline crossover), and gfx950 f32 MFMA runs at
// the f32 vector rate anyway (MI355X_MICROARCH.md "Matrix cores": equal
// rate, not a missing opcode) — so the speed-of-light design is a
// single-pass vector kernel at HBM rate, which this is.
//
// Work layout per wave (64 lanes):
//   lane l owns features {j*256 + l*4 .. +3} for j < n_chunks (dwordx4
//   coalesced loads of X rows), W lives transposed in LDS ([K][F], lane
//   reads ds_read_b128, conflict-free), margins are cross-lane
//   shfl_xor-reduced, gradient accumulates in registers and is flushed
//   once per wave via global f32 atomics (-munsafe-fp-atomics).

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <cstdint>

#define L_CHECK_GPU(x) TORCH_CHECK(x.is_cuda(), #x " must be on GPU")
#define L_CHECK_CONTIG(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")

namespace {

inline int64_t lceil_div(int64_t a, int64_t b) { return (a + b - 1) / b; }

// NC = number of 256-feature chunks (F <= NC*256), K = classes.
// WREG: cache each lane's OWN weight slice in registers — lane l only
// ever multiplies features {j*256 + l*4 .. +3}, so the [K][F] LDS tile
// (and its NC*4*K ds_read_b32 per row per lane) is replaced by NC*4*K
// registers loaded once (measured: removes the per-row LDS issue traffic
// that capped the kernel at ~3 TB/s of the ~6.3 TB/s roofline).
template <int NC, int K, bool WREG, int RR = 2>
__global__ __launch_bounds__(256) void logreg_loss_grad_kernel(
    float* __restrict__ payload,     // [1 + (F+1)*K]: loss, grad[F][K], gbias[K]
    const float* __restrict__ x,     // [N, F]
    const int* __restrict__ y,       // [N]
    const float* __restrict__ w,     // [N]
    const float* __restrict__ wmat,  // [FP, K] row-major (bias row last if has_bias)
    int64_t n, int F, int has_bias) {
  extern __shared__ float wlds[];  // [K][F] transposed weights (!WREG)
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  float wreg[WREG ? NC * 4 : 1][K];
  if (WREG) {
#pragma unroll
    for (int j = 0; j < NC; ++j)
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        const int f = j * 256 + lane * 4 + c;
#pragma unroll
        for (int k = 0; k < K; ++k)
          wreg[j * 4 + c][k] = (f < F) ? wmat[(int64_t)f * K + k] : 0.0f;
      }
  } else {
    for (int i = tid; i < F * K; i += blockDim.x) {
      const int f = i / K, k = i - f * K;
      wlds[k * F + f] = wmat[i];
    }
    __syncthreads();
  }

  float bias[K];
#pragma unroll
  for (int k = 0; k < K; ++k)
    bias[k] = has_bias ? wmat[(int64_t)F * K + k] : 0.0f;

  float greg[NC * 4][K];
#pragma unroll
  for (int j = 0; j < NC * 4; ++j)
#pragma unroll
    for (int k = 0; k < K; ++k) greg[j][k] = 0.0f;
  float gbias[K];
#pragma unroll
  for (int k = 0; k < K; ++k) gbias[k] = 0.0f;
  float loss_acc = 0.0f;

  const int64_t wave_id = (int64_t)blockIdx.x * (blockDim.x >> 6) + wave;
  const int64_t n_waves = (int64_t)gridDim.x * (blockDim.x >> 6);

  // RR rows per iteration: the shfl_xor margin-reduction chains of the
  // rows interleave (independent), and — the round-2 finding — the
  // kernel is INFLIGHT-BYTES-limited (~128 B/wave at RR=2 ≈ the measured
  // 3 TB/s), so RR=4 for the small-K shapes doubles the outstanding
  // loads per wave at an acceptable register cost
  for (int64_t r0 = wave_id * RR; r0 < n; r0 += n_waves * RR) {
    const int nr = (int)std::min<int64_t>(RR, n - r0);
    float xf[RR][NC][4];
    float acc[RR][K];
#pragma unroll
    for (int u = 0; u < RR; ++u)
#pragma unroll
      for (int k = 0; k < K; ++k) acc[u][k] = 0.0f;
#pragma unroll
    for (int u = 0; u < RR; ++u) {
      if (u >= nr) break;
      const float* xr = x + (r0 + u) * F;
#pragma unroll
      for (int j = 0; j < NC; ++j) {
        const int f0 = j * 256 + lane * 4;
        if (f0 + 3 < F) {
          // nontemporal: each x row is read ONCE per launch — keep the
          // stream out of L1/L2 (MI355X_MICROARCH.md nt-weights row:
          // ~18% lower issued->landed latency on streamed loads)
          typedef float vfloat4 __attribute__((ext_vector_type(4)));
          const vfloat4 v = __builtin_nontemporal_load(
              reinterpret_cast<const vfloat4*>(xr + f0));
          xf[u][j][0] = v.x; xf[u][j][1] = v.y;
          xf[u][j][2] = v.z; xf[u][j][3] = v.w;
        } else {
#pragma unroll
          for (int c = 0; c < 4; ++c)
            xf[u][j][c] = (f0 + c < F) ? xr[f0 + c] : 0.0f;
        }
#pragma unroll
        for (int c = 0; c < 4; ++c) {
          const int f = f0 + c;
          if (WREG) {
#pragma unroll
            for (int k = 0; k < K; ++k)
              acc[u][k] = fmaf(xf[u][j][c], wreg[j * 4 + c][k], acc[u][k]);
          } else if (f < F) {
#pragma unroll
            for (int k = 0; k < K; ++k)
              acc[u][k] = fmaf(xf[u][j][c], wlds[k * F + f], acc[u][k]);
          }
        }
      }
    }
    // cross-lane reduce both rows' margins with interleaved chains
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
#pragma unroll
      for (int u = 0; u < RR; ++u)
#pragma unroll
        for (int k = 0; k < K; ++k)
          acc[u][k] += __shfl_xor(acc[u][k], off, 64);
#pragma unroll
    for (int u = 0; u < RR; ++u) {
      if (u >= nr) break;
      const int64_t r = r0 + u;
#pragma unroll
      for (int k = 0; k < K; ++k) acc[u][k] += bias[k];
      // softmax loss + gradient scale (identical in all lanes)
      float m = acc[u][0];
#pragma unroll
      for (int k = 1; k < K; ++k) m = fmaxf(m, acc[u][k]);
      float se = 0.0f;
#pragma unroll
      for (int k = 0; k < K; ++k) se += __expf(acc[u][k] - m);
      const float logz = m + __logf(se);
      const int yr = y[r];
      const float wr = w[r];
      if (lane == 0) loss_acc += wr * (logz - acc[u][yr]);
      float g[K];
#pragma unroll
      for (int k = 0; k < K; ++k) {
        g[k] = (__expf(acc[u][k] - logz) - (k == yr ? 1.0f : 0.0f)) * wr;
        if (lane == 0) gbias[k] += g[k];
      }
      // gradient outer-product accumulate
#pragma unroll
      for (int j = 0; j < NC; ++j)
#pragma unroll
        for (int c = 0; c < 4; ++c)
#pragma unroll
          for (int k = 0; k < K; ++k)
            greg[j * 4 + c][k] =
                fmaf(xf[u][j][c], g[k], greg[j * 4 + c][k]);
    }
  }

  // flush: one atomic per (feature, class) per wave
  float* grad = payload + 1;
#pragma unroll
  for (int j = 0; j < NC; ++j) {
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int f = j * 256 + lane * 4 + c;
      if (f < F) {
#pragma unroll
        for (int k = 0; k < K; ++k)
          atomicAdd(grad + (int64_t)f * K + k, greg[j * 4 + c][k]);
      }
    }
  }
  if (lane == 0) {
    atomicAdd(payload, loss_acc);
    if (has_bias) {
#pragma unroll
      for (int k = 0; k < K; ++k)
        atomicAdd(grad + (int64_t)F * K + k, gbias[k]);
    }
  }
}

template <int NC, int K>
void launch_logreg(float* payload, const float* x, const int* y,
                   const float* w, const float* wmat, int64_t n, int F,
                   int has_bias, hipStream_t stream) {
  const int64_t waves_needed = lceil_div(n, 16);  // >=16 rows per wave
  int blocks = (int)std::min<int64_t>(lceil_div(waves_needed, 4), 4096);
  blocks = std::max(blocks, 1);
  // measured (gpurun_out/r02_logreg.json): W-in-LDS 3.1 TB/s vs
  // W-in-registers 2.86 TB/s at 5M x 1024 K=2 — the LDS reads hide
  // behind the x-load latency and the extra registers cost occupancy,
  // so LDS stays the default; SEA_LOGREG_WREG=1 flips for probing,
  // SEA_LOGREG_RR2=1 forces the narrow 2-row pipeline
  static const bool use_lds = []() {
    const char* e = getenv("SEA_LOGREG_WREG");
    return !(e && e[0] == '1');
  }();
  // RR=4 measured SLOWER (2.21 vs 2.98 TB/s): the xf registers cost a
  // wave per SIMD, shrinking total inflight bytes per CU below the RR=2
  // level — opt-in only, kept for probing on future parts
  static const bool force_rr4 = []() {
    const char* e = getenv("SEA_LOGREG_RR4");
    return e && e[0] == '1';
  }();
  const size_t lds = use_lds ? (size_t)F * K * 4 : 0;
  constexpr bool rr4_ok = (K <= 2 && NC <= 4);
  if (rr4_ok && force_rr4) {
    if (use_lds)
      hipLaunchKernelGGL((logreg_loss_grad_kernel<NC, K, false, 4>),
                         dim3(blocks), dim3(256), lds, stream, payload, x, y,
                         w, wmat, n, F, has_bias);
    else
      hipLaunchKernelGGL((logreg_loss_grad_kernel<NC, K, true, 4>),
                         dim3(blocks), dim3(256), 0, stream, payload, x, y,
                         w, wmat, n, F, has_bias);
    return;
  }
  if (use_lds) {
    hipLaunchKernelGGL((logreg_loss_grad_kernel<NC, K, false, 2>),
                       dim3(blocks), dim3(256), lds, stream, payload, x, y,
                       w, wmat, n, F, has_bias);
  } else {
    hipLaunchKernelGGL((logreg_loss_grad_kernel<NC, K, true, 2>),
                       dim3(blocks), dim3(256), 0, stream, payload, x, y, w,
                       wmat, n, F, has_bias);
  }
}

}  // namespace

// Returns true if this (F, K) combination is supported by the fused kernel.
bool logreg_fused_supported(int64_t F, int64_t K) {
  if (K < 2 || K > 8) return false;
  const int64_t nc = lceil_div(F, 256);
  return nc >= 1 && nc * 4 * K <= 64;  // register-budget cap
}

void logreg_loss_grad(torch::Tensor payload, torch::Tensor x, torch::Tensor y,
                      torch::Tensor w, torch::Tensor wmat, bool has_bias) {
  L_CHECK_GPU(payload); L_CHECK_GPU(x); L_CHECK_GPU(y); L_CHECK_GPU(w);
  L_CHECK_GPU(wmat);
  L_CHECK_CONTIG(payload); L_CHECK_CONTIG(x); L_CHECK_CONTIG(y);
  L_CHECK_CONTIG(w); L_CHECK_CONTIG(wmat);
  const int64_t n = x.size(0);
  const int F = (int)x.size(1);
  const int K = (int)wmat.size(1);
  TORCH_CHECK(wmat.size(0) == F + (has_bias ? 1 : 0), "wmat rows mismatch");
  TORCH_CHECK(payload.numel() == 1 + (int64_t)(F + 1) * K, "payload size");
  TORCH_CHECK(logreg_fused_supported(F, K),
              "unsupported (F, K) for fused logreg: ", F, ", ", K);
  TORCH_CHECK(y.scalar_type() == torch::kInt, "y must be int32");
  auto stream = at::hip::getCurrentHIPStream();
  const int nc = (int)lceil_div(F, 256);
#define LR_LAUNCH(NCC, KK)                                                   \
  launch_logreg<NCC, KK>(payload.data_ptr<float>(), x.data_ptr<float>(),     \
                         y.data_ptr<int>(), w.data_ptr<float>(),             \
                         wmat.data_ptr<float>(), n, F, has_bias ? 1 : 0,     \
                         stream)
  bool done = false;
  switch (K) {
    case 2:
      switch (nc) {
        case 1: LR_LAUNCH(1, 2); done = true; break;
        case 2: LR_LAUNCH(2, 2); done = true; break;
        case 3: LR_LAUNCH(3, 2); done = true; break;
        case 4: LR_LAUNCH(4, 2); done = true; break;
        case 5: LR_LAUNCH(5, 2); done = true; break;
        case 6: LR_LAUNCH(6, 2); done = true; break;
        case 7: LR_LAUNCH(7, 2); done = true; break;
        case 8: LR_LAUNCH(8, 2); done = true; break;
      }
      break;
    case 3:
      switch (nc) {
        case 1: LR_LAUNCH(1, 3); done = true; break;
        case 2: LR_LAUNCH(2, 3); done = true; break;
        case 3: LR_LAUNCH(3, 3); done = true; break;
        case 4: LR_LAUNCH(4, 3); done = true; break;
        case 5: LR_LAUNCH(5, 3); done = true; break;
      }
      break;
    case 4:
      switch (nc) {
        case 1: LR_LAUNCH(1, 4); done = true; break;
        case 2: LR_LAUNCH(2, 4); done = true; break;
        case 3: LR_LAUNCH(3, 4); done = true; break;
        case 4: LR_LAUNCH(4, 4); done = true; break;
      }
      break;
    case 5:
      switch (nc) {
        case 1: LR_LAUNCH(1, 5); done = true; break;
        case 2: LR_LAUNCH(2, 5); done = true; break;
        case 3: LR_LAUNCH(3, 5); done = true; break;
      }
      break;
    case 6:
      switch (nc) {
        case 1: LR_LAUNCH(1, 6); done = true; break;
        case 2: LR_LAUNCH(2, 6); done = true; break;
      }
      break;
    case 7:
      switch (nc) {
        case 1: LR_LAUNCH(1, 7); done = true; break;
        case 2: LR_LAUNCH(2, 7); done = true; break;
      }
      break;
    case 8:
      switch (nc) {
        case 1: LR_LAUNCH(1, 8); done = true; break;
        case 2: LR_LAUNCH(2, 8); done = true; break;
      }
      break;
  }
#undef LR_LAUNCH
  TORCH_CHECK(done, "fused logreg dispatch failed for F=", F, " K=", K);
}
