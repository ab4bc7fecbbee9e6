// spark_ensemble_amd gfx950 (CDNA4 / MI355X) kernels.
//
// Hand-written HIP for the framework's hot path (SURVEY.md section 2.7 maps
// each kernel to the reference semantics it replaces):
//   * hist_build      — LDS-staged per-(node, feature, bin) grad/hess/count
//                       histograms (replaces MLlib DecisionTree's
//                       treeAggregate histogram rounds)
//   * partition_rows  — block-aggregated two-pass node partition
//   * bin_features    — quantile binning (raw f32 -> uint8 bin ids)
//   * tree_predict /
//     forest_predict  — batched node-array tree walks
//   * sample_weights  — counter-based Poisson/Bernoulli row sampling
//   * grad_hess /
//     line_search_eval— fused per-row GBM loss work (one pass over the
//                       shard + one short reduction payload)
//
// Design notes (MI355X_MICROARCH.md): 64-wide waves, 256-thread blocks;
// histograms live in LDS (dynamic, <= 48 KiB keeps 3 blocks/CU resident);
// chunk sizes are chosen host-side so the grid is ~2-3x the resident block
// capacity (minimizes the global-atomic flush that dominated the naive
// version); single-chunk nodes flush with plain stores.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include <chrono>
#include <cstring>
#include <vector>

#define CHECK_GPU(x) TORCH_CHECK(x.is_cuda(), #x " must be on GPU")
#define CHECK_CONTIG(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")

static inline int64_t ceil_div(int64_t a, int64_t b) { return (a + b - 1) / b; }

// Persistent pinned staging for the small per-launch host tables (chunk
// lists, cursors): a pageable .to(device) blocks the HOST until the copy
// completes, and the stream-ordered copy waits for every queued kernel —
// serializing each tree level behind the previous one.  Pinned source +
// non_blocking copy keeps the host running ahead.  Slots rotate per call
// site; reuse is safe because every level ends in a host sync (the split
// fetch) before the slot is touched again.
static torch::Tensor h2d_async(const void* src, size_t bytes, int slot,
                               const torch::Device& dev) {
  static thread_local torch::Tensor pin[16];
  static thread_local hipEvent_t done[16] = {};
  static thread_local unsigned char flip[8] = {};
  // the PREVIOUS copy from this slot may still be pending on the
  // stream: overwriting (or freeing, when the slot grows) the pinned
  // staging before the DMA reads it feeds the kernels garbage
  // descriptors — the depth-12 / 100M-row GPU memory faults.  Each
  // slot PING-PONGS between two buffers so the completion wait below
  // lands on the copy from TWO calls ago — with the host running at
  // most one level ahead that copy has long executed, preserving the
  // run-ahead pipelining the level loop depends on.
  flip[slot] ^= 1;
  const int bi = slot * 2 + flip[slot];
  auto& b = pin[bi];
  if (done[bi]) (void)hipEventSynchronize(done[bi]);
  if (!b.defined() || (size_t)b.numel() < bytes) {
    size_t cap = 4096;
    while (cap < bytes) cap *= 2;
    b = torch::empty({(int64_t)cap},
                     torch::TensorOptions().dtype(torch::kByte)
                         .pinned_memory(true));
  }
  std::memcpy(b.data_ptr(), src, bytes);
  auto d = torch::empty({(int64_t)bytes},
                        torch::TensorOptions().dtype(torch::kByte).device(dev));
  // raw async copy on the current stream: torch's copy_(non_blocking)
  // proved to still block the host behind queued kernels here
  auto stream = at::hip::getCurrentHIPStream();
  (void)hipMemcpyAsync(d.data_ptr(), b.data_ptr(), bytes,
                       hipMemcpyHostToDevice, stream);
  if (!done[bi])
    (void)hipEventCreateWithFlags(&done[bi], hipEventDisableTiming);
  (void)hipEventRecord(done[bi], stream);
  return d;
}

// ---------------------------------------------------------------------------
// sample_weights: counter-based RNG (splitmix64), one state per row
// ---------------------------------------------------------------------------

__device__ inline uint64_t splitmix64(uint64_t z) {
  z += 0x9e3779b97f4a7c15ull;
  z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ull;
  z = (z ^ (z >> 27)) * 0x94d049bb133111ebull;
  return z ^ (z >> 31);
}

__device__ inline float u01(uint64_t z) {
  return ((z >> 40) + 1) * (1.0f / 16777216.0f);  // upper 24 bits -> (0, 1]
}

__global__ void sample_weights_kernel(float* __restrict__ out, int64_t n,
                                      int replacement, float ratio,
                                      uint64_t seed) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < n; i += stride) {
    uint64_t s = seed ^ (0x9e3779b97f4a7c15ull * (uint64_t)(i + 1));
    if (!replacement) {
      uint64_t z = splitmix64(s);
      out[i] = (u01(z) <= ratio) ? 1.0f : 0.0f;
    } else {
      // Knuth Poisson(ratio): E[iters] = ratio + 1 (ratio <= 1 in practice)
      float L = __expf(-ratio);
      float p = 1.0f;
      int k = 0;
      uint64_t z = s;
      do {
        z = splitmix64(z);
        p *= u01(z);
        k++;
      } while (p > L && k < 64);
      out[i] = (float)(k - 1);
    }
  }
}

void sample_weights(torch::Tensor out, bool replacement, double ratio,
                    int64_t seed, int64_t rank) {
  CHECK_GPU(out);
  CHECK_CONTIG(out);
  int64_t n = out.numel();
  auto stream = at::hip::getCurrentHIPStream();
  uint64_t s = (uint64_t)seed * 0x100000001b3ull + (uint64_t)rank * 0x9e3779b9ull;
  int threads = 256;
  int blocks = (int)std::min<int64_t>(ceil_div(n, threads), 8192);
  hipLaunchKernelGGL(sample_weights_kernel, dim3(blocks), dim3(threads), 0,
                     stream, out.data_ptr<float>(), n, replacement ? 1 : 0,
                     (float)ratio, s);
}

// ---------------------------------------------------------------------------
// bin_features: [N, F] f32 + [F, B-1] edges -> [N, F] u8
//   edges for one feature stay in registers? no — binary search in L1/L2.
//   Thread covers one (row, feature); consecutive threads cover consecutive
//   features of one row so the x reads coalesce.
// ---------------------------------------------------------------------------

// Feature-tiled: grid (row_chunks, F/16); each block stages its 16
// features' quantile edges in LDS (16 x nedges f32 <= 16 KiB at 255
// edges), then loops rows with 64-B coalesced x row-slice loads and
// LDS binary searches — the naive per-element global-edge search ran at
// 34 GB/s, this at HBM row-read rate.
__global__ void bin_features_kernel(uint8_t* __restrict__ out,
                                    const float* __restrict__ x,
                                    const float* __restrict__ edges, int64_t n,
                                    int F, int nedges) {
  constexpr int FT = 16;
  extern __shared__ float elds[];  // [FT][nedges]
  const int f0 = blockIdx.y * FT;
  const int nf = min(FT, F - f0);
  for (int i = threadIdx.x; i < nf * nedges; i += blockDim.x)
    elds[i] = edges[(int64_t)f0 * nedges + i];
  __syncthreads();

  const bool full = (nf == FT) && ((F & 15) == 0);
  int64_t r = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t rstride = gridDim.x * (int64_t)blockDim.x;
  for (; r < n; r += rstride) {
    float v[FT];
    if (full) {
      const float4* xr = reinterpret_cast<const float4*>(x + r * F + f0);
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const float4 t = xr[q];
        v[q * 4 + 0] = t.x; v[q * 4 + 1] = t.y;
        v[q * 4 + 2] = t.z; v[q * 4 + 3] = t.w;
      }
    } else {
      for (int j = 0; j < nf; ++j) v[j] = x[r * F + f0 + j];
    }
    uint8_t b[FT];
    for (int j = 0; j < nf; ++j) {
      const float* e = elds + j * nedges;
      int lo = 0, hi = nedges;  // first index with e[idx] >= v
      while (lo < hi) {
        const int mid = (lo + hi) >> 1;
        if (e[mid] >= v[j]) hi = mid; else lo = mid + 1;
      }
      b[j] = (uint8_t)lo;
    }
    if (full) {
      uint4 packed;
      unsigned* w = reinterpret_cast<unsigned*>(&packed);
#pragma unroll
      for (int q = 0; q < 4; ++q)
        w[q] = (unsigned)b[q * 4] | ((unsigned)b[q * 4 + 1] << 8) |
               ((unsigned)b[q * 4 + 2] << 16) | ((unsigned)b[q * 4 + 3] << 24);
      *reinterpret_cast<uint4*>(out + r * F + f0) = packed;
    } else {
      for (int j = 0; j < nf; ++j) out[r * F + f0 + j] = b[j];
    }
  }
}

void bin_features(torch::Tensor out, torch::Tensor x, torch::Tensor edges) {
  CHECK_GPU(out); CHECK_GPU(x); CHECK_GPU(edges);
  CHECK_CONTIG(out); CHECK_CONTIG(x); CHECK_CONTIG(edges);
  int64_t n = x.size(0);
  int F = (int)x.size(1);
  int nedges = (int)edges.size(1);
  TORCH_CHECK(nedges <= 1024, "bin_features: too many edges for LDS staging");
  auto stream = at::hip::getCurrentHIPStream();
  const int threads = 256;
  const int fgroups = (int)ceil_div(F, 16);
  // 4 blocks/CU: a 1-block/CU grid (old 4096/fgroups cap) left the
  // float4 row loads latency-bound at ~1.2 TB/s (rocprof r02)
  int rblocks = (int)std::min<int64_t>(
      ceil_div(n, threads), std::max<int64_t>(1, 16384 / fgroups));
  const size_t lds = (size_t)16 * nedges * 4;
  hipLaunchKernelGGL(bin_features_kernel, dim3(rblocks, fgroups),
                     dim3(threads), lds, stream, out.data_ptr<uint8_t>(),
                     x.data_ptr<float>(), edges.data_ptr<float>(), n, F,
                     nedges);
}

// ---------------------------------------------------------------------------
// hist_build
//   grid = (row_chunks, feature_groups); LDS histogram [FG][B][CELLS] u64.
//
//   KEY gfx950 design point (measured, tools/probe_hist2.hip +
//   profiles/r01_hist_probe.md): LDS f32 atomicAdd (ds_add_f32) runs ~13x
//   slower than LDS u64 integer atomicAdd (ds_add_u64) — 101 vs 1395 G
//   bump/s at B=256 — so the histogram accumulates in FIXED POINT: each
//   64-bit cell packs one SIGNED channel (gradient, high 32) and one
//   NON-NEGATIVE channel (hessian/count, low 32) and is bumped with ONE
//   ds_add_u64.  Channel convention (tree_grower.py): channels [0, D) are
//   signed gradients, channels [D, C) are non-negative hess/count, so
//   CELLS = max(D, C - D).  Host-chosen per-channel scales guarantee
//   |sum| < 2^30 per chunk per bin (no field overflow, low->high carries
//   impossible since low fields are non-negative and bounded).
//
//   chunks: int32 [n_chunks, 5] = (node, start, len, single_chunk_flag,
//   col0); quantization scales ride in the same upload (one H2D per
//   launch).  ``col0`` is the node's base column in a gh matrix of row
//   stride CH — 0/CH==C for single-tree builds; a FOREST build (K trees
//   or K classes grown level-synchronously, reference
//   GBMClassifier.scala:377-411 parallel futures) interleaves per-tree
//   channel groups [g_t, h_t(, cnt_t)] and sets col0 = tree * C, so one
//   launch histograms every active node of every tree.
//   Flush converts back to f32 into out [n_nodes, F, B, C]; single-chunk
//   nodes use plain stores, multi-chunk nodes integer staging atomics.
// ---------------------------------------------------------------------------

template <int DC, int NC>
__global__ void hist_build_kernel(
    float* __restrict__ out,            // [n_nodes, F, B, C]
    long long* __restrict__ stage,      // [n_nodes, F, B, C] i64 (multi-chunk)
    const uint8_t* __restrict__ bins,   // [N, F]
    const float* __restrict__ gh,       // [N, CH]
    const int* __restrict__ row_idx,    // [M]
    const int* __restrict__ chunks,     // [2*C + n_chunks*5] (scales first)
    int F, int B, int FG, int CH, int identity_rows) {
  constexpr int C = DC + NC;
  constexpr int CELLS = DC > NC ? DC : NC;
  extern __shared__ unsigned long long lds64[];  // FG * B * CELLS
  const float* scales = reinterpret_cast<const float*>(chunks);
  const int* chk = chunks + 2 * C + blockIdx.x * 5;
  const int fg = blockIdx.y;
  const int f0 = fg * FG;
  const int nf = min(FG, F - f0);
  const int node = chk[0];
  const int start = chk[1];
  const int len = chk[2];
  const int single = chk[3];
  const int col0 = chk[4];

  const int lds_cells = FG * B * CELLS;
  for (int i = threadIdx.x; i < lds_cells; i += blockDim.x) lds64[i] = 0ull;
  __syncthreads();

  // bins rows are read FG bytes at a time in 16-B uint4 pieces when the
  // group is 16-aligned and fully inside the row
  const bool vec = (FG == 16 || FG == 32 || FG == 64) && (f0 + FG <= F) &&
                   ((F % FG) == 0);
  const int nh = FG >> 4;

  for (int i = threadIdx.x; i < len; i += blockDim.x) {
    const int r = identity_rows ? start + i : row_idx[start + i];
    const float* g = gh + (int64_t)r * CH + col0;
    // quantize once per row, pack one u64 addend per cell
    unsigned long long addend[CELLS];
#pragma unroll
    for (int c = 0; c < CELLS; ++c) addend[c] = 0ull;
#pragma unroll
    for (int d = 0; d < DC; ++d) {
      const int iv = __float2int_rn(g[d] * scales[d]);
      addend[d] |= ((unsigned long long)(unsigned)iv) << 32;
    }
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      const int iv = __float2int_rn(g[DC + c] * scales[DC + c]);
      addend[c] |= (unsigned)iv;
    }
    if (vec) {
      for (int hh = 0; hh < nh; ++hh) {
        const uint4 bv = *reinterpret_cast<const uint4*>(
            bins + (int64_t)r * F + f0 + 16 * hh);
        const unsigned w[4] = {bv.x, bv.y, bv.z, bv.w};
#pragma unroll
        for (int q = 0; q < 4; ++q) {
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            const int b = (w[q] >> (8 * j)) & 0xff;
            unsigned long long* cell =
                lds64 + (((16 * hh + q * 4 + j) * B) + b) * CELLS;
#pragma unroll
            for (int c = 0; c < CELLS; ++c) atomicAdd(cell + c, addend[c]);
          }
        }
      }
    } else {
      const uint8_t* br = bins + (int64_t)r * F + f0;
      for (int f = 0; f < nf; ++f) {
        const int b = br[f];
        unsigned long long* cell = lds64 + ((f * B) + b) * CELLS;
#pragma unroll
        for (int c = 0; c < CELLS; ++c) atomicAdd(cell + c, addend[c]);
      }
    }
  }
  __syncthreads();

  // flush. Single-chunk nodes: decode fixed point -> f32 straight into
  // out (plain stores).  Multi-chunk nodes: accumulate INTEGER partials
  // into the i64 staging buffer — integer atomics are order-independent,
  // so histograms stay bitwise deterministic across runs (decode kernel
  // below converts once at the end).
  if (single) {
    float* dst = out + (((int64_t)node * F + f0) * B) * C;
    for (int i = threadIdx.x; i < nf * B; i += blockDim.x) {
      const int f = i / B, b = i - f * B;
      const unsigned long long* cell = lds64 + ((f * B) + b) * CELLS;
      float* o = dst + ((int64_t)f * B + b) * C;
#pragma unroll
      for (int d = 0; d < DC; ++d)
        o[d] = (float)(int)(unsigned)(cell[d] >> 32) / scales[d];
#pragma unroll
      for (int c = 0; c < NC; ++c)
        o[DC + c] =
            (float)(int)(unsigned)(cell[c] & 0xFFFFFFFFull) / scales[DC + c];
    }
  } else {
    long long* sdst = stage + (((int64_t)node * F + f0) * B) * C;
    for (int i = threadIdx.x; i < nf * B; i += blockDim.x) {
      const int f = i / B, b = i - f * B;
      const unsigned long long* cell = lds64 + ((f * B) + b) * CELLS;
      long long* o = sdst + ((int64_t)f * B + b) * C;
#pragma unroll
      for (int d = 0; d < DC; ++d) {
        const int v = (int)(unsigned)(cell[d] >> 32);
        if (v) atomicAdd((unsigned long long*)(o + d),
                         (unsigned long long)(long long)v);
      }
#pragma unroll
      for (int c = 0; c < NC; ++c) {
        const int v = (int)(unsigned)(cell[c] & 0xFFFFFFFFull);
        if (v) atomicAdd((unsigned long long*)(o + DC + c),
                         (unsigned long long)(long long)v);
      }
    }
  }
}

// decode the i64 staging sums of multi-chunk nodes into f32 out
__global__ void hist_decode_kernel(float* __restrict__ out,
                                   const long long* __restrict__ stage,
                                   const int* __restrict__ nodes,  // multi-chunk node ids (after scales)
                                   const float* __restrict__ scales, int FBC,
                                   int C) {
  const int node = nodes[blockIdx.y];
  const int64_t base = (int64_t)node * FBC;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < FBC;
       i += gridDim.x * (int64_t)blockDim.x) {
    const int c = (int)(i % C);
    out[base + i] = (float)stage[base + i] / scales[c];
  }
}

void hist_build(torch::Tensor out, torch::Tensor bins, torch::Tensor gh,
                torch::Tensor row_idx, torch::Tensor node_offsets,
                int64_t num_bins, int64_t d_dims, torch::Tensor max_abs,
                bool identity_rows, torch::Tensor node_col0,
                int64_t c_per_node) {
  CHECK_GPU(out); CHECK_GPU(bins); CHECK_GPU(gh); CHECK_GPU(row_idx);
  CHECK_CONTIG(out); CHECK_CONTIG(bins); CHECK_CONTIG(gh); CHECK_CONTIG(row_idx);
  TORCH_CHECK(!node_offsets.is_cuda(), "node_offsets stays on host");
  TORCH_CHECK(!max_abs.is_cuda(), "max_abs stays on host");
  const int F = (int)bins.size(1);
  const int B = (int)num_bins;
  const int CH = (int)gh.size(1);  // gh row stride
  // channels PER NODE: gh width for single-tree builds; explicit for
  // forest builds where gh interleaves per-tree channel groups
  const int C = c_per_node > 0 ? (int)c_per_node : CH;
  const bool forest = node_col0.numel() > 0;
  TORCH_CHECK(!forest || node_col0.numel() == node_offsets.numel() - 1,
              "node_col0 must have one entry per node");
  const int D = d_dims > 0 ? (int)d_dims : (C >= 3 ? C - 2 : C - 1);
  const int NN = C - D;
  TORCH_CHECK(C <= 8, "gh channels capped at 8 (grad dims + hess + count)");
  TORCH_CHECK(D >= 1 && NN >= 1 && NN <= 2,
              "hist_build: channels must be D grads + 1-2 nonneg, got C=", C,
              " D=", D);
  TORCH_CHECK(max_abs.numel() == C, "max_abs must have C entries");
  const int CELLS = std::max(D, NN);

  // feature-group size: 16 (32-KiB LDS at CELLS=1) or 32 (64-KiB, halves
  // the random-row read amplification at deep levels); SEA_HIST_FG overrides
  static const int fg_env = []() {
    const char* e = getenv("SEA_HIST_FG");
    return e ? atoi(e) : 0;
  }();
  static const int lds_budget = []() {
    const char* e = getenv("SEA_HIST_LDS");
    return e ? atoi(e) : 163840;  // up to the full 160 KiB LDS per CU
  }();
  // measured (profiles/r01_hist_probe3): FG=64 + 1024 threads + 128 KiB
  // LDS (1 block/CU, 16 waves) runs 1.9x FG=16/256 — 64-B row slices
  // quarter the random-row fetch amplification and the wider block keeps
  // the LDS atomic pipe fed
  // probe-measured optima (profiles/r01_hist_probe3): CELLS==1 ->
  // FG=64 / 1024 threads / 128 KiB; CELLS==2 -> FG=32 / 1024 / 128 KiB;
  // CELLS>=3 (wide multiclass) keeps the conservative 64-KiB config
  // CELLS>=3 (multiclass): FG=16 fits in ~99 KiB with the B+1-free
  // layout and keeps the vectorized uint4 row loads; the old 64-KiB cap
  // forced FG=8 byte loads and DOUBLED the per-row gh re-reads (one per
  // feature group)
  const int budget = lds_budget;
  int FG = std::max<int>(1, std::min<int>(F, budget / (B * CELLS * 8)));
  if (CELLS == 1 && FG >= 64 && (F % 64) == 0) FG = 64;
  else if (CELLS <= 2 && FG >= 32 && (F % 32) == 0) FG = 32;
  else if (FG >= 16) FG = 16;
  else if (FG >= 8) FG = 8;
  else if (FG >= 4) FG = 4;
  if (fg_env == 16 || fg_env == 32 || fg_env == 64)
    FG = std::min(FG, fg_env);
  const int threads =
      (FG >= 64 || (CELLS == 2 && FG >= 32)) ? 1024
      : ((FG >= 32 || (CELLS >= 3 && FG >= 16)) ? 512 : 256);
  const int n_groups = (int)ceil_div(F, FG);

  // ---- adaptive chunking: target ~resident-grid x OVERSUB blocks --------
  const int n_nodes = (int)node_offsets.numel() - 1;
  auto offs = node_offsets.accessor<int64_t, 1>();
  int64_t total_rows = 0;
  for (int nd = 0; nd < n_nodes; ++nd) total_rows += offs[nd + 1] - offs[nd];
  const size_t lds_bytes_pre = (size_t)FG * B * CELLS * 8;
  const int blocks_per_cu =
      std::max<int>(1, (int)(163840 / std::max<size_t>(1, lds_bytes_pre)));
  const int64_t resident = (int64_t)256 * blocks_per_cu;
  // oversubscription: exactly-resident grids measured fastest at 1
  // block/CU (probe3); modest oversub amortizes imbalance otherwise
  static const double oversub_env = []() {
    const char* e = getenv("SEA_HIST_OVERSUB");
    return e ? atof(e) : 0.0;
  }();
  const double oversub =
      oversub_env > 0 ? oversub_env : (blocks_per_cu == 1 ? 1.0 : 1.5);
  const int64_t target_chunks = std::max<int64_t>(
      1, (int64_t)(resident * oversub) / std::max(1, n_groups));
  int64_t chunk_rows = std::max<int64_t>(
      4096, ceil_div(total_rows, target_chunks));
  // fixed-point headroom: per-chunk per-bin |sum| must stay < 2^30
  const int64_t CHUNK_CAP = 1 << 20;
  chunk_rows = std::min(chunk_rows, CHUNK_CAP);

  // quantization scales: scale_c = 2^30 / (chunk_rows * max_abs_c)
  auto max_abs_f = max_abs.to(torch::kFloat32).contiguous();
  auto ma = max_abs_f.accessor<float, 1>();
  std::vector<int> chunk_v(2 * C);
  float* scales_f = reinterpret_cast<float*>(chunk_v.data());
  for (int c = 0; c < C; ++c) {
    float m = ma[c];
    if (!(m > 0.0f) || !std::isfinite(m)) m = 1.0f;
    double s = (double)(1u << 30) / ((double)chunk_rows * (double)m);
    // clamp so a single value cannot overflow int32 either
    s = std::min(s, (double)(1u << 30) / (double)m);
    scales_f[c] = (float)s;
  }
  const int* col0_p = forest ? node_col0.data_ptr<int>() : nullptr;
  std::vector<int> multi_nodes;
  for (int nd = 0; nd < n_nodes; ++nd) {
    int64_t s = offs[nd], e = offs[nd + 1];
    const int single = (e - s) <= chunk_rows ? 1 : 0;
    if (!single) multi_nodes.push_back(nd);
    if (e == s) {
      // `out` is allocated uninitialized (flush/decode fully overwrite
      // every chunked node) — a ZERO-ROW node emits no chunk, so its
      // slice must be zeroed here (rank-local empty nodes are real in
      // sharded fits: this node's rows may all live on other ranks)
      out.slice(0, nd, nd + 1).zero_();
    }
    for (int64_t c = s; c < e; c += chunk_rows) {
      chunk_v.push_back(nd);
      chunk_v.push_back((int)c);
      chunk_v.push_back((int)std::min<int64_t>(chunk_rows, e - c));
      chunk_v.push_back(single);
      chunk_v.push_back(col0_p ? col0_p[nd] : 0);
    }
  }
  if ((int)chunk_v.size() == 2 * C) {
    out.zero_();  // dispatch allocates `out` uninitialized
    return;
  }
  const int n_chunks = (int)((chunk_v.size() - 2 * C) / 5);
  auto chunks_b = h2d_async(chunk_v.data(), chunk_v.size() * 4, 0,
                            bins.device());
  auto chunks = chunks_b.view(torch::kInt32);

  // i64 integer staging for multi-chunk nodes (order-independent flush =>
  // bitwise-deterministic histograms)
  torch::Tensor stage;
  long long* stage_ptr = nullptr;
  if (!multi_nodes.empty()) {
    stage = torch::zeros({(int64_t)n_nodes, (int64_t)F, (int64_t)B, (int64_t)C},
                         out.options().dtype(torch::kInt64));
    stage_ptr = reinterpret_cast<long long*>(stage.data_ptr<int64_t>());
  }

  auto stream = at::hip::getCurrentHIPStream();
  const size_t lds_bytes = lds_bytes_pre;
#define HB_LAUNCH(DD, NNN)                                                   \
  do {                                                                       \
    if (lds_bytes > 65536)                                                   \
      (void)hipFuncSetAttribute(                                             \
          reinterpret_cast<const void*>(&hist_build_kernel<DD, NNN>),        \
          hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds_bytes);       \
    hipLaunchKernelGGL((hist_build_kernel<DD, NNN>),                         \
                       dim3(n_chunks, n_groups), dim3(threads), lds_bytes,   \
                       stream, out.data_ptr<float>(), stage_ptr,             \
                       bins.data_ptr<uint8_t>(),                             \
                       gh.data_ptr<float>(), row_idx.data_ptr<int>(),        \
                       chunks.data_ptr<int>(), F, B, FG, CH,                 \
                       identity_rows ? 1 : 0);                               \
  } while (0)
  const int key = D * 10 + NN;
  switch (key) {
    case 11: HB_LAUNCH(1, 1); break;
    case 12: HB_LAUNCH(1, 2); break;
    case 21: HB_LAUNCH(2, 1); break;
    case 22: HB_LAUNCH(2, 2); break;
    case 31: HB_LAUNCH(3, 1); break;
    case 32: HB_LAUNCH(3, 2); break;
    case 41: HB_LAUNCH(4, 1); break;
    case 42: HB_LAUNCH(4, 2); break;
    case 51: HB_LAUNCH(5, 1); break;
    case 52: HB_LAUNCH(5, 2); break;
    case 61: HB_LAUNCH(6, 1); break;
    case 62: HB_LAUNCH(6, 2); break;
    case 71: HB_LAUNCH(7, 1); break;
    default: TORCH_CHECK(false, "hist_build: unsupported D/NN ", D, "/", NN);
  }
#undef HB_LAUNCH
  if (!multi_nodes.empty()) {
    auto nodes_b = h2d_async(multi_nodes.data(), multi_nodes.size() * 4, 1,
                             bins.device());
    auto nodes_t = nodes_b.view(torch::kInt32);
    const int FBC = F * B * C;
    const int dblocks = (int)std::min<int64_t>(ceil_div(FBC, 256), 1024);
    hipLaunchKernelGGL(hist_decode_kernel,
                       dim3(dblocks, (int)multi_nodes.size()), dim3(256), 0,
                       stream, out.data_ptr<float>(), stage_ptr,
                       nodes_t.data_ptr<int>(),
                       reinterpret_cast<const float*>(chunks.data_ptr<int>()),
                       FBC, C);
  }
}

// ---------------------------------------------------------------------------
// partition_rows: two passes per chunk, ONE bins read.
//   pass 1: decide left/right per row, stash the decision as a ballot
//           bitmask in LDS (1 bit/row, chunk <= 64 Ki rows = 8 KiB), count
//           via popcount, ONE global atomic per block per side
//   pass 2: replay from the LDS bitmask (no second random bins gather),
//           scatter via ballot prefix + LDS block cursors
// ---------------------------------------------------------------------------

__global__ void partition_kernel(
    int* __restrict__ new_rows,        // [M]
    int* __restrict__ cursors,         // [n_nodes, 2] = {lcur, rcur}
    const uint8_t* __restrict__ bins,  // [N, F]
    const uint8_t* __restrict__ bins_t,  // [F, N] or null
    const int* __restrict__ row_idx,   // [M]
    const int* __restrict__ chunks,    // [n_chunks, 3]
    const int* __restrict__ feat,      // [n_nodes]
    const int* __restrict__ thr,       // [n_nodes]
    int F, long long Nrows) {
  __shared__ int base_l, base_r, loc_l, loc_r;
  __shared__ int wl[8];
  extern __shared__ unsigned long long bits[];  // ceil(chunk_rows/64)
  const int chunk = blockIdx.x;
  const int node = chunks[chunk * 3 + 0];
  const int start = chunks[chunk * 3 + 1];
  const int len = chunks[chunk * 3 + 2];
  const int f = feat[node];
  const int t = thr[node];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  // pass 1: decide + stash bitmask + count
  int my_l = 0;
  for (int i = threadIdx.x; i < ((len + 63) & ~63); i += blockDim.x) {
    const bool active = i < len;
    bool left = false;
    if (active) {
      const int r = row_idx[start + i];
      if (f < 0) {
        left = true;  // unsplit node: every row goes left (no load!)
      } else {
        // the transposed matrix turns the per-row gather (1 useful byte
        // per 64-B line with the row-major layout) into a locally-dense
        // read: a node's rows at level L are ~2^-L dense in [0, N), so
        // a line yields ~64/2^L useful bytes at shallow levels
        const uint8_t bv = bins_t ? bins_t[(int64_t)f * Nrows + r]
                                  : bins[(int64_t)r * F + f];
        left = bv <= t;
      }
    }
    const unsigned long long m = __ballot(active && left);
    if (lane == 0) {
      bits[i >> 6] = m;
      my_l += (int)__popcll(m);
    }
  }
  if (lane == 0) wl[wave] = my_l;
  __syncthreads();
  if (threadIdx.x == 0) {
    int tot = 0;
    for (int w0 = 0; w0 < (int)(blockDim.x / 64); ++w0) tot += wl[w0];
    base_l = atomicAdd(cursors + node * 2 + 0, tot);
    base_r = atomicAdd(cursors + node * 2 + 1, -(len - tot)) - (len - tot);
    loc_l = 0;
    loc_r = 0;
  }
  __syncthreads();

  // pass 2: scatter from the stashed decisions (row_idx re-read is
  // sequential; bins is NOT touched again)
  for (int i = threadIdx.x; i < ((len + 63) & ~63); i += blockDim.x) {
    const bool active = i < len;
    const unsigned long long word = bits[i >> 6];
    const bool left = (word >> lane) & 1ull;
    const unsigned long long lmask = word;
    const unsigned long long rmask = __ballot(active) & ~word;
    int lbase = 0, rbase = 0;
    if (lane == 0) {
      lbase = atomicAdd(&loc_l, (int)__popcll(lmask));
      rbase = atomicAdd(&loc_r, (int)__popcll(rmask));
    }
    lbase = __shfl(lbase, 0, 64);
    rbase = __shfl(rbase, 0, 64);
    if (active) {
      const int r = row_idx[start + i];
      if (left) {
        const int pos = base_l + lbase + (int)__popcll(lmask & ((1ull << lane) - 1ull));
        new_rows[pos] = r;
      } else {
        const int pos = base_r + rbase + (int)__popcll(rmask & ((1ull << lane) - 1ull));
        new_rows[pos] = r;
      }
    }
  }
}

void partition_rows(torch::Tensor new_rows, torch::Tensor left_counts,
                    torch::Tensor bins, torch::Tensor bins_t,
                    torch::Tensor row_idx,
                    torch::Tensor node_offsets, torch::Tensor feat,
                    torch::Tensor thr) {
  auto tpA = std::chrono::steady_clock::now();
  CHECK_GPU(new_rows); CHECK_GPU(bins); CHECK_GPU(row_idx);
  CHECK_GPU(feat); CHECK_GPU(thr); CHECK_GPU(left_counts);
  const int F = (int)bins.size(1);
  const int n_nodes = (int)node_offsets.numel() - 1;
  auto offs = node_offsets.accessor<int64_t, 1>();

  std::vector<int> cur_v(n_nodes * 2);
  std::vector<int> chunk_v;
  int64_t total_rows = offs[n_nodes];
  int64_t chunk_rows =
      std::max<int64_t>(8192, ceil_div(total_rows, (int64_t)2048));
  chunk_rows = std::min<int64_t>(chunk_rows, 65536);  // 8 KiB LDS bitmask
  for (int nd = 0; nd < n_nodes; ++nd) {
    cur_v[nd * 2 + 0] = (int)offs[nd];
    cur_v[nd * 2 + 1] = (int)offs[nd + 1];
    for (int64_t c = offs[nd]; c < offs[nd + 1]; c += chunk_rows) {
      chunk_v.push_back(nd);
      chunk_v.push_back((int)c);
      chunk_v.push_back((int)std::min<int64_t>(chunk_rows, offs[nd + 1] - c));
    }
  }
  auto stream = at::hip::getCurrentHIPStream();
  static const bool part_dbg = getenv("SEA_PART_DEBUG") != nullptr;
  auto tp0 = std::chrono::steady_clock::now();
  auto cursors_b = h2d_async(cur_v.data(), cur_v.size() * 4, 2, bins.device());
  auto cursors = cursors_b.view(torch::kInt32);
  auto tp1 = std::chrono::steady_clock::now();
  if (!chunk_v.empty()) {
    auto chunks_b = h2d_async(chunk_v.data(), chunk_v.size() * 4, 3,
                              bins.device());
    auto chunks = chunks_b.view(torch::kInt32);
    auto tp2 = std::chrono::steady_clock::now();
    const int n_chunks = (int)(chunk_v.size() / 3);
    const size_t bit_lds = (size_t)((chunk_rows + 63) / 64) * 8;
    const uint8_t* bt_ptr =
        bins_t.defined() && bins_t.numel() ? bins_t.data_ptr<uint8_t>()
                                           : nullptr;
    hipLaunchKernelGGL(partition_kernel, dim3(n_chunks), dim3(256), bit_lds,
                       stream,
                       new_rows.data_ptr<int>(), cursors.data_ptr<int>(),
                       bins.data_ptr<uint8_t>(), bt_ptr,
                       row_idx.data_ptr<int>(),
                       chunks.data_ptr<int>(), feat.data_ptr<int>(),
                       thr.data_ptr<int>(), F, (long long)bins.size(0));
    if (part_dbg) {
      auto tp3 = std::chrono::steady_clock::now();
      auto us = [](auto a, auto b) {
        return std::chrono::duration_cast<std::chrono::microseconds>(b - a)
            .count();
      };
      fprintf(stderr,
              "[part c++] entry=%ldus cursors=%ldus chunks=%ldus launch=%ldus\n",
              (long)us(tpA, tp0), (long)us(tp0, tp1), (long)us(tp1, tp2),
              (long)us(tp2, tp3));
    }
  }
  auto lcur = cursors.view({n_nodes, 2}).select(1, 0);
  // NOTE: this upload was a pageable .to(device) — a SYNCHRONOUS copy that
  // stalled the host ~1 ms/level behind the queued hist/split kernels
  // (found via SEA_PART_DEBUG bracketing); pinned async staging fixes it
  std::vector<int> seg_v(n_nodes);
  for (int nd = 0; nd < n_nodes; ++nd) seg_v[nd] = (int)offs[nd];
  auto seg_b = h2d_async(seg_v.data(), seg_v.size() * 4, 4, bins.device());
  auto seg_start = seg_b.view(torch::kInt32);
  left_counts.copy_(lcur - seg_start);
}

// ---------------------------------------------------------------------------
// tree walks
// ---------------------------------------------------------------------------

__global__ void tree_predict_kernel(float* __restrict__ out,       // [N, D]
                                    const float* __restrict__ x,   // [N, F]
                                    const int* __restrict__ feat,  // [nodes]
                                    const float* __restrict__ thr,
                                    const int* __restrict__ left,
                                    const float* __restrict__ leaf,  // [nodes, D]
                                    int64_t n, int F, int D) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < n; i += stride) {
    const float* xr = x + i * F;
    int node = 0;
    int f = feat[0];
    while (f >= 0) {
      node = left[node] + (xr[f] <= thr[node] ? 0 : 1);
      f = feat[node];
    }
    const float* lv = leaf + (int64_t)node * D;
    float* o = out + i * D;
    for (int d = 0; d < D; ++d) o[d] = lv[d];
  }
}

void tree_predict(torch::Tensor out, torch::Tensor x, torch::Tensor feat,
                  torch::Tensor thr, torch::Tensor left, torch::Tensor leaf) {
  CHECK_GPU(out); CHECK_GPU(x);
  int64_t n = x.size(0);
  int F = (int)x.size(1);
  int D = (int)leaf.size(1);
  auto stream = at::hip::getCurrentHIPStream();
  int blocks = (int)std::min<int64_t>(ceil_div(n, 256), 16384);
  hipLaunchKernelGGL(tree_predict_kernel, dim3(blocks), dim3(256), 0, stream,
                     out.data_ptr<float>(), x.data_ptr<float>(),
                     feat.data_ptr<int>(), thr.data_ptr<float>(),
                     left.data_ptr<int>(), leaf.data_ptr<float>(), n, F, D);
}

__global__ void forest_predict_kernel(
    float* __restrict__ out,           // [N, D] (+=)
    const float* __restrict__ x,       // [N, F]
    const int* __restrict__ feat,      // concat nodes
    const float* __restrict__ thr,
    const int* __restrict__ left,
    const float* __restrict__ leaf,    // concat [nodes, D]
    const int* __restrict__ tree_off,  // [T]
    const float* __restrict__ w,       // [T]
    int64_t n, int F, int D, int T) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < n; i += stride) {
    const float* xr = x + i * F;
    float* o = out + i * D;
    for (int t = 0; t < T; ++t) {
      const int base = tree_off[t];
      int node = base;
      int f = feat[node];
      while (f >= 0) {
        node = base + left[node] + (xr[f] <= thr[node] ? 0 : 1);
        f = feat[node];
      }
      const float* lv = leaf + (int64_t)node * D;
      const float wt = w[t];
      for (int d = 0; d < D; ++d) o[d] += wt * lv[d];
    }
  }
}

void forest_predict(torch::Tensor out, torch::Tensor x, torch::Tensor feat,
                    torch::Tensor thr, torch::Tensor left, torch::Tensor leaf,
                    torch::Tensor tree_off, torch::Tensor w, int64_t D) {
  CHECK_GPU(out); CHECK_GPU(x);
  int64_t n = x.size(0);
  int F = (int)x.size(1);
  int T = (int)tree_off.numel();
  auto stream = at::hip::getCurrentHIPStream();
  int blocks = (int)std::min<int64_t>(ceil_div(n, 256), 16384);
  hipLaunchKernelGGL(forest_predict_kernel, dim3(blocks), dim3(256), 0, stream,
                     out.data_ptr<float>(), x.data_ptr<float>(),
                     feat.data_ptr<int>(), thr.data_ptr<float>(),
                     left.data_ptr<int>(), leaf.data_ptr<float>(),
                     tree_off.data_ptr<int>(), w.data_ptr<float>(), n, F,
                     (int)D, T);
}

// ---------------------------------------------------------------------------
// forest_predict2: LDS-staged tree-group-tiled forest inference.
//
//   The v1 kernel above walks ALL T trees per row with every node fetch a
//   dependent global load (L1-thrashing: 64 lanes x ~800 scattered node +
//   feature reads) — measured 72.7M rows/s for a 100-tree depth-8 forest
//   (docs/performance.md r01).  v2 restructures serving the MI355X way:
//
//   * nodes are packed host-side into ONE u64 per node:
//       low 16  = feature (s16, -1 leaf)   | mid 16 = left-child (s16,
//       sibling right = left+1)            | high 32 = f32 bits of the
//       threshold — or, for D == 1, the WEIGHTED leaf value when feat < 0
//       (leaf tensors never touched in-kernel for scalar outputs).
//   * trees are tiled into groups of <= 16 Ki nodes (128 KiB); each block
//     stages its group's packed nodes in LDS once (ds_read_b64 is 256
//     B/clk/CU vs ~200-900 cyc global latency per dependent node hop),
//     then grid-strides over rows: per (row, group) all walks hit LDS.
//   * one f32 atomicAdd per (row, channel, group) merges group partials —
//     different rows never contend.
//
//   grid (row_chunks, n_groups) x 1024 threads; 1 block/CU at 128 KiB LDS
//   = 16 waves covering the ds_read dependent-latency chain.
//   Reference semantics: the model transform loops at reference
//   BaggingRegressor.scala:221-228, GBMClassifier.scala:567-589.
// ---------------------------------------------------------------------------

// BINNED == false: x is [N, F] f32, payload high-32 = f32 threshold bits.
// BINNED == true:  x is u8 rank-transformed rows (each feature value
//   replaced by its lower-bound rank within the forest's OWN sorted
//   per-feature threshold set — an EXACT transform: x <= thr  <=>
//   rank(x) <= rank(thr)), payload high-32 = the threshold's rank.
//   Rows shrink 4x (u8 vs f32), so the divergent per-lane gathers stop
//   thrashing L1 (256 B rows: 4 cache lines instead of 16).
//   xt != 0: the u8 matrix is TRANSPOSED [F, N] — consecutive lanes hold
//   consecutive rows, and all lanes of a wave sit on the SAME node for
//   the first tree levels (they entered the tree together), so their
//   gathers land on consecutive bytes of one feature column: hop h costs
//   <= min(2^h, 64) cache lines instead of 64.
template <bool D1, bool BINNED>
__global__ void forest_predict2_kernel(
    float* __restrict__ out,                       // [N, D] (+=, pre-zeroed)
    const void* __restrict__ xv_,                  // [N, F] f32 | u8
    const unsigned long long* __restrict__ nodes,  // [total] packed
    const float* __restrict__ leaf,                // [total, D] (D > 1)
    const int* __restrict__ tree_off,              // [T] global node base
    const float* __restrict__ w,                   // [T]
    const int* __restrict__ groups,                // [G, 4]
    int64_t n, int F, int D, int xt) {
  extern __shared__ unsigned long long tlds[];  // group nodes
  const int* grp = groups + blockIdx.y * 4;
  const int first_tree = grp[0];
  const int n_trees = grp[1];
  const int node_base = grp[2];
  const int n_nodes = grp[3];
  for (int i = threadIdx.x; i < n_nodes; i += blockDim.x)
    tlds[i] = nodes[node_base + i];
  __syncthreads();

  // one row per lane: a 2-row-per-lane ILP variant was measured 1.5-4x
  // SLOWER (merged divergent walk loops serialize both chains through
  // every iteration); wave-level parallelism already covers the latency
  int64_t r = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; r < n; r += stride) {
    const float* xr = BINNED ? nullptr : (const float*)xv_ + r * F;
    const uint8_t* br =
        BINNED ? (const uint8_t*)xv_ + (xt ? r : r * F) : nullptr;
    // transposed stride: br[f * n] addresses feature f of this row
    float acc[8];
    if (!D1)
#pragma unroll
      for (int d = 0; d < 8; ++d) acc[d] = 0.0f;
    float acc1 = 0.0f;
    for (int t = 0; t < n_trees; ++t) {
      const int toff = tree_off[first_tree + t] - node_base;
      int node = toff;
      unsigned long long nd = tlds[node];
      int f = (short)(nd & 0xFFFFu);
      while (f >= 0) {
        const int left = (short)((nd >> 16) & 0xFFFFu);
        bool go_left;
        if (BINNED) {
          const uint8_t bv = xt ? br[(int64_t)f * n] : br[f];
          go_left = bv <= (unsigned)(nd >> 32);
        } else {
          go_left = xr[f] <= __uint_as_float((unsigned)(nd >> 32));
        }
        node = toff + left + (go_left ? 0 : 1);
        nd = tlds[node];
        f = (short)(nd & 0xFFFFu);
      }
      if (D1) {
        acc1 += __uint_as_float((unsigned)(nd >> 32));  // weighted leaf
      } else {
        const float wt = w[first_tree + t];
        const float* lv = leaf + (int64_t)(node_base + node) * D;
        for (int d = 0; d < D; ++d) acc[d] += wt * lv[d];
      }
    }
    if (D1) {
      atomicAdd(out + r, acc1);
    } else {
      float* o = out + r * D;
      for (int d = 0; d < D; ++d)
        if (acc[d] != 0.0f) atomicAdd(o + d, acc[d]);
    }
  }
}

// ---------------------------------------------------------------------------
// transpose_u8: [N, F] u8 -> [F, N] u8 for the transposed serving walk.
// torch's generic u8 2D transpose measured 19 ms at 10M x 256 (~26x off
// bandwidth); this is a 64x64 LDS-tiled transpose moving u32 quads on
// both sides (full tiles) — byte path only on edge tiles.
// ---------------------------------------------------------------------------
// gather_ranges / leaf_scatter: the forest grower's per-level row-arena
// maintenance.  Both replace a 3-kernel torch expansion
// (repeat_interleave + cumsum + arange gather — the reference's per-node
// driver loops have no on-GPU analog) with ONE pass:
// each output position binary-searches the tiny per-segment prefix table
// (L2-resident) and streams contiguous source ranges.
// ---------------------------------------------------------------------------

__device__ inline int seg_of(const long long* __restrict__ cum, int n_segs,
                             long long p) {
  int lo = 0, hi = n_segs;  // cum[0] = 0, cum[n_segs] = total
  while (hi - lo > 1) {
    const int mid = (lo + hi) >> 1;
    if (cum[mid] <= p) lo = mid; else hi = mid;
  }
  return lo;
}

__global__ void gather_ranges_kernel(
    int* __restrict__ out,               // [total]
    const int* __restrict__ src,
    const long long* __restrict__ seg,   // [n_segs+1 cum..., n_segs starts...]
    int n_segs, long long total) {
  const long long* cum = seg;
  const long long* starts = seg + n_segs + 1;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long p = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       p < total; p += stride) {
    const int s = seg_of(cum, n_segs, p);
    out[p] = src[starts[s] + (p - cum[s])];
  }
}

__global__ void leaf_scatter_kernel(
    float* __restrict__ tp,              // [N, T] row-major
    const int* __restrict__ row_idx,     // level row arena
    const long long* __restrict__ seg,   // [n_segs+1 cum..., n_segs starts...]
    const int* __restrict__ tree,        // [n_segs]
    const float* __restrict__ val,       // [n_segs]
    int T, int n_segs, long long total) {
  const long long* cum = seg;
  const long long* starts = seg + n_segs + 1;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long p = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       p < total; p += stride) {
    const int s = seg_of(cum, n_segs, p);
    const int r = row_idx[starts[s] + (p - cum[s])];
    tp[(long long)r * T + tree[s]] = val[s];
  }
}

void gather_ranges(torch::Tensor out, torch::Tensor src,
                   torch::Tensor starts, torch::Tensor lens) {
  CHECK_GPU(out); CHECK_GPU(src);
  const int n_segs = (int)starts.numel();
  const long long total = out.numel();
  if (n_segs == 0 || total == 0) return;
  auto st = starts.accessor<int64_t, 1>();
  auto ln = lens.accessor<int64_t, 1>();
  std::vector<long long> seg_v(2 * n_segs + 1);
  seg_v[0] = 0;
  for (int i = 0; i < n_segs; ++i) {
    seg_v[i + 1] = seg_v[i] + ln[i];
    seg_v[n_segs + 1 + i] = st[i];
  }
  TORCH_CHECK(seg_v[n_segs] == total, "gather_ranges: lens sum != out size");
  auto seg_b = h2d_async(seg_v.data(), seg_v.size() * 8, 5, src.device());
  auto stream = at::hip::getCurrentHIPStream();
  const int blocks =
      (int)std::min<long long>(8192, ceil_div(total, (long long)1024));
  hipLaunchKernelGGL(gather_ranges_kernel, dim3(blocks), dim3(256), 0, stream,
                     out.data_ptr<int>(), src.data_ptr<int>(),
                     (const long long*)seg_b.data_ptr(), n_segs, total);
}

void leaf_scatter(torch::Tensor tp, torch::Tensor row_idx,
                  torch::Tensor starts, torch::Tensor lens,
                  torch::Tensor tree, torch::Tensor val) {
  CHECK_GPU(tp); CHECK_GPU(row_idx);
  const int n_segs = (int)starts.numel();
  if (n_segs == 0) return;
  const int T = (int)tp.size(1);
  auto st = starts.accessor<int64_t, 1>();
  auto ln = lens.accessor<int64_t, 1>();
  std::vector<long long> seg_v(2 * n_segs + 1);
  seg_v[0] = 0;
  for (int i = 0; i < n_segs; ++i) {
    seg_v[i + 1] = seg_v[i] + ln[i];
    seg_v[n_segs + 1 + i] = st[i];
  }
  const long long total = seg_v[n_segs];
  if (total == 0) return;
  auto seg_b = h2d_async(seg_v.data(), seg_v.size() * 8, 5, tp.device());
  std::vector<int> tv(n_segs);
  std::vector<float> vv(n_segs);
  auto ta = tree.accessor<int64_t, 1>();
  auto va = val.accessor<float, 1>();
  for (int i = 0; i < n_segs; ++i) { tv[i] = (int)ta[i]; vv[i] = va[i]; }
  auto tree_b = h2d_async(tv.data(), tv.size() * 4, 6, tp.device());
  std::vector<char> vb(n_segs * 4);
  std::memcpy(vb.data(), vv.data(), n_segs * 4);
  auto val_b = h2d_async(vb.data(), vb.size(), 7, tp.device());
  auto stream = at::hip::getCurrentHIPStream();
  const int blocks =
      (int)std::min<long long>(8192, ceil_div(total, (long long)1024));
  hipLaunchKernelGGL(leaf_scatter_kernel, dim3(blocks), dim3(256), 0, stream,
                     tp.data_ptr<float>(), row_idx.data_ptr<int>(),
                     (const long long*)seg_b.data_ptr(),
                     (const int*)tree_b.data_ptr(),
                     (const float*)val_b.data_ptr(), T, n_segs, total);
}

// ---------------------------------------------------------------------------

__global__ void transpose_u8_kernel(uint8_t* __restrict__ out,  // [F, N]
                                    const uint8_t* __restrict__ in,  // [N, F]
                                    int64_t n, int F) {
  __shared__ uint8_t tile[64][68];  // row stride 68 = 4*17: aligned, unbanked
  const int64_t r0 = (int64_t)blockIdx.x * 64;
  const int f0 = blockIdx.y * 64;
  const int tid = threadIdx.x;  // 256 threads
  const bool full = (r0 + 64 <= n) && (f0 + 64 <= F) && ((F & 3) == 0) &&
                    ((n & 3) == 0);
  if (full) {
    for (int i = tid; i < 1024; i += 256) {
      const int rr = i >> 4;
      const int cq = i & 15;
      const unsigned v = *reinterpret_cast<const unsigned*>(
          in + (r0 + rr) * F + f0 + cq * 4);
      *reinterpret_cast<unsigned*>(&tile[rr][cq * 4]) = v;
    }
    __syncthreads();
    for (int i = tid; i < 1024; i += 256) {
      const int ff = i >> 4;
      const int rq = i & 15;
      const unsigned v =
          (unsigned)tile[rq * 4 + 0][ff] |
          ((unsigned)tile[rq * 4 + 1][ff] << 8) |
          ((unsigned)tile[rq * 4 + 2][ff] << 16) |
          ((unsigned)tile[rq * 4 + 3][ff] << 24);
      *reinterpret_cast<unsigned*>(out + (int64_t)(f0 + ff) * n + r0 +
                                   rq * 4) = v;
    }
  } else {
    for (int i = tid; i < 4096; i += 256) {
      const int rr = i >> 6;
      const int ff = i & 63;
      if (r0 + rr < n && f0 + ff < F)
        tile[rr][ff] = in[(r0 + rr) * F + f0 + ff];
    }
    __syncthreads();
    for (int i = tid; i < 4096; i += 256) {
      const int ff = i >> 6;
      const int rr = i & 63;
      if (r0 + rr < n && f0 + ff < F)
        out[(int64_t)(f0 + ff) * n + r0 + rr] = tile[rr][ff];
    }
  }
}

void transpose_u8(torch::Tensor out, torch::Tensor in) {
  CHECK_GPU(out); CHECK_GPU(in);
  CHECK_CONTIG(out); CHECK_CONTIG(in);
  TORCH_CHECK(in.scalar_type() == torch::kUInt8, "u8 only");
  const int64_t n = in.size(0);
  const int F = (int)in.size(1);
  TORCH_CHECK(out.size(0) == F && out.size(1) == n, "shape mismatch");
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(transpose_u8_kernel,
                     dim3((unsigned)ceil_div(n, 64), (unsigned)ceil_div(F, 64)),
                     dim3(256), 0, stream, out.data_ptr<uint8_t>(),
                     in.data_ptr<uint8_t>(), n, F);
}

void forest_predict2(torch::Tensor out, torch::Tensor x, torch::Tensor nodes,
                     torch::Tensor leaf, torch::Tensor tree_off,
                     torch::Tensor w, torch::Tensor groups, int64_t D,
                     int64_t max_group_nodes, int64_t x_transposed) {
  CHECK_GPU(out); CHECK_GPU(x); CHECK_GPU(nodes); CHECK_GPU(groups);
  CHECK_CONTIG(out); CHECK_CONTIG(x); CHECK_CONTIG(nodes);
  TORCH_CHECK(D <= 8, "forest_predict2: D <= 8");
  const bool binned = x.scalar_type() == torch::kUInt8;
  TORCH_CHECK(!x_transposed || binned, "x_transposed implies u8 input");
  int64_t n = x.size(x_transposed ? 1 : 0);
  int F = (int)x.size(x_transposed ? 0 : 1);
  const int G = (int)groups.size(0);
  const size_t lds = (size_t)max_group_nodes * 8;
  TORCH_CHECK(lds <= 163840, "forest_predict2: group too big for LDS");
  auto stream = at::hip::getCurrentHIPStream();
  const int threads = 1024;
  int rblocks = (int)std::min<int64_t>(ceil_div(n, threads), 8192);
  // fill the chip even for few groups
  rblocks = std::max(rblocks, (int)std::min<int64_t>(
                                  ceil_div(512, (int64_t)G), ceil_div(n, 64)));
#define FP2_LAUNCH(DD1, BB)                                                   \
  do {                                                                        \
    if (lds > 65536)                                                          \
      (void)hipFuncSetAttribute(                                              \
          reinterpret_cast<const void*>(&forest_predict2_kernel<DD1, BB>),    \
          hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);              \
    hipLaunchKernelGGL((forest_predict2_kernel<DD1, BB>), dim3(rblocks, G),   \
                       dim3(threads), lds, stream, out.data_ptr<float>(),     \
                       x.data_ptr(),                                          \
                       (const unsigned long long*)nodes.data_ptr<int64_t>(),  \
                       leaf.numel() ? leaf.data_ptr<float>() : nullptr,       \
                       tree_off.data_ptr<int>(), w.data_ptr<float>(),         \
                       groups.data_ptr<int>(), n, F, (int)D,                  \
                       (int)x_transposed);                                    \
  } while (0)
  if (D == 1 && binned) FP2_LAUNCH(true, true);
  else if (D == 1) FP2_LAUNCH(true, false);
  else if (binned) FP2_LAUNCH(false, true);
  else FP2_LAUNCH(false, false);
#undef FP2_LAUNCH
}

// ---------------------------------------------------------------------------
// fused GBM loss kernels
//   loss ids match spark_ensemble_amd/boosting/losses.py LOSS_IDS
// ---------------------------------------------------------------------------

#define L_SQUARED 0
#define L_ABSOLUTE 1
#define L_LOGCOSH 2
#define L_SCALEDLOGCOSH 3
#define L_HUBER 4
#define L_QUANTILE 5
#define L_LOGLOSS 6
#define L_EXPONENTIAL 7
#define L_BERNOULLI 8

__device__ inline float logcosh_f(float d) {
  float a = fabsf(d);
  return a + log1pf(__expf(-2.0f * a)) - 0.6931471805599453f;
}

// scalar (dim=1) losses: returns loss, writes d(loss)/d(pred) and hessian
__device__ inline float scalar_loss_grad(int loss_id, float param, float y,
                                         float p, float* grad, float* hess) {
  switch (loss_id) {
    case L_SQUARED: {
      float d = y - p;
      *grad = -d;
      *hess = 1.0f;
      return 0.5f * d * d;
    }
    case L_ABSOLUTE: {
      float d = y - p;
      *grad = -copysignf(1.0f, d);
      *hess = 0.0f;
      return fabsf(d);
    }
    case L_LOGCOSH: {
      float d = y - p;
      float t = tanhf(d);
      *grad = -t;
      *hess = 1.0f - t * t;
      return logcosh_f(d);
    }
    case L_SCALEDLOGCOSH: {
      float d = y - p;
      float s = (y > p) ? param : 1.0f - param;
      float t = tanhf(d);
      *grad = -s * t;
      *hess = s * (1.0f - t * t);
      return s * logcosh_f(d);
    }
    case L_HUBER: {
      float d = y - p;
      float ad = fabsf(d);
      if (ad <= param) {
        *grad = -d;
        *hess = 1.0f;
        return 0.5f * d * d;
      }
      *grad = -param * copysignf(1.0f, d);
      *hess = 0.0f;
      return param * (ad - 0.5f * param);
    }
    case L_QUANTILE: {
      float d = y - p;
      *grad = (d > 0.0f) ? -param : (1.0f - param);
      *hess = 0.0f;
      return (d > 0.0f) ? param * d : (param - 1.0f) * d;
    }
    case L_EXPONENTIAL: {
      // y in {-1, 1}
      float e = __expf(-y * p);
      *grad = -y * e;
      *hess = e;  // y^2 = 1
      return e;
    }
    case L_BERNOULLI: {
      // overflow-safe forms: exp(z) -> inf past |z|~88 made the naive
      // hess 4e/(1+e)^2 = inf/inf = NaN once boosting margins grow,
      // kicking the stage-weight search from 3-6 Newton evals to ~25
      // Brent evals per round (measured +4 ms/round after round ~40).
      // sigmoid(-z) = t/(1+t) with t = exp(-|z|) in (0,1] is finite
      // everywhere; grad = -2y*sigmoid(-z), hess = 4*sig*(1-sig).
      float z = 2.0f * y * p;
      float az = fabsf(z);
      float t = __expf(-az);
      float sig_neg = (z >= 0.0f) ? t / (1.0f + t) : 1.0f / (1.0f + t);
      *grad = -2.0f * y * sig_neg;
      *hess = 4.0f * sig_neg * (1.0f - sig_neg);
      return fmaxf(-z, 0.0f) + log1pf(t);
    }
  }
  *grad = 0.0f;
  *hess = 0.0f;
  return 0.0f;
}

// grad_hess: per-row fused pseudo-residual inputs.
//   dim == 1 scalar losses, or logloss with dim = K (softmax).
__global__ void grad_hess_kernel(float* __restrict__ grad,  // [N, D]
                                 float* __restrict__ hess,  // [N, D] or null
                                 const float* __restrict__ label,  // [N, D]
                                 const float* __restrict__ pred,   // [N, D]
                                 int64_t n, int D, int loss_id, float param,
                                 int want_hess) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < n; i += stride) {
    if (loss_id == L_LOGLOSS) {
      // two passes over d; no local arrays (rule: runtime-indexed locals
      // become register-select chains or scratch)
      const float* pr = pred + i * D;
      const float* lr = label + i * D;
      float m = pr[0];
      for (int d = 1; d < D; ++d) m = fmaxf(m, pr[d]);
      float s = 0.0f;
      for (int d = 0; d < D; ++d) s += __expf(pr[d] - m);
      float inv = 1.0f / s;
      for (int d = 0; d < D; ++d) {
        float q = __expf(pr[d] - m) * inv;
        grad[i * D + d] = q - lr[d];
        if (want_hess) hess[i * D + d] = q * (1.0f - q);
      }
    } else {
      float g, h;
      scalar_loss_grad(loss_id, param, label[i], pred[i], &g, &h);
      grad[i] = g;
      if (want_hess) hess[i] = h;
    }
  }
}

void grad_hess(torch::Tensor grad, torch::Tensor hess, torch::Tensor label,
               torch::Tensor pred, int64_t loss_id, double param,
               bool want_hess) {
  CHECK_GPU(grad); CHECK_GPU(label); CHECK_GPU(pred);
  int64_t n = pred.size(0);
  int D = (int)pred.size(1);
  auto stream = at::hip::getCurrentHIPStream();
  int blocks = (int)std::min<int64_t>(ceil_div(n, 256), 16384);
  hipLaunchKernelGGL(grad_hess_kernel, dim3(blocks), dim3(256), 0, stream,
                     grad.data_ptr<float>(),
                     want_hess ? hess.data_ptr<float>() : nullptr,
                     label.data_ptr<float>(), pred.data_ptr<float>(), n, D,
                     (int)loss_id, (float)param, want_hess ? 1 : 0);
}

// line_search_eval: one pass computing
//   payload[0]   = sum_i w_i * loss(y_i, p_i + a . d_i)
//   payload[1+d] = sum_i w_i * d_id * dloss/dp_id
//   payload[1+D] = sum_i w_i * d_i^2 * d2loss/dp_i^2   (want_hess, D == 1;
//                  feeds the safeguarded-Newton stage-weight search)
template <int D>
__global__ void line_search_eval_kernel(
    float* __restrict__ payload,        // [1 + D (+1)] (pre-zeroed)
    const float* __restrict__ label,    // [N, D]
    const float* __restrict__ pred,     // [N, D]
    const float* __restrict__ dir,      // [N, D]
    const float* __restrict__ weight,   // [N]
    const float* __restrict__ coeff,    // [D]
    int64_t n, int loss_id, float param, int want_hess) {
  __shared__ float wacc[4][2 + D];  // per-wave partials (256 threads = 4 waves)
  __syncthreads();

  float cf[D];
#pragma unroll
  for (int d = 0; d < D; ++d) cf[d] = coeff[d];

  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  float loss_sum = 0.0f;
  float hsum = 0.0f;
  float gsum[D];
#pragma unroll
  for (int d = 0; d < D; ++d) gsum[d] = 0.0f;

  for (; i < n; i += stride) {
    const float w = weight[i];
    if (D > 1 || loss_id == L_LOGLOSS) {
      const float* pr = pred + i * D;
      const float* dr = dir + i * D;
      const float* lr = label + i * D;
      float pv[D];
      float m = -1e30f;
#pragma unroll
      for (int d = 0; d < D; ++d) {
        pv[d] = pr[d] + cf[d] * dr[d];
        m = fmaxf(m, pv[d]);
      }
      float s = 0.0f;
#pragma unroll
      for (int d = 0; d < D; ++d) s += __expf(pv[d] - m);
      float lse = m + __logf(s);
      float l = 0.0f;
      float inv = 1.0f / s;
#pragma unroll
      for (int d = 0; d < D; ++d) {
        l += -lr[d] * (pv[d] - lse);
        float q = __expf(pv[d] - m) * inv;
        gsum[d] += w * dr[d] * (q - lr[d]);
      }
      loss_sum += w * l;
    } else {
      float g, h;
      float l = scalar_loss_grad(loss_id, param, label[i],
                                 pred[i] + cf[0] * dir[i], &g, &h);
      if (!isfinite(l)) l = 3.0e38f;
      loss_sum += w * l;
      gsum[0] += w * dir[i] * g;
      if (want_hess) hsum += w * dir[i] * dir[i] * h;
    }
  }

  // wave reduce then LDS
  for (int off = 32; off > 0; off >>= 1) {
    loss_sum += __shfl_down(loss_sum, off, 64);
    hsum += __shfl_down(hsum, off, 64);
#pragma unroll
    for (int d = 0; d < D; ++d) gsum[d] += __shfl_down(gsum[d], off, 64);
  }
  if ((threadIdx.x & 63) == 0) {
    const int wv = threadIdx.x >> 6;
    wacc[wv][0] = loss_sum;
    wacc[wv][1 + D] = hsum;
#pragma unroll
    for (int d = 0; d < D; ++d) wacc[wv][1 + d] = gsum[d];
  }
  __syncthreads();
  // per-block partial, waves summed in FIXED order (no atomics anywhere)
  // -> bitwise deterministic; final fixed-order reduction in
  // ls_reduce_kernel
  if (threadIdx.x == 0) {
    float* row = payload + (int64_t)blockIdx.x * (2 + D);
    const int nw = (int)(blockDim.x >> 6);
    for (int k = 0; k < 2 + D; ++k) {
      float v = 0.0f;
      for (int wv = 0; wv < nw; ++wv) v += wacc[wv][k];
      row[k] = v;
    }
  }
}

// sum the [n_blocks, stride] partials with a FIXED thread->block
// assignment (grid of one block; lane j sums rows j, j+J, ... then lane 0
// of each slot adds the J lane-partials in order => bitwise deterministic)
__global__ void ls_reduce_kernel(float* __restrict__ out,      // [out_width]
                                 const float* __restrict__ partials,
                                 int n_blocks, int stride, int out_width) {
  __shared__ float lp[256];
  const int k = threadIdx.x % stride;        // payload slot
  const int j = threadIdx.x / stride;        // lane within slot
  const int J = blockDim.x / stride;         // lanes per slot
  float s = 0.0f;
  if (j < J) {
    for (int b = j; b < n_blocks; b += J)
      s += partials[(int64_t)b * stride + k];
  }
  lp[threadIdx.x] = s;
  __syncthreads();
  if (threadIdx.x < out_width) {
    float v = 0.0f;
    for (int jj = 0; jj < J; ++jj) v += lp[jj * stride + threadIdx.x];
    out[threadIdx.x] = v;
  }
}

// Safeguarded-Newton update on the 1-D stage-weight search state —
// the device side of the chained line search (line_search.py _newton_1d):
// state = [a, blo, bhi, best_a, best_f, evals, done]; payload = [f, g, h]
// from the preceding eval at coeff = &state[0].  Mirrors the host loop
// exactly so a chain interrupted mid-way can resume on the host.
__global__ void newton_update_kernel(float* __restrict__ st,
                                     const float* __restrict__ p,
                                     float tol) {
  if (st[6] != 0.0f) return;
  const float f = p[0], g = p[1], h = p[2];
  if (!isfinite(f) || !isfinite(g) || !isfinite(h)) { st[6] = 2.0f; return; }
  st[5] += 1.0f;
  const float a = st[0];
  if (f < st[4]) { st[4] = f; st[3] = a; }
  if (g > 0.0f) st[2] = a; else st[1] = a;
  if (fabsf(g) <= tol * fmaxf(1.0f, fabsf(f)) || (st[2] - st[1]) <= tol) {
    st[6] = 1.0f;
    return;
  }
  float nxt;
  if (h > 1e-12f) {
    nxt = a - g / h;
    if (!(st[1] < nxt && nxt < st[2])) nxt = 0.5f * (st[1] + st[2]);
    if (fabsf(nxt - a) <= tol * fmaxf(1.0f, fabsf(a))) {
      st[0] = nxt;
      st[6] = 1.0f;
      return;
    }
  } else {
    nxt = 0.5f * (st[1] + st[2]);
  }
  st[0] = nxt;
}

void newton_chain_1d(torch::Tensor state, torch::Tensor payloads,
                     torch::Tensor label, torch::Tensor pred,
                     torch::Tensor dir, torch::Tensor weight,
                     int64_t loss_id, double param, double tol,
                     int64_t iters) {
  // queue `iters` x (eval -> reduce -> update) with NO host sync: the
  // eval kernel reads the live alpha from state[0] (its coeff pointer),
  // the update kernel advances it.  Steps after convergence evaluate at
  // a frozen alpha and are discarded by the update's done guard — the
  // caller sizes `iters` from the previous round's eval count, so the
  // overshoot is ~0 in steady state.
  CHECK_GPU(state); CHECK_GPU(pred);
  const int64_t n = pred.size(0);
  TORCH_CHECK(pred.size(1) == 1, "newton_chain_1d is scalar-dim only");
  TORCH_CHECK(state.numel() >= 7 && payloads.numel() >= 3 * iters, "sizes");
  auto stream = at::hip::getCurrentHIPStream();
  const int blocks = (int)std::min<int64_t>(ceil_div(n, 256 * 8), 2048);
  auto partials = torch::empty({(int64_t)blocks * 3}, state.options());
  const int rthreads = (256 / 3) * 3;
  for (int k = 0; k < (int)iters; ++k) {
    float* pay = payloads.data_ptr<float>() + 3 * k;
    hipLaunchKernelGGL(line_search_eval_kernel<1>, dim3(blocks), dim3(256), 0,
                       stream, partials.data_ptr<float>(),
                       label.data_ptr<float>(), pred.data_ptr<float>(),
                       dir.data_ptr<float>(), weight.data_ptr<float>(),
                       state.data_ptr<float>(), n, (int)loss_id,
                       (float)param, 1);
    hipLaunchKernelGGL(ls_reduce_kernel, dim3(1), dim3(rthreads), 0, stream,
                       pay, partials.data_ptr<float>(), blocks, 3, 3);
    hipLaunchKernelGGL(newton_update_kernel, dim3(1), dim3(1), 0, stream,
                       state.data_ptr<float>(), pay, (float)tol);
  }
}

void line_search_eval(torch::Tensor payload, torch::Tensor label,
                      torch::Tensor pred, torch::Tensor dir,
                      torch::Tensor weight, torch::Tensor coeff,
                      int64_t loss_id, double param, bool want_hess) {
  CHECK_GPU(payload); CHECK_GPU(pred);
  int64_t n = pred.size(0);
  int D = (int)pred.size(1);
  TORCH_CHECK(D <= 8, "line_search_eval supports dim <= 8");
  TORCH_CHECK(D == 1 || loss_id == L_LOGLOSS,
              "vector line search only for logloss");
  TORCH_CHECK(!want_hess || D == 1, "hessian accumulation is scalar-only");
  TORCH_CHECK(payload.numel() >= 1 + D + (want_hess ? 1 : 0), "payload size");
  auto stream = at::hip::getCurrentHIPStream();
  int blocks = (int)std::min<int64_t>(ceil_div(n, 256 * 8), 2048);
  const int width = 2 + D;
  auto partials = torch::empty({(int64_t)blocks * width},
                               payload.options());
#define LS_LAUNCH(DD)                                                         \
  hipLaunchKernelGGL(line_search_eval_kernel<DD>, dim3(blocks), dim3(256), 0, \
                     stream, partials.data_ptr<float>(),                      \
                     label.data_ptr<float>(), pred.data_ptr<float>(),         \
                     dir.data_ptr<float>(), weight.data_ptr<float>(),         \
                     coeff.data_ptr<float>(), n, (int)loss_id, (float)param,  \
                     want_hess ? 1 : 0)
  switch (D) {
    case 1: LS_LAUNCH(1); break;
    case 2: LS_LAUNCH(2); break;
    case 3: LS_LAUNCH(3); break;
    case 4: LS_LAUNCH(4); break;
    case 5: LS_LAUNCH(5); break;
    case 6: LS_LAUNCH(6); break;
    case 7: LS_LAUNCH(7); break;
    case 8: LS_LAUNCH(8); break;
  }
#undef LS_LAUNCH
  const int rthreads = (256 / width) * width;  // whole lanes per slot
  hipLaunchKernelGGL(ls_reduce_kernel, dim3(1), dim3(rthreads), 0, stream,
                     payload.data_ptr<float>(), partials.data_ptr<float>(),
                     blocks, width, std::min(width, (int)payload.numel()));
}

// ---------------------------------------------------------------------------
// split_argmax: fused best-split search over node histograms.
//   Replaces the eager chain cumsum -> score -> where -> argmax -> gather of
//   reference.split_search (ops/reference.py:150-204; semantics: XGBoost-style
//   newton gain |G|^2/(H+lam), reference analog: the split evaluation inside
//   MLlib DecisionTree invoked via fitBaseLearner, ensembleParams.scala:64-81).
//   One wave scans one feature's B bins (lane-chunked + wave prefix scan in
//   registers), packs (sortable gain, feat*B+bin) into a u64 and atomicMax's
//   a per-node cell — no [n,F,B,C] cumsum/gain tensors ever touch HBM.
// ---------------------------------------------------------------------------

__device__ inline unsigned f32_sortable(float f) {
  unsigned u = __float_as_uint(f);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}
__device__ inline float sortable_f32(unsigned s) {
  unsigned u = (s & 0x80000000u) ? (s & 0x7FFFFFFFu) : ~s;
  return __uint_as_float(u);
}

template <int C>
__global__ void split_argmax_kernel(
    unsigned long long* __restrict__ best,  // [n_nodes] pre-zeroed
    const float* __restrict__ hist,         // [n_nodes, F, B, C]
    int F, int B, int D, float lam, float min_child_weight,
    float min_instances) {
  const int node = blockIdx.x;
  const int f = blockIdx.y * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  if (f >= F) return;
  const float* h = hist + (((int64_t)node * F + f) * (int64_t)B) * C;
  const int BPL = (B + 63) >> 6;  // <= 4 for B <= 256
  const int idx_c = C - 1;        // count channel

  float loc[4][C];
  float tot[C];
#pragma unroll
  for (int c = 0; c < C; ++c) tot[c] = 0.0f;
  for (int i = 0; i < BPL; ++i) {
    const int b = lane * BPL + i;
#pragma unroll
    for (int c = 0; c < C; ++c) loc[i][c] = 0.0f;
    if (b < B) {
#pragma unroll
      for (int c = 0; c < C; ++c) {
        loc[i][c] = h[(int64_t)b * C + c];
        tot[c] += loc[i][c];
      }
    }
  }
  // wave-wide exclusive prefix + parent totals (shfl scan per channel)
  float pre[C], parent[C];
#pragma unroll
  for (int c = 0; c < C; ++c) {
    float v = tot[c];
    for (int off = 1; off < 64; off <<= 1) {
      const float u = __shfl_up(v, off, 64);
      if (lane >= off) v += u;
    }
    pre[c] = v - tot[c];
    parent[c] = __shfl(v, 63, 64);
  }
  float pg2 = 0.0f;
#pragma unroll
  for (int d = 0; d < C; ++d)
    if (d < D) pg2 += parent[d] * parent[d];
  const float parent_score = pg2 / (parent[D] + lam);

  float run[C];
#pragma unroll
  for (int c = 0; c < C; ++c) run[c] = pre[c];
  unsigned best_s = 0u;
  int best_b = -1;
  for (int i = 0; i < BPL; ++i) {
    const int b = lane * BPL + i;
    if (b >= B) break;
#pragma unroll
    for (int c = 0; c < C; ++c) run[c] += loc[i][c];
    if (b < B - 1) {  // last bin cannot split
      const float hl = run[D], hr = parent[D] - run[D];
      const float cl = run[idx_c], cr = parent[idx_c] - run[idx_c];
      if (hl >= min_child_weight && hr >= min_child_weight &&
          cl >= min_instances && cr >= min_instances) {
        float gl = 0.0f, gr = 0.0f;
#pragma unroll
        for (int d = 0; d < C; ++d)
          if (d < D) {
            const float l = run[d], r = parent[d] - run[d];
            gl += l * l;
            gr += r * r;
          }
        const float gain =
            gl / (hl + lam) + gr / (hr + lam) - parent_score;
        const unsigned s = f32_sortable(gain);
        if (s > best_s) { best_s = s; best_b = b; }
      }
    }
  }
  // wave argmax
  for (int off = 32; off > 0; off >>= 1) {
    const unsigned os = __shfl_down(best_s, off, 64);
    const int ob = __shfl_down(best_b, off, 64);
    if (os > best_s) { best_s = os; best_b = ob; }
  }
  if (lane == 0 && best_b >= 0) {
    const unsigned long long packed =
        ((unsigned long long)best_s << 32) | (unsigned)(f * B + best_b);
    atomicMax(&best[node], packed);
  }
}

template <int C>
__global__ void split_decode_kernel(
    float* __restrict__ gain, int* __restrict__ feat, int* __restrict__ bin,
    float* __restrict__ left_stats,  // [n, C]
    const unsigned long long* __restrict__ best,
    const float* __restrict__ hist, int F, int B, float min_info_gain) {
  const int node = blockIdx.x;
  const unsigned long long p = best[node];
  const float g = sortable_f32((unsigned)(p >> 32));
  __shared__ float acc[C];
  if (threadIdx.x < C) acc[threadIdx.x] = 0.0f;
  __syncthreads();
  if (p == 0ull || !(g >= min_info_gain)) {
    if (threadIdx.x == 0) {
      gain[node] = -INFINITY;
      feat[node] = -1;
      bin[node] = -1;
    }
    if (threadIdx.x < C) left_stats[node * C + threadIdx.x] = 0.0f;
    return;
  }
  const int fb = (int)(p & 0xFFFFFFFFull);
  const int f = fb / B, b = fb % B;
  const float* h = hist + (((int64_t)node * F + f) * (int64_t)B) * C;
  float part[C];
#pragma unroll
  for (int c = 0; c < C; ++c) part[c] = 0.0f;
  for (int i = threadIdx.x; i <= b; i += blockDim.x)
#pragma unroll
    for (int c = 0; c < C; ++c) part[c] += h[(int64_t)i * C + c];
#pragma unroll
  for (int c = 0; c < C; ++c)
    if (part[c] != 0.0f) atomicAdd(&acc[c], part[c]);
  __syncthreads();
  if (threadIdx.x == 0) {
    gain[node] = g;
    feat[node] = f;
    bin[node] = b;
  }
  if (threadIdx.x < C) left_stats[node * C + threadIdx.x] = acc[threadIdx.x];
}

void split_argmax(torch::Tensor gain, torch::Tensor feat, torch::Tensor bin,
                  torch::Tensor left_stats, torch::Tensor hist, int64_t d_dims,
                  double lam, double min_child_weight, double min_instances,
                  double min_info_gain) {
  CHECK_GPU(hist); CHECK_CONTIG(hist);
  const int n = (int)hist.size(0);
  const int F = (int)hist.size(1);
  const int B = (int)hist.size(2);
  const int C = (int)hist.size(3);
  const int D = d_dims > 0 ? (int)d_dims : C - 2;
  TORCH_CHECK(C >= 2 && C <= 8, "split_argmax: 2 <= C <= 8");
  TORCH_CHECK(B >= 2 && B <= 256, "split_argmax: 2 <= B <= 256");
  TORCH_CHECK(D >= 1 && D < C, "split_argmax: bad D");
  auto best = torch::zeros({n}, hist.options().dtype(torch::kInt64));
  auto stream = at::hip::getCurrentHIPStream();
  const int fpb = 4;  // features (waves) per block
#define SA_LAUNCH(CC)                                                         \
  do {                                                                        \
    hipLaunchKernelGGL(split_argmax_kernel<CC>,                               \
                       dim3(n, (F + fpb - 1) / fpb), dim3(64 * fpb), 0,       \
                       stream,                                                \
                       (unsigned long long*)best.data_ptr<int64_t>(),         \
                       hist.data_ptr<float>(), F, B, D, (float)lam,           \
                       (float)min_child_weight, (float)min_instances);        \
    hipLaunchKernelGGL(split_decode_kernel<CC>, dim3(n), dim3(64), 0, stream, \
                       gain.data_ptr<float>(), feat.data_ptr<int>(),          \
                       bin.data_ptr<int>(), left_stats.data_ptr<float>(),     \
                       (unsigned long long*)best.data_ptr<int64_t>(),         \
                       hist.data_ptr<float>(), F, B, (float)min_info_gain);   \
  } while (0)
  switch (C) {
    case 2: SA_LAUNCH(2); break;
    case 3: SA_LAUNCH(3); break;
    case 4: SA_LAUNCH(4); break;
    case 5: SA_LAUNCH(5); break;
    case 6: SA_LAUNCH(6); break;
    case 7: SA_LAUNCH(7); break;
    case 8: SA_LAUNCH(8); break;
  }
#undef SA_LAUNCH
}

// csrc/linear.hip
void logreg_loss_grad(torch::Tensor payload, torch::Tensor x, torch::Tensor y,
                      torch::Tensor w, torch::Tensor wmat, bool has_bias);
bool logreg_fused_supported(int64_t F, int64_t K);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("logreg_loss_grad", &logreg_loss_grad,
        "single-pass fused logistic loss+gradient");
  m.def("logreg_fused_supported", &logreg_fused_supported);
  m.def("split_argmax", &split_argmax,
        "fused best-split gain scan + argmax over node histograms");
  m.def("sample_weights", &sample_weights, "counter-based Poisson/Bernoulli row weights");
  m.def("bin_features", &bin_features, "quantile binning f32 -> u8");
  m.def("hist_build", &hist_build, "LDS-staged node histograms");
  m.def("partition_rows", &partition_rows, "block-aggregated node partition");
  m.def("tree_predict", &tree_predict, "single-tree batched predict");
  m.def("forest_predict", &forest_predict, "packed-forest weighted predict");
  m.def("forest_predict2", &forest_predict2,
        "LDS-staged tree-group-tiled forest predict");
  m.def("transpose_u8", &transpose_u8, "tiled u8 matrix transpose");
  m.def("gather_ranges", &gather_ranges,
        "one-pass concat of contiguous index ranges");
  m.def("leaf_scatter", &leaf_scatter,
        "scatter per-segment leaf values into the [N, T] train-pred matrix");
  m.def("grad_hess", &grad_hess, "fused per-row loss gradient/hessian");
  m.def("line_search_eval", &line_search_eval, "fused loss+grad line-search payload");
  m.def("newton_chain_1d", &newton_chain_1d,
        "device-chained safeguarded-Newton stage-weight search");
}
