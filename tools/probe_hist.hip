// Standalone hist_build bottleneck probe (no torch): times kernel variants
// that isolate {global loads, LDS atomics, LDS writes} so the dominant cost
// is measured, not guessed.
//   hipcc --offload-arch=gfx950 -O3 -munsafe-fp-atomics tools/probe_hist.hip -o probe_hist
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdio>
#include <vector>

#define HIP_CHECK(x) do { hipError_t e = (x); if (e) { printf("HIPERR %s @%d\n", hipGetErrorString(e), __LINE__); return 1; } } while (0)

constexpr int F = 256;
constexpr int B = 256;
constexpr int C = 2;
constexpr int FG = 16;
constexpr int64_t N = 10'000'000;

// MODE: 0 full, 1 loads-only, 2 atomics-only (no loads), 3 ds_write (racy), 4 full+unroll4
template <int MODE>
__global__ void probe_kernel(float* __restrict__ out,
                             const uint8_t* __restrict__ bins,
                             const float* __restrict__ gh,
                             int rows_per_block) {
  __shared__ float lds[FG * B * C];
  const int fg = blockIdx.y;
  const int f0 = fg * FG;
  const int64_t start = (int64_t)blockIdx.x * rows_per_block;
  const int64_t len = min((int64_t)rows_per_block, N - start);

  for (int i = threadIdx.x; i < FG * B * C; i += blockDim.x) lds[i] = 0.0f;
  __syncthreads();

  float sink = 0.f;
  for (int64_t i = threadIdx.x; i < len; i += blockDim.x) {
    const int64_t r = start + i;
    float g0, g1;
    uint4 bv;
    if (MODE != 2) {
      const float* g = gh + r * C;
      g0 = g[0]; g1 = g[1];
      bv = *reinterpret_cast<const uint4*>(bins + r * F + f0);
    } else {
      unsigned h = (unsigned)(r * 2654435761u);
      bv = make_uint4(h, h * 97, h * 131, h * 181);
      g0 = 1.0f; g1 = 2.0f;
    }
    if (MODE == 1) {
      sink += g0 + g1 + (float)(bv.x ^ bv.y ^ bv.z ^ bv.w);
      continue;
    }
    const unsigned w[4] = {bv.x, bv.y, bv.z, bv.w};
#pragma unroll
    for (int q = 0; q < 4; ++q) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int b = (w[q] >> (8 * j)) & 0xff;
        float* cell = lds + (((q * 4 + j) * B) + b) * C;
        if (MODE == 3) {
          cell[0] = g0;
          cell[1] = g1;
        } else {
          atomicAdd(cell + 0, g0);
          atomicAdd(cell + 1, g1);
        }
      }
    }
  }
  __syncthreads();
  // flush (plain store; separate buffer per block to avoid atomics)
  float* dst = out + ((int64_t)(blockIdx.x % 64) * F + f0) * B * C;
  for (int i = threadIdx.x; i < FG * B * C; i += blockDim.x)
    dst[i] = lds[i] + sink;
}

template <int MODE>
float run(const char* name, uint8_t* bins, float* gh, float* out, int n_chunks) {
  int rows_per_block = (int)((N + n_chunks - 1) / n_chunks);
  dim3 grid(n_chunks, F / FG);
  hipEvent_t a, b;
  hipEventCreate(&a); hipEventCreate(&b);
  // warmup
  hipLaunchKernelGGL(probe_kernel<MODE>, grid, dim3(256), 0, 0, out, bins, gh, rows_per_block);
  hipDeviceSynchronize();
  hipEventRecord(a);
  for (int it = 0; it < 3; ++it)
    hipLaunchKernelGGL(probe_kernel<MODE>, grid, dim3(256), 0, 0, out, bins, gh, rows_per_block);
  hipEventRecord(b);
  hipDeviceSynchronize();
  float ms;
  hipEventElapsedTime(&ms, a, b);
  ms /= 3;
  printf("%-38s %8.2f ms\n", name, ms);
  return ms;
}

__global__ void fill_kernel(uint8_t* bins, float* gh) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < N * F; i += stride) {
    unsigned h = (unsigned)(i * 2654435761u);
    bins[i] = (uint8_t)(h >> 24);
    if (i < N * C) gh[i] = (float)(h & 0xff) / 255.0f;
  }
}

int main() {
  uint8_t* bins; float* gh; float* out;
  HIP_CHECK(hipMalloc(&bins, N * F));
  HIP_CHECK(hipMalloc(&gh, N * C * 4));
  HIP_CHECK(hipMalloc(&out, (int64_t)64 * F * B * C * 4));
  hipLaunchKernelGGL(fill_kernel, dim3(4096), dim3(256), 0, 0, bins, gh);
  HIP_CHECK(hipDeviceSynchronize());

  run<1>("loads only", bins, gh, out, 144);
  run<2>("atomics only (synthetic bins)", bins, gh, out, 144);
  run<3>("ds_write instead of atomic (racy)", bins, gh, out, 144);
  run<0>("full", bins, gh, out, 144);
  run<0>("full, 432 chunks", bins, gh, out, 432);
  run<0>("full, 48 chunks", bins, gh, out, 48);
  printf("done\n");
  return 0;
}
