// Probe #4: row_idx indirection cost + software-pipelined prefetch.
//   hipcc --offload-arch=gfx950 -O3 -munsafe-fp-atomics tools/probe_hist4.hip -o probe_hist4
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdio>

#define HIP_CHECK(x) do { hipError_t e = (x); if (e) { printf("HIPERR %s @%d\n", hipGetErrorString(e), __LINE__); return 1; } } while (0)

constexpr int F = 256;
constexpr int B = 256;
constexpr int64_t N = 10'000'000;

// MODE 0: no row_idx (identity). MODE 1: row_idx gather (naive).
// MODE 2: row_idx gather, 2-stage software pipeline (prefetch next index+row ptr).
template <int FG, int T, int MODE>
__global__ __launch_bounds__(T) void probe_kernel(
    float* __restrict__ out, const uint8_t* __restrict__ bins,
    const float* __restrict__ gh, const int* __restrict__ row_idx,
    int rows_per_block) {
  extern __shared__ unsigned long long lds64[];
  const int fg = blockIdx.y;
  const int f0 = fg * FG;
  const int64_t start = (int64_t)blockIdx.x * rows_per_block;
  const int64_t len = min((int64_t)rows_per_block, N - start);
  for (int i = threadIdx.x; i < FG * B; i += blockDim.x) lds64[i] = 0ull;
  __syncthreads();

  if (MODE == 2) {
    int64_t i = threadIdx.x;
    if (i < len) {
      int r = row_idx[start + i];
      for (; i < len;) {
        const int64_t inext = i + blockDim.x;
        int rnext = 0;
        if (inext < len) rnext = row_idx[start + inext];  // prefetch
        const float* g = gh + (int64_t)r * 2;
        const unsigned long long packed =
            ((unsigned long long)(unsigned)__float2int_rn(g[0] * 65536.f) << 32) |
            (unsigned)__float2int_rn(g[1] * 65536.f);
        const uint8_t* br = bins + (int64_t)r * F + f0;
#pragma unroll
        for (int q = 0; q < FG / 16; ++q) {
          const uint4 bv = *reinterpret_cast<const uint4*>(br + 16 * q);
          const unsigned w[4] = {bv.x, bv.y, bv.z, bv.w};
#pragma unroll
          for (int qq = 0; qq < 4; ++qq)
#pragma unroll
            for (int j = 0; j < 4; ++j)
              atomicAdd(lds64 + ((16 * q + qq * 4 + j) * B) +
                            ((w[qq] >> (8 * j)) & 0xff),
                        packed);
        }
        i = inext;
        r = rnext;
      }
    }
  } else {
    for (int64_t i = threadIdx.x; i < len; i += blockDim.x) {
      const int64_t r = MODE == 0 ? start + i : row_idx[start + i];
      const float* g = gh + r * 2;
      const unsigned long long packed =
          ((unsigned long long)(unsigned)__float2int_rn(g[0] * 65536.f) << 32) |
          (unsigned)__float2int_rn(g[1] * 65536.f);
      const uint8_t* br = bins + r * F + f0;
#pragma unroll
      for (int q = 0; q < FG / 16; ++q) {
        const uint4 bv = *reinterpret_cast<const uint4*>(br + 16 * q);
        const unsigned w[4] = {bv.x, bv.y, bv.z, bv.w};
#pragma unroll
        for (int qq = 0; qq < 4; ++qq)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            atomicAdd(lds64 + ((16 * q + qq * 4 + j) * B) +
                          ((w[qq] >> (8 * j)) & 0xff),
                      packed);
      }
    }
  }
  __syncthreads();
  float* dst = out + ((int64_t)(blockIdx.x % 32) * F + f0) * B * 2;
  for (int i = threadIdx.x; i < FG * B; i += blockDim.x) {
    dst[2 * i] = (float)(int)(unsigned)(lds64[i] >> 32);
    dst[2 * i + 1] = (float)(int)(unsigned)(lds64[i] & 0xFFFFFFFFull);
  }
}

template <int FG, int T, int MODE>
float run(const char* name, uint8_t* bins, float* gh, int* rows, float* out,
          int n_chunks) {
  int rows_per_block = (int)((N + n_chunks - 1) / n_chunks);
  dim3 grid(n_chunks, F / FG);
  size_t lds = (size_t)FG * B * 8;
  (void)hipFuncSetAttribute(
      reinterpret_cast<const void*>(&probe_kernel<FG, T, MODE>),
      hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);
  hipEvent_t a, b;
  hipEventCreate(&a); hipEventCreate(&b);
  hipLaunchKernelGGL((probe_kernel<FG, T, MODE>), grid, dim3(T), lds, 0, out,
                     bins, gh, rows, rows_per_block);
  if (hipGetLastError()) { printf("%s launch failed\n", name); return -1; }
  hipDeviceSynchronize();
  hipEventRecord(a);
  for (int it = 0; it < 3; ++it)
    hipLaunchKernelGGL((probe_kernel<FG, T, MODE>), grid, dim3(T), lds, 0, out,
                       bins, gh, rows, rows_per_block);
  hipEventRecord(b);
  hipDeviceSynchronize();
  float ms;
  hipEventElapsedTime(&ms, a, b);
  ms /= 3;
  printf("%-42s %8.2f ms  %6.1f G bump/s\n", name, ms, (double)N * F / ms / 1e6);
  return ms;
}

__global__ void fill_kernel(uint8_t* bins, float* gh, int* rows_id,
                            int* rows_shuf) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < N * F; i += stride) {
    unsigned h = (unsigned)(i * 2654435761u);
    bins[i] = (uint8_t)(h >> 24);
    if (i < N * 2) gh[i] = (float)(h & 0xff) / 255.0f;
    if (i < N) {
      rows_id[i] = (int)i;
      // pseudo-shuffle: bit-reversal-ish permutation
      unsigned v = (unsigned)i;
      v = ((v >> 16) | (v << 16));
      rows_shuf[i] = (int)(((unsigned long long)v * 2654435761ull) % N);
    }
  }
}

int main() {
  uint8_t* bins; float* gh; float* out; int *rid, *rshuf;
  HIP_CHECK(hipMalloc(&bins, N * F));
  HIP_CHECK(hipMalloc(&gh, N * 2 * 4));
  HIP_CHECK(hipMalloc(&out, (int64_t)32 * F * B * 2 * 4));
  HIP_CHECK(hipMalloc(&rid, N * 4));
  HIP_CHECK(hipMalloc(&rshuf, N * 4));
  hipLaunchKernelGGL(fill_kernel, dim3(4096), dim3(256), 0, 0, bins, gh, rid, rshuf);
  HIP_CHECK(hipDeviceSynchronize());
  run<64, 1024, 0>("FG64 identity (no row_idx)", bins, gh, rid, out, 64);
  run<64, 1024, 1>("FG64 row_idx identity-values", bins, gh, rid, out, 64);
  run<64, 1024, 2>("FG64 row_idx identity pipelined", bins, gh, rid, out, 64);
  run<64, 1024, 1>("FG64 row_idx SHUFFLED", bins, gh, rshuf, out, 64);
  run<64, 1024, 2>("FG64 row_idx SHUFFLED pipelined", bins, gh, rshuf, out, 64);
  printf("done\n");
  return 0;
}
