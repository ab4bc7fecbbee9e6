// Probe #3: FG=64 (64-B row slices, 128 KiB LDS, 1 block/CU) vs FG=16.
//   hipcc --offload-arch=gfx950 -O3 -munsafe-fp-atomics tools/probe_hist3.hip -o probe_hist3
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdio>

#define HIP_CHECK(x) do { hipError_t e = (x); if (e) { printf("HIPERR %s @%d\n", hipGetErrorString(e), __LINE__); return 1; } } while (0)

constexpr int F = 256;
constexpr int B = 256;
constexpr int64_t N = 10'000'000;

template <int FG, int T, int CELLS>
__global__ __launch_bounds__(T) void probe_kernel(
    float* __restrict__ out, const uint8_t* __restrict__ bins,
    const float* __restrict__ gh, int rows_per_block) {
  extern __shared__ unsigned long long lds64[];  // FG * B * CELLS
  const int fg = blockIdx.y;
  const int f0 = fg * FG;
  const int64_t start = (int64_t)blockIdx.x * rows_per_block;
  const int64_t len = min((int64_t)rows_per_block, N - start);
  for (int i = threadIdx.x; i < FG * B * CELLS; i += blockDim.x) lds64[i] = 0ull;
  __syncthreads();
  for (int64_t i = threadIdx.x; i < len; i += blockDim.x) {
    const int64_t r = start + i;
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
        for (int j = 0; j < 4; ++j) {
          const int b = (w[qq] >> (8 * j)) & 0xff;
          unsigned long long* cell =
              lds64 + (((16 * q + qq * 4 + j) * B) + b) * CELLS;
#pragma unroll
          for (int c = 0; c < CELLS; ++c) atomicAdd(cell + c, packed);
        }
    }
  }
  __syncthreads();
  float* dst = out + ((int64_t)(blockIdx.x % 32) * F + f0) * B * 2;
  for (int i = threadIdx.x; i < FG * B; i += blockDim.x) {
    dst[2 * i] = (float)(int)(unsigned)(lds64[i * CELLS] >> 32);
    dst[2 * i + 1] = (float)(int)(unsigned)(lds64[i * CELLS] & 0xFFFFFFFFull);
  }
}

template <int FG, int T, int CELLS = 1>
float run(const char* name, uint8_t* bins, float* gh, float* out, int n_chunks) {
  int rows_per_block = (int)((N + n_chunks - 1) / n_chunks);
  dim3 grid(n_chunks, F / FG);
  size_t lds = (size_t)FG * B * CELLS * 8;
  hipError_t e = hipFuncSetAttribute(
      reinterpret_cast<const void*>(&probe_kernel<FG, T, CELLS>),
      hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);
  if (e) printf("(funcattr %s) ", hipGetErrorString(e));
  hipEvent_t a, b;
  hipEventCreate(&a); hipEventCreate(&b);
  hipLaunchKernelGGL((probe_kernel<FG, T, CELLS>), grid, dim3(T), lds, 0, out, bins, gh, rows_per_block);
  hipError_t le = hipGetLastError();
  if (le) { printf("%-36s launch failed: %s\n", name, hipGetErrorString(le)); return -1; }
  hipDeviceSynchronize();
  hipEventRecord(a);
  for (int it = 0; it < 3; ++it)
    hipLaunchKernelGGL((probe_kernel<FG, T, CELLS>), grid, dim3(T), lds, 0, out, bins, gh, rows_per_block);
  hipEventRecord(b);
  hipDeviceSynchronize();
  float ms;
  hipEventElapsedTime(&ms, a, b);
  ms /= 3;
  printf("%-36s %8.2f ms  %6.1f G bump/s\n", name, ms, (double)N * F / ms / 1e6);
  return ms;
}

__global__ void fill_kernel(uint8_t* bins, float* gh) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < N * F; i += stride) {
    unsigned h = (unsigned)(i * 2654435761u);
    bins[i] = (uint8_t)(h >> 24);
    if (i < N * 2) gh[i] = (float)(h & 0xff) / 255.0f;
  }
}

int main() {
  uint8_t* bins; float* gh; float* out;
  HIP_CHECK(hipMalloc(&bins, N * F));
  HIP_CHECK(hipMalloc(&gh, N * 2 * 4));
  HIP_CHECK(hipMalloc(&out, (int64_t)32 * F * B * 2 * 4));
  hipLaunchKernelGGL(fill_kernel, dim3(4096), dim3(256), 0, 0, bins, gh);
  HIP_CHECK(hipDeviceSynchronize());
  run<16, 256>("FG16 T256 (baseline), 768 chunks", bins, gh, out, 768 / 16);
  run<64, 1024>("FG64 T1024, 64 chunks", bins, gh, out, 64);
  printf("-- CELLS=2 (two u64 atomics per bump; the C=3/4 shape) --\n");
  run<16, 256, 2>("c2 FG16 T256 64KiB, 32 chunks", bins, gh, out, 32);
  run<16, 256, 2>("c2 FG16 T256 64KiB, 128 chunks", bins, gh, out, 128);
  run<16, 512, 2>("c2 FG16 T512 64KiB, 64 chunks", bins, gh, out, 64);
  run<32, 1024, 2>("c2 FG32 T1024 128KiB, 32 chunks", bins, gh, out, 32);
  run<32, 512, 2>("c2 FG32 T512 128KiB, 32 chunks", bins, gh, out, 32);
  run<16, 1024, 2>("c2 FG16 T1024 64KiB, 64 chunks", bins, gh, out, 64);
  printf("done\n");
  return 0;
}
