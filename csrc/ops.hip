// spark_ensemble_amd gfx950 (CDNA4 / MI355X) kernels.
//
// Hand-written HIP for the framework's hot path (SURVEY.md section 2.7 maps
// each kernel to the reference semantics it replaces):
//   * hist_build      — LDS-staged per-(node, feature, bin) grad/hess/count
//                       histograms (replaces MLlib DecisionTree's
//                       treeAggregate histogram rounds)
//   * partition_rows  — single-pass two-ended node partition
//   * bin_features    — quantile binning (raw f32 -> uint8 bin ids)
//   * tree_predict /
//     forest_predict  — batched node-array tree walks (per-row model.predict
//                       loops of every ensemble model)
//   * sample_weights  — counter-based Poisson/Bernoulli row sampling
//                       (RDD.sample semantics as weight vectors)
//
// Design notes (MI355X_MICROARCH.md): 64-wide waves, 256-thread blocks;
// histograms live in LDS (dynamic, <= 64 KiB per block keeps 2 blocks/CU);
// global accumulation via device atomics once per block per bin; grids are
// (row-chunk x feature-group) so launches have >> 256 workgroups.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include <vector>

#define CHECK_GPU(x) TORCH_CHECK(x.is_cuda(), #x " must be on GPU")
#define CHECK_CONTIG(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")

static inline int64_t ceil_div(int64_t a, int64_t b) { return (a + b - 1) / b; }

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
  // upper 24 bits -> (0, 1]
  return ((z >> 40) + 1) * (1.0f / 16777216.0f);
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
// ---------------------------------------------------------------------------

__global__ void bin_features_kernel(uint8_t* __restrict__ out,
                                    const float* __restrict__ x,
                                    const float* __restrict__ edges, int64_t n,
                                    int f, int nedges) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t total = n * f;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < total; i += stride) {
    int fi = (int)(i % f);
    float v = x[i];
    const float* e = edges + (int64_t)fi * nedges;
    // first index with e[idx] >= v  (torch.searchsorted right=False)
    int lo = 0, hi = nedges;
    while (lo < hi) {
      int mid = (lo + hi) >> 1;
      if (e[mid] >= v) hi = mid; else lo = mid + 1;
    }
    out[i] = (uint8_t)lo;
  }
}

void bin_features(torch::Tensor out, torch::Tensor x, torch::Tensor edges) {
  CHECK_GPU(out); CHECK_GPU(x); CHECK_GPU(edges);
  CHECK_CONTIG(out); CHECK_CONTIG(x); CHECK_CONTIG(edges);
  int64_t n = x.size(0);
  int f = (int)x.size(1);
  int nedges = (int)edges.size(1);
  auto stream = at::hip::getCurrentHIPStream();
  int threads = 256;
  int blocks = (int)std::min<int64_t>(ceil_div(n * f, threads), 16384);
  hipLaunchKernelGGL(bin_features_kernel, dim3(blocks), dim3(threads), 0,
                     stream, out.data_ptr<uint8_t>(), x.data_ptr<float>(),
                     edges.data_ptr<float>(), n, f, nedges);
}

// ---------------------------------------------------------------------------
// hist_build
//   grid = (row_chunks, feature_groups); LDS histogram [FG][B][C] f32.
//   chunks: int32 [n_chunks, 3] = (node, start, len) segments of row_idx.
// ---------------------------------------------------------------------------

__global__ void hist_build_kernel(
    float* __restrict__ out,            // [n_nodes, F, B, C]
    const uint8_t* __restrict__ bins,   // [N, F]
    const float* __restrict__ gh,       // [N, C]
    const int* __restrict__ row_idx,    // [M]
    const int* __restrict__ chunks,     // [n_chunks, 3]
    int F, int B, int C, int FG) {
  extern __shared__ float lds[];  // FG * B * C
  const int chunk = blockIdx.x;
  const int fg = blockIdx.y;
  const int f0 = fg * FG;
  const int nf = min(FG, F - f0);
  const int node = chunks[chunk * 3 + 0];
  const int start = chunks[chunk * 3 + 1];
  const int len = chunks[chunk * 3 + 2];

  const int lds_size = FG * B * C;
  for (int i = threadIdx.x; i < lds_size; i += blockDim.x) lds[i] = 0.0f;
  __syncthreads();

  for (int i = threadIdx.x; i < len; i += blockDim.x) {
    const int r = row_idx[start + i];
    const float* g = gh + (int64_t)r * C;
    const uint8_t* br = bins + (int64_t)r * F + f0;
    for (int f = 0; f < nf; ++f) {
      const int b = br[f];
      float* cell = lds + ((f * B) + b) * C;
      for (int c = 0; c < C; ++c) {
        atomicAdd(cell + c, g[c]);
      }
    }
  }
  __syncthreads();

  // flush LDS -> global (atomic: several chunks may hit one node)
  float* dst = out + (((int64_t)node * F + f0) * B) * C;
  for (int i = threadIdx.x; i < nf * B * C; i += blockDim.x) {
    float v = lds[i];
    if (v != 0.0f) atomicAdd(dst + i, v);
  }
}

void hist_build(torch::Tensor out, torch::Tensor bins, torch::Tensor gh,
                torch::Tensor row_idx, torch::Tensor node_offsets,
                int64_t num_bins) {
  CHECK_GPU(out); CHECK_GPU(bins); CHECK_GPU(gh); CHECK_GPU(row_idx);
  CHECK_CONTIG(out); CHECK_CONTIG(bins); CHECK_CONTIG(gh); CHECK_CONTIG(row_idx);
  TORCH_CHECK(!node_offsets.is_cuda(), "node_offsets stays on host");
  const int F = (int)bins.size(1);
  const int B = (int)num_bins;
  const int C = (int)gh.size(1);

  // feature-group size: keep LDS <= 48 KiB so >= 3 blocks/CU stay resident
  int FG = std::max<int>(1, std::min<int>(F, 49152 / (B * C * 4)));
  const int n_groups = (int)ceil_div(F, FG);

  // chunk table on host
  const int64_t CHUNK = 16384;
  auto offs = node_offsets.accessor<int64_t, 1>();
  std::vector<int> chunk_v;
  const int n_nodes = (int)node_offsets.numel() - 1;
  for (int nd = 0; nd < n_nodes; ++nd) {
    int64_t s = offs[nd], e = offs[nd + 1];
    for (int64_t c = s; c < e; c += CHUNK) {
      chunk_v.push_back(nd);
      chunk_v.push_back((int)c);
      chunk_v.push_back((int)std::min<int64_t>(CHUNK, e - c));
    }
  }
  if (chunk_v.empty()) return;
  const int n_chunks = (int)(chunk_v.size() / 3);
  auto chunks = torch::from_blob(chunk_v.data(), {(int64_t)chunk_v.size()},
                                 torch::kInt32)
                    .to(bins.device(), /*non_blocking=*/false);

  auto stream = at::hip::getCurrentHIPStream();
  const size_t lds_bytes = (size_t)FG * B * C * 4;
  hipLaunchKernelGGL(hist_build_kernel, dim3(n_chunks, n_groups), dim3(256),
                     lds_bytes, stream, out.data_ptr<float>(),
                     bins.data_ptr<uint8_t>(), gh.data_ptr<float>(),
                     row_idx.data_ptr<int>(), chunks.data_ptr<int>(), F, B, C,
                     FG);
}

// ---------------------------------------------------------------------------
// partition_rows: single pass, two-ended (left fills up, right fills down)
// ---------------------------------------------------------------------------

__global__ void partition_kernel(
    int* __restrict__ new_rows,        // [M]
    int* __restrict__ cursors,         // [n_nodes, 2] = {lcur, rcur}
    const uint8_t* __restrict__ bins,  // [N, F]
    const int* __restrict__ row_idx,   // [M]
    const int* __restrict__ chunks,    // [n_chunks, 3]
    const int* __restrict__ feat,      // [n_nodes]
    const int* __restrict__ thr,       // [n_nodes]
    int F) {
  const int chunk = blockIdx.x;
  const int node = chunks[chunk * 3 + 0];
  const int start = chunks[chunk * 3 + 1];
  const int len = chunks[chunk * 3 + 2];
  const int f = feat[node];
  const int t = thr[node];
  for (int i = threadIdx.x; i < len; i += blockDim.x) {
    const int r = row_idx[start + i];
    bool left = (f < 0) || (bins[(int64_t)r * F + f] <= t);
    int pos;
    if (left) {
      pos = atomicAdd(cursors + node * 2 + 0, 1);
    } else {
      pos = atomicAdd(cursors + node * 2 + 1, -1) - 1;
    }
    new_rows[pos] = r;
  }
}

void partition_rows(torch::Tensor new_rows, torch::Tensor left_counts,
                    torch::Tensor bins, torch::Tensor row_idx,
                    torch::Tensor node_offsets, torch::Tensor feat,
                    torch::Tensor thr) {
  CHECK_GPU(new_rows); CHECK_GPU(bins); CHECK_GPU(row_idx);
  CHECK_GPU(feat); CHECK_GPU(thr); CHECK_GPU(left_counts);
  const int F = (int)bins.size(1);
  const int n_nodes = (int)node_offsets.numel() - 1;
  auto offs = node_offsets.accessor<int64_t, 1>();

  std::vector<int> cur_v(n_nodes * 2);
  std::vector<int> chunk_v;
  const int64_t CHUNK = 16384;
  for (int nd = 0; nd < n_nodes; ++nd) {
    cur_v[nd * 2 + 0] = (int)offs[nd];
    cur_v[nd * 2 + 1] = (int)offs[nd + 1];
    for (int64_t c = offs[nd]; c < offs[nd + 1]; c += CHUNK) {
      chunk_v.push_back(nd);
      chunk_v.push_back((int)c);
      chunk_v.push_back((int)std::min<int64_t>(CHUNK, offs[nd + 1] - c));
    }
  }
  auto stream = at::hip::getCurrentHIPStream();
  auto cursors = torch::from_blob(cur_v.data(), {n_nodes * 2}, torch::kInt32)
                     .to(bins.device());
  if (!chunk_v.empty()) {
    auto chunks = torch::from_blob(chunk_v.data(), {(int64_t)chunk_v.size()},
                                   torch::kInt32)
                      .to(bins.device());
    const int n_chunks = (int)(chunk_v.size() / 3);
    hipLaunchKernelGGL(partition_kernel, dim3(n_chunks), dim3(256), 0, stream,
                       new_rows.data_ptr<int>(), cursors.data_ptr<int>(),
                       bins.data_ptr<uint8_t>(), row_idx.data_ptr<int>(),
                       chunks.data_ptr<int>(), feat.data_ptr<int>(),
                       thr.data_ptr<int>(), F);
  }
  // left_counts[nd] = final lcur - seg_start
  auto lcur = cursors.view({n_nodes, 2}).select(1, 0);
  auto seg_start =
      torch::from_blob(cur_v.data(), {n_nodes, 2}, torch::kInt32)
          .select(1, 0)
          .clone()
          .to(bins.device());
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

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("sample_weights", &sample_weights, "counter-based Poisson/Bernoulli row weights");
  m.def("bin_features", &bin_features, "quantile binning f32 -> u8");
  m.def("hist_build", &hist_build, "LDS-staged node histograms");
  m.def("partition_rows", &partition_rows, "two-ended node partition");
  m.def("tree_predict", &tree_predict, "single-tree batched predict");
  m.def("forest_predict", &forest_predict, "packed-forest weighted predict");
}
