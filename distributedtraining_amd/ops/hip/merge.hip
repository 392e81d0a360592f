// Averager merge plane: fused multi-source weighted merge and the
// meta-gradient (grad_W) segmented reduction.
//
// Reference semantics being fused (averaging_logic.py):
//   merged[e]  = sum_i W[i, seg(e)] * (base[e] + delta_i[e])   (:422-470)
//   grad_W[i,j]= sum_{e in seg j} g[e] * (base[e]+delta_i[e]-merged[e])
//                                                              (:512-522)
// The reference re-loads every miner model from disk per batch; here all
// deltas are HBM-resident [N, P] fp32 and each pass streams them once.
//
// seg(e) is found by binary search over the LDS-cached offsets table
// (param-tensor boundaries; ~dozens to a few hundred entries).
#include "dta_common.h"
#include "dta_kernels.h"

namespace {

constexpr int MAX_SEGS = 1024;   // parameter tensors per model (GPT-2: 75)
constexpr int MAX_MODELS = 32;   // miners per node (8 GPUs typical)

DEV int find_seg(const int64_t* offs, int n_segs, int64_t e) {
  int lo = 0, hi = n_segs;  // invariant: offs[lo] <= e < offs[hi]
  while (hi - lo > 1) {
    int mid = (lo + hi) >> 1;
    if (e < offs[mid]) hi = mid; else lo = mid;
  }
  return lo;
}

// one pass over P; W cached in LDS ([N,S] fp32, N*S <= 32K floats)
__global__ void weighted_merge_k(const float* __restrict__ base,
                                 const float* __restrict__ deltas,
                                 const float* __restrict__ W,
                                 const int64_t* __restrict__ offsets,
                                 int n_models, int n_segs, int64_t P,
                                 float* __restrict__ out) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  int64_t* s_off = reinterpret_cast<int64_t*>(smem);
  float* s_w = reinterpret_cast<float*>(smem + (n_segs + 1) * sizeof(int64_t)
                                        + ((n_segs + 1) & 1) * 8);
  for (int i = threadIdx.x; i <= n_segs; i += blockDim.x)
    s_off[i] = offsets[i];
  for (int i = threadIdx.x; i < n_models * n_segs; i += blockDim.x)
    s_w[i] = W[i];
  __syncthreads();

  int64_t e = int64_t(blockIdx.x) * blockDim.x + threadIdx.x;
  const int64_t stride = int64_t(gridDim.x) * blockDim.x;
  for (; e < P; e += stride) {
    const int seg = find_seg(s_off, n_segs, e);
    const float b = base[e];
    float acc = 0.f;
    for (int i = 0; i < n_models; ++i)
      acc = fmaf(s_w[i * n_segs + seg], b + deltas[int64_t(i) * P + e], acc);
    out[e] = acc;
  }
}

// grad_W: grid = (chunks, n_models); each block reduces one aligned chunk
// (single segment) for one miner, then one atomicAdd into grad_w[i, seg].
// chunks: [n_chunks, 3] int64 rows (start, end, seg) built host-side from
// the offsets so no chunk crosses a segment boundary.
template <bool G_BF16>
__global__ void grad_w_k(const void* __restrict__ gp,
                         const float* __restrict__ base,
                         const float* __restrict__ deltas,
                         const float* __restrict__ merged,
                         const int64_t* __restrict__ chunks, int n_segs,
                         int64_t P, float* __restrict__ grad_w) {
  __shared__ float lds[16];
  const int64_t start = chunks[blockIdx.x * 3 + 0];
  const int64_t end = chunks[blockIdx.x * 3 + 1];
  const int seg = int(chunks[blockIdx.x * 3 + 2]);
  const int i = blockIdx.y;
  const float* di = deltas + int64_t(i) * P;
  float acc = 0.f;
  for (int64_t e = start + threadIdx.x; e < end; e += blockDim.x) {
    float g = G_BF16 ? bf2f(reinterpret_cast<const ushort*>(gp)[e])
                     : reinterpret_cast<const float*>(gp)[e];
    acc = fmaf(g, base[e] + di[e] - merged[e], acc);
  }
  acc = block_sum<16>(acc, lds);
  if (threadIdx.x == 0) atomicAdd(grad_w + int64_t(i) * n_segs + seg, acc);
}

}  // namespace

void launch_weighted_merge(const float* base, const float* deltas,
                           const float* W, const int64_t* offsets,
                           int n_models, int n_segs, int64_t P, float* out,
                           hipStream_t s) {
  const int block = 256;
  const int grid = elementwise_grid(P, block, 1);
  size_t shmem = (n_segs + 1) * sizeof(int64_t) + 8 +
                 size_t(n_models) * n_segs * sizeof(float);
  weighted_merge_k<<<grid, block, shmem, s>>>(base, deltas, W, offsets,
                                              n_models, n_segs, P, out);
}

void launch_grad_merge_weights(const void* g, bool g_is_bf16,
                               const float* base, const float* deltas,
                               const float* merged, const int64_t* chunks,
                               int n_models, int n_chunks, int64_t P,
                               float* grad_w, int n_segs, hipStream_t s) {
  dim3 grid(n_chunks, n_models);
  if (g_is_bf16)
    grad_w_k<true><<<grid, 256, 0, s>>>(g, base, deltas, merged, chunks,
                                        n_segs, P, grad_w);
  else
    grad_w_k<false><<<grid, 256, 0, s>>>(g, base, deltas, merged, chunks,
                                         n_segs, P, grad_w);
}
