// Fused log-softmax + cross-entropy over the vocab dimension, fwd + bwd.
// This is the dominant memory cost of GPT-2-small training (logits are
// [tokens, 50257] bf16); each pass streams the logits exactly once at
// HBM rate. Replaces the reference's transformers-internal CE
// (`labels=` path, training_manager.py:380-385).
//
// fwd: one block (4 waves) per row: online max+sumexp in one pass;
//      loss_row = lse - x_target; atomicAdd of the block's loss into
//      loss_sum; lse saved for backward.
// bwd: dlogits = scale_row * (softmax - onehot), one streaming pass.
#include "dta_common.h"
#include "dta_kernels.h"

namespace {

constexpr int CE_BLOCK = 256;

__global__ void ce_fwd_k(const ushort* __restrict__ logits,
                         const int64_t* __restrict__ targets, int64_t rows,
                         int64_t vocab, int64_t ignore_index,
                         float* __restrict__ lse, float* __restrict__ loss_sum,
                         int* __restrict__ count) {
  __shared__ float lds[16];
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const ushort* xr = logits + row * vocab;
    const int64_t tgt = targets[row];
    // online max + sumexp (per-thread, then block-combined)
    float m = -INFINITY, ssum = 0.f;
    int64_t i = int64_t(threadIdx.x) * 8;
    const int64_t stride = int64_t(CE_BLOCK) * 8;
    for (; i + 8 <= vocab; i += stride) {
      s16x8 vx = *reinterpret_cast<const s16x8*>(xr + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f(ushort(vx[j]));
        if (f > m) { ssum *= expf(m - f); m = f; }
        ssum += expf(f - m);
      }
    }
    if (i < vocab && i + 8 > vocab)
      for (; i < vocab; ++i) {
        float f = bf2f(xr[i]);
        if (f > m) { ssum *= expf(m - f); m = f; }
        ssum += expf(f - m);
      }
    // combine across block: M = max; S = sum of s_t * exp(m_t - M)
    float M = block_max<16>(m, lds);
    float S = block_sum<16>(ssum * expf(m - M), lds);
    float l = M + logf(S);
    if (threadIdx.x == 0) {
      lse[row] = l;
      if (tgt != ignore_index) {
        float xt = bf2f(xr[tgt]);
        atomicAdd(loss_sum, l - xt);
        atomicAdd(count, 1);
      }
    }
    __syncthreads();
  }
}

__global__ void ce_bwd_k(const ushort* __restrict__ logits,
                         const int64_t* __restrict__ targets,
                         const float* __restrict__ lse, float scale,
                         int64_t ignore_index, ushort* __restrict__ dlogits,
                         int64_t rows, int64_t vocab) {
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const ushort* xr = logits + row * vocab;
    ushort* dxr = dlogits + row * vocab;
    const int64_t tgt = targets[row];
    const float l = lse[row];
    const float sc = (tgt == ignore_index) ? 0.f : scale;
    int64_t i = int64_t(threadIdx.x) * 8;
    const int64_t stride = int64_t(CE_BLOCK) * 8;
    for (; i + 8 <= vocab; i += stride) {
      s16x8 vx = *reinterpret_cast<const s16x8*>(xr + i);
      s16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float p = expf(bf2f(ushort(vx[j])) - l);
        float g = sc * (p - ((i + j) == tgt ? 1.f : 0.f));
        o[j] = f2bf(g);
      }
      *reinterpret_cast<s16x8*>(dxr + i) = o;
    }
    if (i < vocab && i + 8 > vocab)
      for (; i < vocab; ++i) {
        float p = expf(bf2f(xr[i]) - l);
        dxr[i] = f2bf(sc * (p - (i == tgt ? 1.f : 0.f)));
      }
  }
}

}  // namespace

void launch_ce_fwd(const bf16_t* logits, const int64_t* targets, int64_t rows,
                   int64_t vocab, int64_t ignore_index, float* lse,
                   float* loss_sum, int* count, hipStream_t s) {
  const int grid = int(rows < 2048 ? (rows > 0 ? rows : 1) : 2048);
  ce_fwd_k<<<grid, CE_BLOCK, 0, s>>>(logits, targets, rows, vocab,
                                     ignore_index, lse, loss_sum, count);
}

void launch_ce_bwd(const bf16_t* logits, const int64_t* targets,
                   const float* lse, float scale, int64_t ignore_index,
                   bf16_t* dlogits, int64_t rows, int64_t vocab,
                   hipStream_t s) {
  const int grid = int(rows < 2048 ? (rows > 0 ? rows : 1) : 2048);
  ce_bwd_k<<<grid, CE_BLOCK, 0, s>>>(logits, targets, lse, scale,
                                     ignore_index, dlogits, rows, vocab);
}
