// Fused log-softmax + cross-entropy over the vocab dimension, fwd + bwd.
// This is the dominant memory cost of GPT-2-small training (logits are
// [tokens, 50257] bf16). Replaces the reference's transformers-internal CE
// (`labels=` path, training_manager.py:380-385).
//
// fwd: one block per row, ONE online pass: per s16x8 vector take the local
//      max (cheap fmax tree), then one branch-free accumulator rescale
//      s = s*exp2(m_old−m_new) + Σ exp2(x_j−m_new) — 9 exp2 per 8 elements.
//      (A per-ELEMENT rescale branch measured 3x slower than two-pass; the
//      per-VECTOR unconditional rescale beats both: single HBM pass.)
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
    // online pass: TWO independent per-thread (m, s) streams (the single
    // running pair is a serial rescale chain — no ILP across vectors),
    // each reading a WAVE-CONTIGUOUS 1 KiB pass (stream B one full
    // block-sweep ahead — per-thread window widening shatters wave
    // coalescing, see ce_bwd_k)
    float m0 = -INFINITY, s0 = 0.f, m1 = -INFINITY, s1 = 0.f;
    const int64_t CB = int64_t(CE_BLOCK) * 8;
    int64_t i = int64_t(threadIdx.x) * 8;
    for (; i + CB + 8 <= vocab; i += 2 * CB) {
      s16x8 va = *reinterpret_cast<const s16x8*>(xr + i);
      s16x8 vb = *reinterpret_cast<const s16x8*>(xr + i + CB);
      float fa[8], fb[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        fa[j] = bf2f(ushort(va[j])) * LOG2E;
        fb[j] = bf2f(ushort(vb[j])) * LOG2E;
      }
      const float mxa = fmaxf(fmaxf(fmaxf(fa[0], fa[1]), fmaxf(fa[2], fa[3])),
                              fmaxf(fmaxf(fa[4], fa[5]), fmaxf(fa[6], fa[7])));
      const float mxb = fmaxf(fmaxf(fmaxf(fb[0], fb[1]), fmaxf(fb[2], fb[3])),
                              fmaxf(fmaxf(fb[4], fb[5]), fmaxf(fb[6], fb[7])));
      const float na = fmaxf(m0, mxa), nb = fmaxf(m1, mxb);
      float pa = 0.f, pb = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        pa += __builtin_exp2f(fa[j] - na);
        pb += __builtin_exp2f(fb[j] - nb);
      }
      s0 = s0 * __builtin_exp2f(m0 - na) + pa;
      s1 = s1 * __builtin_exp2f(m1 - nb) + pb;
      m0 = na;
      m1 = nb;
    }
    for (; i + 8 <= vocab; i += CB) {
      s16x8 va = *reinterpret_cast<const s16x8*>(xr + i);
      float fa[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) fa[j] = bf2f(ushort(va[j])) * LOG2E;
      const float mxa = fmaxf(fmaxf(fmaxf(fa[0], fa[1]), fmaxf(fa[2], fa[3])),
                              fmaxf(fmaxf(fa[4], fa[5]), fmaxf(fa[6], fa[7])));
      const float na = fmaxf(m0, mxa);
      float pa = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) pa += __builtin_exp2f(fa[j] - na);
      s0 = s0 * __builtin_exp2f(m0 - na) + pa;
      m0 = na;
    }
    if (i < vocab)
      for (; i < vocab; ++i) {
        const float f = bf2f(xr[i]) * LOG2E;
        const float m_new = fmaxf(m0, f);
        s0 = s0 * __builtin_exp2f(m0 - m_new) +
             __builtin_exp2f(f - m_new);
        m0 = m_new;
      }
    // merge the two streams (guard empty ones: exp2(-inf - -inf) is NaN
    // for threads whose slice starts past a small vocab)
    const float m_run = fmaxf(m0, m1);
    float s_run = 0.f;
    if (m0 != -INFINITY) s_run += s0 * __builtin_exp2f(m0 - m_run);
    if (m1 != -INFINITY) s_run += s1 * __builtin_exp2f(m1 - m_run);
    // combine the per-thread (m, s) pairs across the block
    const float M = block_max<16>(m_run, lds);
    const float s_adj = s_run * __builtin_exp2f(m_run - M);
    const float S = block_sum<16>(s_adj, lds);
    const float l = (M + __builtin_log2f(S)) * LN2;  // natural-log lse
    if (threadIdx.x == 0) {
      lse[row] = l;
      const int64_t tgt = targets[row];
      if (tgt != ignore_index) {
        atomicAdd(loss_sum, l - bf2f(xr[tgt]));
        atomicAdd(count, 1);
      }
    }
    __syncthreads();
  }
}

// scale_p: optional device pointer (dloss/count computed on device so the
// whole backward is hipGraph-capturable with no host sync); if null the
// host 'scale' scalar is used. v0: vocab-tile offset (the onehot column
// is targets[row] - v0). NT: nontemporal dlogits stores — right for the
// one-shot 6.5 GB pass (keeps the logits read stream L2-resident), wrong
// for the tiled pipeline (the tile is re-read by the two head GEMMs
// immediately, so it should STAY in cache).
template <bool NT>
__global__ void ce_bwd_k(const ushort* __restrict__ logits,
                         const int64_t* __restrict__ targets,
                         const float* __restrict__ lse, float scale,
                         const float* __restrict__ scale_p,
                         int64_t ignore_index, int64_t v0,
                         ushort* __restrict__ dlogits, int64_t rows,
                         int64_t vocab) {
  const float sc_base = scale_p ? *scale_p : scale;
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const ushort* xr = logits + row * vocab;
    ushort* dxr = dlogits + row * vocab;
    const int64_t tgt_raw = targets[row];
    const int64_t tgt = tgt_raw - v0;
    const float lb = lse[row] * LOG2E;
    const float sc = (tgt_raw == ignore_index) ? 0.f : sc_base;
    // two WAVE-CONTIGUOUS 1 KiB passes in flight per iteration: lane l
    // reads 16 B at 16·l — perfectly coalesced — and the second pass sits
    // one whole block-sweep (CB elements) ahead. (Widening the per-THREAD
    // window instead measured 2.3x SLOWER: 32-element windows stride the
    // wave's lanes 64 B apart and shatter coalescing.)
    const int64_t CB = int64_t(CE_BLOCK) * 8;
    int64_t i = int64_t(threadIdx.x) * 8;
    for (; i + CB + 8 <= vocab; i += 2 * CB) {
      s16x8 va = *reinterpret_cast<const s16x8*>(xr + i);
      s16x8 vb = *reinterpret_cast<const s16x8*>(xr + i + CB);
      s16x8 oa, ob;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float pa = __builtin_exp2f(bf2f(ushort(va[j])) * LOG2E - lb);
        float pb = __builtin_exp2f(bf2f(ushort(vb[j])) * LOG2E - lb);
        oa[j] = f2bf(sc * (pa - ((i + j) == tgt ? 1.f : 0.f)));
        ob[j] = f2bf(sc * (pb - ((i + CB + j) == tgt ? 1.f : 0.f)));
      }
      if (NT) {
        __builtin_nontemporal_store(oa, reinterpret_cast<s16x8*>(dxr + i));
        __builtin_nontemporal_store(ob,
                                    reinterpret_cast<s16x8*>(dxr + i + CB));
      } else {
        *reinterpret_cast<s16x8*>(dxr + i) = oa;
        *reinterpret_cast<s16x8*>(dxr + i + CB) = ob;
      }
    }
    for (; i + 8 <= vocab; i += CB) {
      s16x8 va = *reinterpret_cast<const s16x8*>(xr + i);
      s16x8 oa;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float pa = __builtin_exp2f(bf2f(ushort(va[j])) * LOG2E - lb);
        oa[j] = f2bf(sc * (pa - ((i + j) == tgt ? 1.f : 0.f)));
      }
      if (NT)
        __builtin_nontemporal_store(oa, reinterpret_cast<s16x8*>(dxr + i));
      else
        *reinterpret_cast<s16x8*>(dxr + i) = oa;
    }
    if (i < vocab)
      for (; i < vocab; ++i) {
        float p = __builtin_exp2f(bf2f(xr[i]) * LOG2E - lb);
        dxr[i] = f2bf(sc * (p - (i == tgt ? 1.f : 0.f)));
      }
  }
}

// ---- pipelined head-GEMM + CE forward -------------------------------------
// The LM-head logits ([tokens, 50k] bf16, ~6.5 GB at the flagship shape)
// are produced tile-by-tile by the head GEMM; consuming each tile's
// online-softmax contribution IMMEDIATELY after its GEMM (while the tile
// is L2/Infinity-Cache resident) removes the full-logits HBM re-read of
// the one-shot ce_fwd. State: per-row running (m, s) in exp2 units.
// targets/v0/tlogit: the tile holding a row's target column records the
// target logit into tlogit[row] (exactly one tile matches per row), so
// finalize never re-touches the logits.
__global__ void ce_chunk_k(const ushort* __restrict__ chunk, int64_t ld,
                           int64_t rows, int64_t cols,
                           const int64_t* __restrict__ targets, int64_t v0,
                           float* __restrict__ tlogit,
                           float* __restrict__ m_run,
                           float* __restrict__ s_run) {
  __shared__ float lds[16];
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const ushort* xr = chunk + row * ld;
    float m_l = -INFINITY, s_l = 0.f;
    int64_t i = int64_t(threadIdx.x) * 8;
    const int64_t stride = int64_t(CE_BLOCK) * 8;
    for (; i + 8 <= cols; i += stride) {
      s16x8 vx = *reinterpret_cast<const s16x8*>(xr + i);
      float f[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) f[j] = bf2f(ushort(vx[j])) * LOG2E;
      float mx = fmaxf(fmaxf(fmaxf(f[0], f[1]), fmaxf(f[2], f[3])),
                       fmaxf(fmaxf(f[4], f[5]), fmaxf(f[6], f[7])));
      const float m_new = fmaxf(m_l, mx);
      float ps = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) ps += __builtin_exp2f(f[j] - m_new);
      s_l = s_l * __builtin_exp2f(m_l - m_new) + ps;
      m_l = m_new;
    }
    if (i < cols && i + 8 > cols)
      for (; i < cols; ++i) {
        const float f = bf2f(xr[i]) * LOG2E;
        const float m_new = fmaxf(m_l, f);
        s_l = s_l * __builtin_exp2f(m_l - m_new) +
              __builtin_exp2f(f - m_new);
        m_l = m_new;
      }
    const float M = block_max<16>(m_l, lds);
    const float S = block_sum<16>(s_l * __builtin_exp2f(m_l - M), lds);
    if (threadIdx.x == 0) {
      const float m0 = m_run[row], s0 = s_run[row];
      const float mn = fmaxf(m0, M);
      // first chunk: s0 == 0 (the -inf m0 exp2 underflows to 0 cleanly)
      s_run[row] = s0 * __builtin_exp2f(m0 - mn) +
                   S * __builtin_exp2f(M - mn);
      m_run[row] = mn;
      const int64_t tc = targets[row] - v0;
      if (tc >= 0 && tc < cols) tlogit[row] = bf2f(xr[tc]);
    }
    __syncthreads();
  }
}

// lse[t] = ln-units lse from (m, s); loss_sum/count accumulated over
// non-ignored rows using the tlogit recorded by the chunks.
__global__ void ce_finalize_k(const int64_t* __restrict__ targets,
                              const float* __restrict__ m_run,
                              const float* __restrict__ s_run,
                              const float* __restrict__ tlogit, int64_t rows,
                              int64_t ignore_index,
                              float* __restrict__ lse,
                              float* __restrict__ loss_sum,
                              int* __restrict__ count) {
  float part = 0.f;
  int c = 0;
  for (int64_t t = int64_t(blockIdx.x) * blockDim.x + threadIdx.x; t < rows;
       t += int64_t(gridDim.x) * blockDim.x) {
    const float l = (m_run[t] + __builtin_log2f(s_run[t])) * LN2;
    lse[t] = l;
    const int64_t tg = targets[t];
    if (tg != ignore_index) {
      part += l - tlogit[t];
      ++c;
    }
  }
  part = wave_sum(part);
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) c += __shfl_xor(c, off);
  if ((threadIdx.x & 63) == 0) {
    atomicAdd(loss_sum, part);
    atomicAdd(count, c);
  }
}

}  // namespace

void launch_ce_chunk(const bf16_t* chunk, int64_t ld, int64_t rows,
                     int64_t cols, const int64_t* targets, int64_t v0,
                     float* tlogit, float* m_run, float* s_run,
                     hipStream_t s) {
  const int grid = int(rows < 4096 ? (rows > 0 ? rows : 1) : 4096);
  ce_chunk_k<<<grid, CE_BLOCK, 0, s>>>(chunk, ld, rows, cols, targets, v0,
                                       tlogit, m_run, s_run);
}

void launch_ce_finalize(const int64_t* targets, const float* m_run,
                        const float* s_run, const float* tlogit,
                        int64_t rows, int64_t ignore_index, float* lse,
                        float* loss_sum, int* count, hipStream_t s) {
  const int grid = elementwise_grid(rows, 256, 1);
  ce_finalize_k<<<grid, 256, 0, s>>>(targets, m_run, s_run, tlogit, rows,
                                     ignore_index, lse, loss_sum, count);
}

void launch_ce_fwd(const bf16_t* logits, const int64_t* targets, int64_t rows,
                   int64_t vocab, int64_t ignore_index, float* lse,
                   float* loss_sum, int* count, hipStream_t s) {
  const int grid = int(rows < 4096 ? (rows > 0 ? rows : 1) : 4096);
  ce_fwd_k<<<grid, CE_BLOCK, 0, s>>>(logits, targets, rows, vocab,
                                     ignore_index, lse, loss_sum, count);
}

void launch_ce_bwd(const bf16_t* logits, const int64_t* targets,
                   const float* lse, float scale, const float* scale_p,
                   int64_t ignore_index, int64_t v0, bool nt,
                   bf16_t* dlogits, int64_t rows, int64_t vocab,
                   hipStream_t s) {
  const int grid = int(rows < 4096 ? (rows > 0 ? rows : 1) : 4096);
  if (nt)
    ce_bwd_k<true><<<grid, CE_BLOCK, 0, s>>>(logits, targets, lse, scale,
                                             scale_p, ignore_index, v0,
                                             dlogits, rows, vocab);
  else
    ce_bwd_k<false><<<grid, CE_BLOCK, 0, s>>>(logits, targets, lse, scale,
                                              scale_p, ignore_index, v0,
                                              dlogits, rows, vocab);
}
