// Fused LayerNorm / RMSNorm forward + backward.
// One 64-lane wave per row, values register-resident (single HBM pass);
// bf16x8 vectorized loads (Guideline 13). Lane i owns column chunks
// i, i+64, ... so dγ/dβ accumulate in registers across the block's rows
// and flush with one atomicAdd per element at the end (Guideline 12).
//
// Replaces the implicit LayerNorm of the reference's GPT-2 forward/backward
// (transformers internals; SURVEY.md §2.2 op table).
#include "dta_common.h"
#include "dta_kernels.h"

namespace {

constexpr int ROW_WAVES = 4;  // rows processed concurrently per block

// ---- forward --------------------------------------------------------------
template <int ITERS, bool RMS>
__global__ void norm_fwd_k(const ushort* __restrict__ x,
                           const ushort* __restrict__ w,
                           const ushort* __restrict__ b,
                           ushort* __restrict__ y, float* __restrict__ mean,
                           float* __restrict__ rstd, int64_t rows, int cols,
                           float eps) {
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  const int nchunk = cols >> 3;  // bf16x8 chunks per row
  for (int64_t row = int64_t(blockIdx.x) * ROW_WAVES + wid; row < rows;
       row += int64_t(gridDim.x) * ROW_WAVES) {
    const ushort* xr = x + row * cols;
    float vals[ITERS][8];
    float sum = 0.f, sumsq = 0.f;
#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      const int c = lane + it * 64;
      if (c < nchunk) {
        s16x8 vx = *reinterpret_cast<const s16x8*>(xr + c * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = bf2f(ushort(vx[j]));
          vals[it][j] = f;
          sum += f;
          sumsq = fmaf(f, f, sumsq);
        }
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) vals[it][j] = 0.f;
      }
    }
    sum = wave_sum(sum);
    sumsq = wave_sum(sumsq);
    const float inv_n = 1.0f / cols;
    float mu = RMS ? 0.f : sum * inv_n;
    float var = sumsq * inv_n - (RMS ? 0.f : mu * mu);
    float rs = rsqrtf(var + eps);
    if (lane == 0) {
      if (!RMS) mean[row] = mu;
      rstd[row] = rs;
    }
    ushort* yr = y + row * cols;
#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      const int c = lane + it * 64;
      if (c < nchunk) {
        s16x8 vw = *reinterpret_cast<const s16x8*>(w + c * 8);
        s16x8 vb;
        if (!RMS) vb = *reinterpret_cast<const s16x8*>(b + c * 8);
        s16x8 vy;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float xh = (vals[it][j] - mu) * rs;
          float o = xh * bf2f(ushort(vw[j]));
          if (!RMS) o += bf2f(ushort(vb[j]));
          vy[j] = f2bf(o);
        }
        *reinterpret_cast<s16x8*>(yr + c * 8) = vy;
      }
    }
  }
}

// ---- backward -------------------------------------------------------------
// LN: xh=(x-mu)*rs; dx = rs*(dyw - mean(dyw) - xh*mean(dyw*xh)), dyw=dy*w
// RMS: xh=x*rs;      dx = rs*(dyw - xh*mean(dyw*xh))
// dw[c] += dy*xh ; db[c] += dy (LN only); accumulated over rows in regs.
template <int ITERS, bool RMS>
__global__ void norm_bwd_k(const ushort* __restrict__ dy,
                           const ushort* __restrict__ x,
                           const ushort* __restrict__ w,
                           const float* __restrict__ mean,
                           const float* __restrict__ rstd,
                           ushort* __restrict__ dx, float* __restrict__ dw,
                           float* __restrict__ db, int64_t rows, int cols) {
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  const int nchunk = cols >> 3;
  float dw_acc[ITERS][8];
  float db_acc[ITERS][8];
#pragma unroll
  for (int it = 0; it < ITERS; ++it)
#pragma unroll
    for (int j = 0; j < 8; ++j) { dw_acc[it][j] = 0.f; db_acc[it][j] = 0.f; }

  float wv[ITERS][8];
#pragma unroll
  for (int it = 0; it < ITERS; ++it) {
    const int c = lane + it * 64;
    if (c < nchunk) {
      s16x8 vw = *reinterpret_cast<const s16x8*>(w + c * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) wv[it][j] = bf2f(ushort(vw[j]));
    }
  }

  for (int64_t row = int64_t(blockIdx.x) * ROW_WAVES + wid; row < rows;
       row += int64_t(gridDim.x) * ROW_WAVES) {
    const ushort* xr = x + row * cols;
    const ushort* dyr = dy + row * cols;
    const float mu = RMS ? 0.f : mean[row];
    const float rs = rstd[row];
    float xh[ITERS][8], dyv[ITERS][8];
    float s1 = 0.f, s2 = 0.f;  // sum(dyw), sum(dyw*xh)
#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      const int c = lane + it * 64;
      if (c < nchunk) {
        s16x8 vx = *reinterpret_cast<const s16x8*>(xr + c * 8);
        s16x8 vdy = *reinterpret_cast<const s16x8*>(dyr + c * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float xhj = (bf2f(ushort(vx[j])) - mu) * rs;
          float dyj = bf2f(ushort(vdy[j]));
          float dywj = dyj * wv[it][j];
          xh[it][j] = xhj;
          dyv[it][j] = dyj;
          s1 += dywj;
          s2 = fmaf(dywj, xhj, s2);
        }
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) { xh[it][j] = 0.f; dyv[it][j] = 0.f; }
      }
    }
    s1 = wave_sum(s1) / cols;
    s2 = wave_sum(s2) / cols;
    ushort* dxr = dx + row * cols;
#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      const int c = lane + it * 64;
      if (c < nchunk) {
        s16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float dywj = dyv[it][j] * wv[it][j];
          float v = RMS ? (dywj - xh[it][j] * s2)
                        : (dywj - s1 - xh[it][j] * s2);
          o[j] = f2bf(rs * v);
          dw_acc[it][j] = fmaf(dyv[it][j], xh[it][j], dw_acc[it][j]);
          db_acc[it][j] += dyv[it][j];
        }
        *reinterpret_cast<s16x8*>(dxr + c * 8) = o;
      }
    }
  }
  // flush per-lane column accumulators (fp32 atomics, low contention)
#pragma unroll
  for (int it = 0; it < ITERS; ++it) {
    const int c = lane + it * 64;
    if (c < nchunk) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        atomicAdd(dw + c * 8 + j, dw_acc[it][j]);
        if (!RMS) atomicAdd(db + c * 8 + j, db_acc[it][j]);
      }
    }
  }
}

template <bool RMS>
void dispatch_fwd(const ushort* x, const ushort* w, const ushort* b,
                  ushort* y, float* mean, float* rstd, int64_t rows, int cols,
                  float eps, hipStream_t s) {
  const int nchunk = cols >> 3;
  const int iters = (nchunk + 63) / 64;
  int64_t want = (rows + ROW_WAVES - 1) / ROW_WAVES;
  const int grid = int(want < 4096 ? (want > 0 ? want : 1) : 4096);
  const dim3 blk(64 * ROW_WAVES);
#define CASE_F(I)                                                         \
  case I:                                                                 \
    norm_fwd_k<I, RMS><<<grid, blk, 0, s>>>(x, w, b, y, mean, rstd, rows, \
                                            cols, eps);                   \
    break;
  switch (iters) {
    CASE_F(1) CASE_F(2) CASE_F(3) CASE_F(4) CASE_F(6) CASE_F(8) CASE_F(16)
    default: {
      // generic fallback for odd sizes: round up to next supported
      if (iters <= 6) { norm_fwd_k<6, RMS><<<grid, blk, 0, s>>>(x, w, b, y, mean, rstd, rows, cols, eps); }
      else if (iters <= 8) { norm_fwd_k<8, RMS><<<grid, blk, 0, s>>>(x, w, b, y, mean, rstd, rows, cols, eps); }
      else { norm_fwd_k<16, RMS><<<grid, blk, 0, s>>>(x, w, b, y, mean, rstd, rows, cols, eps); }
    }
  }
#undef CASE_F
}

template <bool RMS>
void dispatch_bwd(const ushort* dy, const ushort* x, const ushort* w,
                  const float* mean, const float* rstd, ushort* dx, float* dw,
                  float* db, int64_t rows, int cols, hipStream_t s) {
  const int nchunk = cols >> 3;
  const int iters = (nchunk + 63) / 64;
  int64_t want = (rows + ROW_WAVES - 1) / ROW_WAVES;
  const int grid = int(want < 512 ? (want > 0 ? want : 1) : 512);
  const dim3 blk(64 * ROW_WAVES);
#define CASE_B(I)                                                          \
  case I:                                                                  \
    norm_bwd_k<I, RMS><<<grid, blk, 0, s>>>(dy, x, w, mean, rstd, dx, dw,  \
                                            db, rows, cols);               \
    break;
  switch (iters) {
    CASE_B(1) CASE_B(2) CASE_B(3) CASE_B(4) CASE_B(6) CASE_B(8)
    default:
      norm_bwd_k<8, RMS><<<grid, blk, 0, s>>>(dy, x, w, mean, rstd, dx, dw,
                                              db, rows, cols);
  }
#undef CASE_B
}

}  // namespace

void launch_layernorm_fwd(const bf16_t* x, const bf16_t* w, const bf16_t* b,
                          bf16_t* y, float* mean, float* rstd, int64_t rows,
                          int cols, float eps, hipStream_t s) {
  dispatch_fwd<false>(x, w, b, y, mean, rstd, rows, cols, eps, s);
}
void launch_layernorm_bwd(const bf16_t* dy, const bf16_t* x, const bf16_t* w,
                          const float* mean, const float* rstd, bf16_t* dx,
                          float* dw, float* db, int64_t rows, int cols,
                          hipStream_t s) {
  dispatch_bwd<false>(dy, x, w, mean, rstd, dx, dw, db, rows, cols, s);
}
void launch_rmsnorm_fwd(const bf16_t* x, const bf16_t* w, bf16_t* y,
                        float* rstd, int64_t rows, int cols, float eps,
                        hipStream_t s) {
  dispatch_fwd<true>(x, w, nullptr, y, nullptr, rstd, rows, cols, eps, s);
}
void launch_rmsnorm_bwd(const bf16_t* dy, const bf16_t* x, const bf16_t* w,
                        const float* rstd, bf16_t* dx, float* dw,
                        int64_t rows, int cols, hipStream_t s) {
  dispatch_bwd<true>(dy, x, w, nullptr, rstd, dx, dw, nullptr, rows, cols, s);
}
