// Fused LayerNorm / RMSNorm forward + backward.
//
// fwd: one 64-lane wave per row, values register-resident, bf16x8 loads
//      (Guideline 13); one HBM pass.
// bwd: split in two streaming kernels (measured: a fused version spent
//      ~95% of its time in a 2048-way atomicAdd flush of dγ/dβ):
//        * dx kernel — pure streaming, no reductions across rows;
//        * dγ/dβ kernel — column-parallel over ~256 row stripes,
//          ≤256 atomic adds per output address (Guideline 12).
//
// Replaces the implicit LayerNorm of the reference's GPT-2 fwd/bwd
// (transformers internals; SURVEY.md §2.2 op table).
#include "dta_common.h"
#include "dta_kernels.h"

namespace {

constexpr int ROW_WAVES = 4;

// ---- forward --------------------------------------------------------------
// RES: residual fusion — vals = round_bf16(x + res) (rounded BEFORE the
// statistics so the saved sum tensor reproduces them exactly in backward),
// and the sum is written out for the ongoing residual stream.
// DROP: counter-RNG dropout fused onto the INCOMING residual branch
// (transformers resid_pdrop: x = x + dropout(branch)) — elementwise
// chain over the branch's flat index, same draws as the standalone
// dropout kernel, so the host gold is ops.droprng.elem_keep_mask.
template <int ITERS, bool RMS, bool RES, bool DROP>
__global__ void norm_fwd_k(const ushort* __restrict__ x,
                           const ushort* __restrict__ res,
                           ushort* __restrict__ sum_out,
                           const ushort* __restrict__ w,
                           const ushort* __restrict__ b,
                           ushort* __restrict__ y, float* __restrict__ mean,
                           float* __restrict__ rstd, int64_t rows, int cols,
                           float eps,
                           const unsigned long long* __restrict__ rng,
                           unsigned long long site, unsigned int thr16,
                           float inv_keep) {
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  const int nchunk = cols >> 3;
  const uint64_t s1d = DROP ? sm64(*rng + site * DTA_RNG_SITE_K) : 0;
  for (int64_t row = int64_t(blockIdx.x) * ROW_WAVES + wid; row < rows;
       row += int64_t(gridDim.x) * ROW_WAVES) {
    const ushort* xr = x + row * cols;
    const ushort* rr = RES ? res + row * cols : nullptr;
    ushort* so = RES ? sum_out + row * cols : nullptr;
    float vals[ITERS][8];
    float sum = 0.f, sumsq = 0.f;
#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      const int c = lane + it * 64;
      if (c < nchunk) {
        s16x8 vx = *reinterpret_cast<const s16x8*>(xr + c * 8);
        s16x8 vr;
        if (RES) vr = *reinterpret_cast<const s16x8*>(rr + c * 8);
        uint64_t h0 = 0, h1 = 0;
        if (DROP) {
          const int64_t i0 = row * cols + int64_t(c) * 8;
          h0 = sm64(s1d + uint64_t(i0 >> 2) * DTA_RNG_IDX_K);
          h1 = sm64(s1d + uint64_t((i0 >> 2) + 1) * DTA_RNG_IDX_K);
        }
        s16x8 vs;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = bf2f(ushort(vx[j]));
          if (RES) {
            float rf = bf2f(ushort(vr[j]));
            if (DROP) {
              const uint64_t h = (j < 4) ? h0 : h1;
              rf = ((uint32_t(h >> (16 * (j & 3))) & 0xFFFF) >= thr16)
                       ? rf * inv_keep : 0.f;
            }
            const ushort sb = f2bf(f + rf);
            vs[j] = sb;
            f = bf2f(sb);
          }
          vals[it][j] = f;
          sum += f;
          sumsq = fmaf(f, f, sumsq);
        }
        if (RES) *reinterpret_cast<s16x8*>(so + c * 8) = vs;
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

// ---- backward dx (streaming, no cross-row state) --------------------------
// LN:  dx = rs*(dyw - mean(dyw) - xh*mean(dyw*xh)),  dyw = dy*w
// RMS: dx = rs*(dyw - xh*mean(dyw*xh))
// DS: residual fusion — dx += ds (the gradient arriving on the sum
// stream from its downstream consumer), saving the separate add kernel.
// DROP: the forward dropped the residual branch, so its gradient is the
// total sum-gradient gated by the regenerated mask: dres = dx ⊙ M/(1-p)
// — one extra streaming write here instead of a separate dropout-bwd
// pass over dx.
// FDW: dγ/dβ partials accumulated IN-REGISTER here (each lane owns fixed
// columns across all its rows) and spilled once per (block, wave) to the
// partial panel — removing the separate dwdb_part kernel's full re-read
// of dy and x (~0.9 ms/step at the flagship shape). Register cost is
// 16·ITERS VGPRs, so the fusion is gated to cols ≤ 1024 (the GPT-2
// family); wider rows keep the two-kernel path.
template <int ITERS, bool RMS, bool DS, bool DROP, bool FDW>
__global__ void norm_bwd_dx_k(const ushort* __restrict__ dy,
                              const ushort* __restrict__ ds,
                              const ushort* __restrict__ x,
                              const ushort* __restrict__ w,
                              const float* __restrict__ mean,
                              const float* __restrict__ rstd,
                              ushort* __restrict__ dx,
                              ushort* __restrict__ dres,
                              float* __restrict__ pdw,
                              float* __restrict__ pdb, int64_t rows,
                              int cols,
                              const unsigned long long* __restrict__ rng,
                              unsigned long long site, unsigned int thr16,
                              float inv_keep) {
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  const int nchunk = cols >> 3;
  const uint64_t s1d = DROP ? sm64(*rng + site * DTA_RNG_SITE_K) : 0;
  float dwp[FDW ? ITERS : 1][8], dbp[(FDW && !RMS) ? ITERS : 1][8];
  if (FDW) {
#pragma unroll
    for (int it = 0; it < (FDW ? ITERS : 1); ++it)
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        dwp[it][j] = 0.f;
        if (!RMS) dbp[it][j] = 0.f;
      }
  }
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
    float xh[ITERS][8], dyw[ITERS][8];
    float s1 = 0.f, s2 = 0.f;
#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      const int c = lane + it * 64;
      if (c < nchunk) {
        s16x8 vx = *reinterpret_cast<const s16x8*>(xr + c * 8);
        s16x8 vdy = *reinterpret_cast<const s16x8*>(dyr + c * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float xhj = (bf2f(ushort(vx[j])) - mu) * rs;
          const float dyv = bf2f(ushort(vdy[j]));
          float dywj = dyv * wv[it][j];
          xh[it][j] = xhj;
          dyw[it][j] = dywj;
          s1 += dywj;
          s2 = fmaf(dywj, xhj, s2);
          if (FDW) {
            dwp[it][j] = fmaf(dyv, xhj, dwp[it][j]);
            if (!RMS) dbp[it][j] += dyv;
          }
        }
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) { xh[it][j] = 0.f; dyw[it][j] = 0.f; }
      }
    }
    s1 = wave_sum(s1) / cols;
    s2 = wave_sum(s2) / cols;
    ushort* dxr = dx + row * cols;
    ushort* drr = DROP ? dres + row * cols : nullptr;
    const ushort* dsr = DS ? ds + row * cols : nullptr;
#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      const int c = lane + it * 64;
      if (c < nchunk) {
        s16x8 vds;
        if (DS) vds = *reinterpret_cast<const s16x8*>(dsr + c * 8);
        uint64_t h0 = 0, h1 = 0;
        if (DROP) {
          const int64_t i0 = row * cols + int64_t(c) * 8;
          h0 = sm64(s1d + uint64_t(i0 >> 2) * DTA_RNG_IDX_K);
          h1 = sm64(s1d + uint64_t((i0 >> 2) + 1) * DTA_RNG_IDX_K);
        }
        s16x8 o, od;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float v = RMS ? (dyw[it][j] - xh[it][j] * s2)
                        : (dyw[it][j] - s1 - xh[it][j] * s2);
          float g = rs * v;
          if (DS) g += bf2f(ushort(vds[j]));
          o[j] = f2bf(g);
          if (DROP) {
            const uint64_t h = (j < 4) ? h0 : h1;
            od[j] = ((uint32_t(h >> (16 * (j & 3))) & 0xFFFF) >= thr16)
                        ? f2bf(g * inv_keep) : ushort(0);
          }
        }
        *reinterpret_cast<s16x8*>(dxr + c * 8) = o;
        if (DROP) *reinterpret_cast<s16x8*>(drr + c * 8) = od;
      }
    }
  }
  if (FDW) {
    // combine the 4 waves in LDS (cols <= 1024 when FDW) and spill ONE
    // panel row per block — keeps the panel small enough for the
    // stripe-parallel reduce (a per-(block,wave) panel was 4x larger
    // and measured 6.6 ms/step in dwdb_reduce)
    __shared__ float lred[FDW ? 2048 : 1];
    float* lw = lred;
    float* lb = lred + 1024;
    for (int i = threadIdx.x; i < cols; i += blockDim.x) {
      lw[i] = 0.f;
      if (!RMS) lb[i] = 0.f;
    }
    __syncthreads();
#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      const int c = lane + it * 64;
      if (c < nchunk) {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          atomicAdd(&lw[c * 8 + j], dwp[it][j]);
          if (!RMS) atomicAdd(&lb[c * 8 + j], dbp[it][j]);
        }
      }
    }
    __syncthreads();
    float* pw = pdw + int64_t(blockIdx.x) * cols;
    float* pb = (!RMS) ? pdb + int64_t(blockIdx.x) * cols : nullptr;
    for (int i = threadIdx.x; i < cols; i += blockDim.x) {
      pw[i] = lw[i];
      if (!RMS) pb[i] = lb[i];
    }
  }
}

// ---- backward dγ/dβ (two-phase column reduction) --------------------------
// Phase 1: each (col-tile, stripe) block accumulates its stripe's rows into
// a [stripes, cols] fp32 partial panel (vectorized s16x8 row reads) — a
// single-phase version with ~1024 same-address atomicAdds per column
// measured 4x slower (per-address L2 serialization). Phase 2 reduces the
// panel 2D with <=32 atomics per output address.
template <bool RMS>
__global__ void norm_bwd_dwdb_part_k(const ushort* __restrict__ dy,
                                     const ushort* __restrict__ x,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ rstd,
                                     float* __restrict__ pdw,
                                     float* __restrict__ pdb, int64_t rows,
                                     int cols) {
  const int c8 = (blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (c8 >= cols) return;
  const int64_t r0 = (rows * blockIdx.y) / gridDim.y;
  const int64_t r1 = (rows * (blockIdx.y + 1)) / gridDim.y;
  float dw_acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  float db_acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  for (int64_t r = r0; r < r1; ++r) {
    const float mu = RMS ? 0.f : mean[r];
    const float rs = rstd[r];
    s16x8 xv = *reinterpret_cast<const s16x8*>(x + r * int64_t(cols) + c8);
    s16x8 dv = *reinterpret_cast<const s16x8*>(dy + r * int64_t(cols) + c8);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float dyv = bf2f(ushort(dv[j]));
      dw_acc[j] = fmaf(dyv, (bf2f(ushort(xv[j])) - mu) * rs, dw_acc[j]);
      db_acc[j] += dyv;
    }
  }
  float* pw = pdw + int64_t(blockIdx.y) * cols + c8;
#pragma unroll
  for (int j = 0; j < 8; ++j) pw[j] = dw_acc[j];
  if (!RMS) {
    float* pb = pdb + int64_t(blockIdx.y) * cols + c8;
#pragma unroll
    for (int j = 0; j < 8; ++j) pb[j] = db_acc[j];
  }
}

// 2D over (col tiles, stripe chunks); <=32 atomics per output address
// (a 1-block serial stripe loop measured 330 us — pure latency chain).
__global__ void dwdb_reduce_k(const float* __restrict__ pdw,
                              const float* __restrict__ pdb,
                              float* __restrict__ dw, float* __restrict__ db,
                              int stripes, int cols) {
  const int c4 = (blockIdx.x * blockDim.x + threadIdx.x) * 4;
  if (c4 >= cols) return;
  const int s0 = (stripes * blockIdx.y) / gridDim.y;
  const int s1 = (stripes * (blockIdx.y + 1)) / gridDim.y;
  f32x4 aw = {0.f, 0.f, 0.f, 0.f};
  f32x4 ab = {0.f, 0.f, 0.f, 0.f};
  for (int s_ = s0; s_ < s1; ++s_) {
    aw += *reinterpret_cast<const f32x4*>(pdw + int64_t(s_) * cols + c4);
    if (pdb)
      ab += *reinterpret_cast<const f32x4*>(pdb + int64_t(s_) * cols + c4);
  }
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    atomicAdd(dw + c4 + j, aw[j]);
    if (pdb) atomicAdd(db + c4 + j, ab[j]);
  }
}

template <bool RMS, bool RES, bool DROP>
void dispatch_fwd(const ushort* x, const ushort* res, ushort* sum_out,
                  const ushort* w, const ushort* b, ushort* y, float* mean,
                  float* rstd, int64_t rows, int cols, float eps,
                  const unsigned long long* rng, unsigned long long site,
                  unsigned int thr16, float ik, hipStream_t s) {
  const int nchunk = cols >> 3;
  const int iters = (nchunk + 63) / 64;
  int64_t want = (rows + ROW_WAVES - 1) / ROW_WAVES;
  const int grid = int(want < 4096 ? (want > 0 ? want : 1) : 4096);
  const dim3 blk(64 * ROW_WAVES);
#define CASE_F(I)                                                         \
  case I:                                                                 \
    norm_fwd_k<I, RMS, RES, DROP><<<grid, blk, 0, s>>>(                   \
        x, res, sum_out, w, b, y, mean, rstd, rows, cols, eps, rng, site, \
        thr16, ik);                                                       \
    break;
  switch (iters) {
    CASE_F(1) CASE_F(2) CASE_F(3) CASE_F(4) CASE_F(6) CASE_F(8) CASE_F(16)
    default: {
      if (iters <= 6) { norm_fwd_k<6, RMS, RES, DROP><<<grid, blk, 0, s>>>(x, res, sum_out, w, b, y, mean, rstd, rows, cols, eps, rng, site, thr16, ik); }
      else if (iters <= 8) { norm_fwd_k<8, RMS, RES, DROP><<<grid, blk, 0, s>>>(x, res, sum_out, w, b, y, mean, rstd, rows, cols, eps, rng, site, thr16, ik); }
      else { norm_fwd_k<16, RMS, RES, DROP><<<grid, blk, 0, s>>>(x, res, sum_out, w, b, y, mean, rstd, rows, cols, eps, rng, site, thr16, ik); }
    }
  }
#undef CASE_F
}

template <bool RMS, bool DS, bool DROP>
void dispatch_bwd(const ushort* dy, const ushort* ds, const ushort* x,
                  const ushort* w, const float* mean, const float* rstd,
                  ushort* dx, ushort* dres, float* dw, float* db, float* pdw,
                  float* pdb, int stripes, int64_t rows, int cols,
                  const unsigned long long* rng, unsigned long long site,
                  unsigned int thr16, float ik, hipStream_t s) {
  const int nchunk = cols >> 3;
  const int iters = (nchunk + 63) / 64;
  const bool fdw = dta_norm_fused_dwdb(cols);
  const int grid = dta_norm_bwd_grid(rows);
  const dim3 blk(64 * ROW_WAVES);
#define CASE_B(I)                                                            \
  case I:                                                                    \
    if (fdw)                                                                 \
      norm_bwd_dx_k<I, RMS, DS, DROP, true><<<grid, blk, 0, s>>>(            \
          dy, ds, x, w, mean, rstd, dx, dres, pdw, pdb, rows, cols, rng,     \
          site, thr16, ik);                                                  \
    else                                                                     \
      norm_bwd_dx_k<I, RMS, DS, DROP, false><<<grid, blk, 0, s>>>(           \
          dy, ds, x, w, mean, rstd, dx, dres, nullptr, nullptr, rows, cols,  \
          rng, site, thr16, ik);                                             \
    break;
  switch (iters) {
    CASE_B(1) CASE_B(2) CASE_B(3) CASE_B(4) CASE_B(6) CASE_B(8)
    default:
      norm_bwd_dx_k<8, RMS, DS, DROP, false><<<grid, blk, 0, s>>>(
          dy, ds, x, w, mean, rstd, dx, dres, nullptr, nullptr, rows, cols,
          rng, site, thr16, ik);
  }
#undef CASE_B
  if (!fdw) {
    const ColRedCfg cfg = dta_colred_cfg(rows, cols);
    dim3 g2(cfg.gx, unsigned(stripes));
    norm_bwd_dwdb_part_k<RMS><<<g2, cfg.threads, 0, s>>>(
        dy, x, mean, rstd, pdw, RMS ? nullptr : pdb, rows, cols);
  }
  const int g3 = (cols / 4 + 255) / 256;
  // stripe-parallelism scales with the panel height (the fused path
  // produces up to 4096 panel rows; 32 y-blocks left 128-deep serial
  // loops on an underfilled chip)
  const int ry = stripes < 32 ? stripes : (stripes > 1024 ? 256 : 32);
  dwdb_reduce_k<<<dim3(g3, ry), 256, 0, s>>>(pdw, RMS ? nullptr : pdb, dw,
                                             db, stripes, cols);
}

}  // namespace

void launch_layernorm_fwd(const bf16_t* x, const bf16_t* res,
                          bf16_t* sum_out, const bf16_t* w, const bf16_t* b,
                          bf16_t* y, float* mean, float* rstd, int64_t rows,
                          int cols, float eps, const unsigned long long* rng,
                          unsigned long long site, unsigned int thr16,
                          float ik, hipStream_t s) {
  if (res && thr16)
    dispatch_fwd<false, true, true>(x, res, sum_out, w, b, y, mean, rstd,
                                    rows, cols, eps, rng, site, thr16, ik,
                                    s);
  else if (res)
    dispatch_fwd<false, true, false>(x, res, sum_out, w, b, y, mean, rstd,
                                     rows, cols, eps, nullptr, 0, 0, 1.f,
                                     s);
  else
    dispatch_fwd<false, false, false>(x, nullptr, nullptr, w, b, y, mean,
                                      rstd, rows, cols, eps, nullptr, 0, 0,
                                      1.f, s);
}
void launch_layernorm_bwd(const bf16_t* dy, const bf16_t* ds,
                          const bf16_t* x, const bf16_t* w,
                          const float* mean, const float* rstd, bf16_t* dx,
                          bf16_t* dres, float* dw, float* db, float* pdw,
                          float* pdb, int stripes, int64_t rows, int cols,
                          const unsigned long long* rng,
                          unsigned long long site, unsigned int thr16,
                          float ik, hipStream_t s) {
  if (thr16) {
    if (ds)
      dispatch_bwd<false, true, true>(dy, ds, x, w, mean, rstd, dx, dres,
                                      dw, db, pdw, pdb, stripes, rows,
                                      cols, rng, site, thr16, ik, s);
    else
      dispatch_bwd<false, false, true>(dy, nullptr, x, w, mean, rstd, dx,
                                       dres, dw, db, pdw, pdb, stripes,
                                       rows, cols, rng, site, thr16, ik, s);
  } else if (ds)
    dispatch_bwd<false, true, false>(dy, ds, x, w, mean, rstd, dx, nullptr,
                                     dw, db, pdw, pdb, stripes, rows, cols,
                                     nullptr, 0, 0, 1.f, s);
  else
    dispatch_bwd<false, false, false>(dy, nullptr, x, w, mean, rstd, dx,
                                      nullptr, dw, db, pdw, pdb, stripes,
                                      rows, cols, nullptr, 0, 0, 1.f, s);
}
void launch_rmsnorm_fwd(const bf16_t* x, const bf16_t* res,
                        bf16_t* sum_out, const bf16_t* w, bf16_t* y,
                        float* rstd, int64_t rows, int cols, float eps,
                        const unsigned long long* rng,
                        unsigned long long site, unsigned int thr16,
                        float ik, hipStream_t s) {
  if (res && thr16)
    dispatch_fwd<true, true, true>(x, res, sum_out, w, nullptr, y, nullptr,
                                   rstd, rows, cols, eps, rng, site, thr16,
                                   ik, s);
  else if (res)
    dispatch_fwd<true, true, false>(x, res, sum_out, w, nullptr, y,
                                    nullptr, rstd, rows, cols, eps,
                                    nullptr, 0, 0, 1.f, s);
  else
    dispatch_fwd<true, false, false>(x, nullptr, nullptr, w, nullptr, y,
                                     nullptr, rstd, rows, cols, eps,
                                     nullptr, 0, 0, 1.f, s);
}
void launch_rmsnorm_bwd(const bf16_t* dy, const bf16_t* ds,
                        const bf16_t* x, const bf16_t* w,
                        const float* rstd, bf16_t* dx, bf16_t* dres,
                        float* dw, float* pdw, int stripes, int64_t rows,
                        int cols, const unsigned long long* rng,
                        unsigned long long site, unsigned int thr16,
                        float ik, hipStream_t s) {
  if (thr16) {
    if (ds)
      dispatch_bwd<true, true, true>(dy, ds, x, w, nullptr, rstd, dx, dres,
                                     dw, nullptr, pdw, nullptr, stripes,
                                     rows, cols, rng, site, thr16, ik, s);
    else
      dispatch_bwd<true, false, true>(dy, nullptr, x, w, nullptr, rstd, dx,
                                      dres, dw, nullptr, pdw, nullptr,
                                      stripes, rows, cols, rng, site,
                                      thr16, ik, s);
  } else if (ds)
    dispatch_bwd<true, true, false>(dy, ds, x, w, nullptr, rstd, dx,
                                    nullptr, dw, nullptr, pdw, nullptr,
                                    stripes, rows, cols, nullptr, 0, 0,
                                    1.f, s);
  else
    dispatch_bwd<true, false, false>(dy, nullptr, x, w, nullptr, rstd, dx,
                                     nullptr, dw, nullptr, pdw, nullptr,
                                     stripes, rows, cols, nullptr, 0, 0,
                                     1.f, s);
}
