// Elementwise kernels: GELU (tanh approx), SwiGLU, flat-plane delta ops,
// counter-based dropout. Covers the implicit elementwise surface of the
// reference's transformers GPT-2 step (GELU/dropout inside
// `self.model(...)`, training_manager.py:380-385) and its flat-plane
// tensor loops (delta = theta - base, training_manager.py:417-421;
// apply theta += delta, validation_logic.py:251-259).
// All memory-bound: vectorized bf16x8 / float4 loads (Guideline 13),
// grid-stride with capped grid (Guideline 11).
#include "dta_common.h"
#include "dta_kernels.h"

namespace {

constexpr float GELU_C = 0.7978845608028654f;   // sqrt(2/pi)
constexpr float GELU_A = 0.044715f;

DEV float gelu_f(float x) {
  float u = GELU_C * (x + GELU_A * x * x * x);
  return 0.5f * x * (1.0f + fast_tanh(u));
}
DEV float gelu_df(float x) {
  float x2 = x * x;
  float u = GELU_C * x * (1.0f + GELU_A * x2);
  float t = fast_tanh(u);
  float sech2 = 1.0f - t * t;
  return 0.5f * (1.0f + t) + 0.5f * x * sech2 * GELU_C * (1.0f + 3.0f * GELU_A * x2);
}
DEV float silu_f(float x) { return x / (1.0f + fast_exp(-x)); }
DEV float silu_df(float x) {
  float s = 1.0f / (1.0f + fast_exp(-x));
  return s * (1.0f + x * (1.0f - s));
}

// -- generic vectorized 1-in-1-out bf16 map ---------------------------------
// two 16 B vectors per iteration: the transcendental-heavy bodies (gelu)
// measured 1.5x memory SOL with one — the second load covers the latency
template <float (*F)(float)>
__global__ void map_bf16_k(const ushort* __restrict__ x, ushort* __restrict__ y,
                           int64_t n) {
  int64_t i = (int64_t(blockIdx.x) * blockDim.x + threadIdx.x) * 16;
  const int64_t stride = int64_t(gridDim.x) * blockDim.x * 16;
  for (; i + 16 <= n; i += stride) {
    s16x8 va = *reinterpret_cast<const s16x8*>(x + i);
    s16x8 vb = *reinterpret_cast<const s16x8*>(x + i + 8);
    s16x8 oa, ob;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      oa[j] = f2bf(F(bf2f(ushort(va[j]))));
      ob[j] = f2bf(F(bf2f(ushort(vb[j]))));
    }
    *reinterpret_cast<s16x8*>(y + i) = oa;
    *reinterpret_cast<s16x8*>(y + i + 8) = ob;
  }
  if (i < n)
    for (; i < n; ++i) y[i] = f2bf(F(bf2f(x[i])));
}

template <float (*F)(float)>
__global__ void map_f32_k(const float* __restrict__ x, float* __restrict__ y,
                          int64_t n) {
  int64_t i = (int64_t(blockIdx.x) * blockDim.x + threadIdx.x) * 4;
  const int64_t stride = int64_t(gridDim.x) * blockDim.x * 4;
  for (; i + 4 <= n; i += stride) {
    f32x4 vx = *reinterpret_cast<const f32x4*>(x + i);
    f32x4 vy;
#pragma unroll
    for (int j = 0; j < 4; ++j) vy[j] = F(vx[j]);
    *reinterpret_cast<f32x4*>(y + i) = vy;
  }
  if (i < n && i + 4 > n)
    for (; i < n; ++i) y[i] = F(x[i]);
}

// dx = dy * f'(x); two vectors per iteration (see map_bf16_k)
template <float (*DF)(float)>
__global__ void map_grad_bf16_k(const ushort* __restrict__ dy,
                                const ushort* __restrict__ x,
                                ushort* __restrict__ dx, int64_t n) {
  int64_t i = (int64_t(blockIdx.x) * blockDim.x + threadIdx.x) * 16;
  const int64_t stride = int64_t(gridDim.x) * blockDim.x * 16;
  for (; i + 16 <= n; i += stride) {
    s16x8 da = *reinterpret_cast<const s16x8*>(dy + i);
    s16x8 db = *reinterpret_cast<const s16x8*>(dy + i + 8);
    s16x8 xa = *reinterpret_cast<const s16x8*>(x + i);
    s16x8 xb = *reinterpret_cast<const s16x8*>(x + i + 8);
    s16x8 oa, ob;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      oa[j] = f2bf(bf2f(ushort(da[j])) * DF(bf2f(ushort(xa[j]))));
      ob[j] = f2bf(bf2f(ushort(db[j])) * DF(bf2f(ushort(xb[j]))));
    }
    *reinterpret_cast<s16x8*>(dx + i) = oa;
    *reinterpret_cast<s16x8*>(dx + i + 8) = ob;
  }
  if (i < n)
    for (; i < n; ++i) dx[i] = f2bf(bf2f(dy[i]) * DF(bf2f(x[i])));
}

template <float (*DF)(float)>
__global__ void map_grad_f32_k(const float* __restrict__ dy,
                               const float* __restrict__ x,
                               float* __restrict__ dx, int64_t n) {
  int64_t i = int64_t(blockIdx.x) * blockDim.x + threadIdx.x;
  const int64_t stride = int64_t(gridDim.x) * blockDim.x;
  for (; i < n; i += stride) dx[i] = dy[i] * DF(x[i]);
}

__global__ void swiglu_fwd_k(const ushort* __restrict__ g,
                             const ushort* __restrict__ u,
                             ushort* __restrict__ y, int64_t n) {
  int64_t i = (int64_t(blockIdx.x) * blockDim.x + threadIdx.x) * 8;
  const int64_t stride = int64_t(gridDim.x) * blockDim.x * 8;
  for (; i + 8 <= n; i += stride) {
    s16x8 vg = *reinterpret_cast<const s16x8*>(g + i);
    s16x8 vu = *reinterpret_cast<const s16x8*>(u + i);
    s16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = f2bf(silu_f(bf2f(ushort(vg[j]))) * bf2f(ushort(vu[j])));
    *reinterpret_cast<s16x8*>(y + i) = o;
  }
  if (i < n && i + 8 > n)
    for (; i < n; ++i) y[i] = f2bf(silu_f(bf2f(g[i])) * bf2f(u[i]));
}

__global__ void swiglu_bwd_k(const ushort* __restrict__ dy,
                             const ushort* __restrict__ g,
                             const ushort* __restrict__ u,
                             ushort* __restrict__ dg, ushort* __restrict__ du,
                             int64_t n) {
  int64_t i = (int64_t(blockIdx.x) * blockDim.x + threadIdx.x) * 8;
  const int64_t stride = int64_t(gridDim.x) * blockDim.x * 8;
  for (; i + 8 <= n; i += stride) {
    s16x8 vdy = *reinterpret_cast<const s16x8*>(dy + i);
    s16x8 vg = *reinterpret_cast<const s16x8*>(g + i);
    s16x8 vu = *reinterpret_cast<const s16x8*>(u + i);
    s16x8 og, ou;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float fdy = bf2f(ushort(vdy[j])), fg = bf2f(ushort(vg[j])),
            fu = bf2f(ushort(vu[j]));
      og[j] = f2bf(fdy * fu * silu_df(fg));
      ou[j] = f2bf(fdy * silu_f(fg));
    }
    *reinterpret_cast<s16x8*>(dg + i) = og;
    *reinterpret_cast<s16x8*>(du + i) = ou;
  }
  if (i < n && i + 8 > n)
    for (; i < n; ++i) {
      float fdy = bf2f(dy[i]), fg = bf2f(g[i]), fu = bf2f(u[i]);
      dg[i] = f2bf(fdy * fu * silu_df(fg));
      du[i] = f2bf(fdy * silu_f(fg));
    }
}

// -- flat-plane: delta = w - base; w += a*x; nan scan; sq-norm --------------
__global__ void delta_sub_k(const float* __restrict__ w,
                            const float* __restrict__ base,
                            float* __restrict__ out, int64_t n) {
  int64_t i = (int64_t(blockIdx.x) * blockDim.x + threadIdx.x) * 4;
  const int64_t stride = int64_t(gridDim.x) * blockDim.x * 4;
  for (; i + 4 <= n; i += stride) {
    f32x4 a = *reinterpret_cast<const f32x4*>(w + i);
    f32x4 b = *reinterpret_cast<const f32x4*>(base + i);
    f32x4 o = {a[0] - b[0], a[1] - b[1], a[2] - b[2], a[3] - b[3]};
    *reinterpret_cast<f32x4*>(out + i) = o;
  }
  if (i < n && i + 4 > n)
    for (; i < n; ++i) out[i] = w[i] - base[i];
}

__global__ void axpy_k(float* __restrict__ w, const float* __restrict__ x,
                       float alpha, int64_t n) {
  int64_t i = (int64_t(blockIdx.x) * blockDim.x + threadIdx.x) * 4;
  const int64_t stride = int64_t(gridDim.x) * blockDim.x * 4;
  for (; i + 4 <= n; i += stride) {
    f32x4 a = *reinterpret_cast<const f32x4*>(w + i);
    f32x4 b = *reinterpret_cast<const f32x4*>(x + i);
    f32x4 o = {fmaf(alpha, b[0], a[0]), fmaf(alpha, b[1], a[1]),
               fmaf(alpha, b[2], a[2]), fmaf(alpha, b[3], a[3])};
    *reinterpret_cast<f32x4*>(w + i) = o;
  }
  if (i < n && i + 4 > n)
    for (; i < n; ++i) w[i] = fmaf(alpha, x[i], w[i]);
}

__global__ void nan_any_k(const float* __restrict__ x, int64_t n,
                          int* __restrict__ flag) {
  int64_t i = (int64_t(blockIdx.x) * blockDim.x + threadIdx.x) * 4;
  const int64_t stride = int64_t(gridDim.x) * blockDim.x * 4;
  bool bad = false;
  for (; i + 4 <= n; i += stride) {
    f32x4 a = *reinterpret_cast<const f32x4*>(x + i);
    bad |= (isnan(a[0]) || isnan(a[1]) || isnan(a[2]) || isnan(a[3]));
  }
  if (i < n && i + 4 > n)
    for (; i < n; ++i) bad |= isnan(x[i]);
  if (__any(bad) && (threadIdx.x & 63) == 0) atomicOr(flag, 1);
}

__global__ void l2norm_sq_k(const float* __restrict__ x, int64_t n,
                            float* __restrict__ out) {
  __shared__ float lds[16];
  int64_t i = (int64_t(blockIdx.x) * blockDim.x + threadIdx.x) * 4;
  const int64_t stride = int64_t(gridDim.x) * blockDim.x * 4;
  float acc = 0.f;
  for (; i + 4 <= n; i += stride) {
    f32x4 a = *reinterpret_cast<const f32x4*>(x + i);
    acc = fmaf(a[0], a[0], acc); acc = fmaf(a[1], a[1], acc);
    acc = fmaf(a[2], a[2], acc); acc = fmaf(a[3], a[3], acc);
  }
  if (i < n && i + 4 > n)
    for (; i < n; ++i) acc = fmaf(x[i], x[i], acc);
  acc = block_sum<16>(acc, lds);
  if (threadIdx.x == 0) atomicAdd(out, acc);
}

__global__ void f32_to_bf16_k(const float* __restrict__ x,
                              ushort* __restrict__ y, int64_t n) {
  int64_t i = (int64_t(blockIdx.x) * blockDim.x + threadIdx.x) * 8;
  const int64_t stride = int64_t(gridDim.x) * blockDim.x * 8;
  for (; i + 8 <= n; i += stride) {
    f32x4 a = *reinterpret_cast<const f32x4*>(x + i);
    f32x4 b = *reinterpret_cast<const f32x4*>(x + i + 4);
    s16x8 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) { o[j] = f2bf(a[j]); o[4 + j] = f2bf(b[j]); }
    *reinterpret_cast<s16x8*>(y + i) = o;
  }
  if (i < n && i + 8 > n)
    for (; i < n; ++i) y[i] = f2bf(x[i]);
}


// ---- column sum (bias gradient) -------------------------------------------
// out[c] = sum_r x[r,c] for x [rows, cols] bf16. Two-phase, deterministic,
// atomic-free: phase 1 writes per-stripe partials (a single-kernel atomic
// version measured 230 us at [32k,768] -- ~1000 same-address fp32 atomics
// per column serialize in L2); phase 2 reduces the [stripes, cols] panel.
// Replaces torch's generic reduce_kernel for dbias (48 calls/step on GPT-2).
__global__ void colsum_part_k(const ushort* __restrict__ x,
                              float* __restrict__ part, int64_t rows,
                              int cols) {
  const int c8 = (blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (c8 >= cols) return;
  const int64_t r0 = (rows * blockIdx.y) / gridDim.y;
  const int64_t r1 = (rows * (blockIdx.y + 1)) / gridDim.y;
  float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  for (int64_t r = r0; r < r1; ++r) {
    s16x8 vx = *reinterpret_cast<const s16x8*>(x + r * int64_t(cols) + c8);
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] += bf2f(ushort(vx[j]));
  }
  float* p = part + int64_t(blockIdx.y) * cols + c8;
#pragma unroll
  for (int j = 0; j < 8; ++j) p[j] = acc[j];
}

// 2D: blockIdx.y owns a chunk of stripes (a 1-block serial loop over 1024
// stripes measured 190 us — pure latency chain); <=32 atomics per output
// address is noise.
__global__ void colsum_reduce_k(const float* __restrict__ part,
                                float* __restrict__ out, int stripes,
                                int cols) {
  const int c4 = (blockIdx.x * blockDim.x + threadIdx.x) * 4;
  if (c4 >= cols) return;
  const int s0 = (stripes * blockIdx.y) / gridDim.y;
  const int s1 = (stripes * (blockIdx.y + 1)) / gridDim.y;
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  for (int s_ = s0; s_ < s1; ++s_)
    acc += *reinterpret_cast<const f32x4*>(part + int64_t(s_) * cols + c4);
#pragma unroll
  for (int j = 0; j < 4; ++j) atomicAdd(out + c4 + j, acc[j]);
}

// -- counter-based dropout (residual/embedding paths; transformers GPT-2
// resid_pdrop/embd_pdrop). One kernel serves fwd (x) and bwd (dy): both
// are y = x * mask * inv_keep with the mask regenerated from
// (counter, site, index) — dta_common.h RNG chain. 8 elements per 16 B
// vector = two sm64 hashes (4 draws each).
__global__ void dropout_k(const ushort* __restrict__ x,
                          ushort* __restrict__ y, int64_t n,
                          const unsigned long long* __restrict__ rng,
                          unsigned long long site, unsigned int thr16,
                          float inv_keep) {
  const uint64_t s1 = sm64(*rng + site * DTA_RNG_SITE_K);
  int64_t i = (int64_t(blockIdx.x) * blockDim.x + threadIdx.x) * 8;
  const int64_t stride = int64_t(gridDim.x) * blockDim.x * 8;
  for (; i + 8 <= n; i += stride) {
    s16x8 vx = *reinterpret_cast<const s16x8*>(x + i);
    const uint64_t h0 = sm64(s1 + uint64_t(i >> 2) * DTA_RNG_IDX_K);
    const uint64_t h1 = sm64(s1 + uint64_t((i >> 2) + 1) * DTA_RNG_IDX_K);
    s16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const uint64_t h = (j < 4) ? h0 : h1;
      const bool keep = (uint32_t(h >> (16 * (j & 3))) & 0xFFFF) >= thr16;
      o[j] = keep ? f2bf(bf2f(ushort(vx[j])) * inv_keep) : ushort(0);
    }
    *reinterpret_cast<s16x8*>(y + i) = o;
  }
  if (i < n && i + 8 > n)
    for (; i < n; ++i) {
      const uint64_t h = sm64(s1 + uint64_t(i >> 2) * DTA_RNG_IDX_K);
      const bool keep = (uint32_t(h >> (16 * (i & 3))) & 0xFFFF) >= thr16;
      y[i] = keep ? f2bf(bf2f(x[i]) * inv_keep) : ushort(0);
    }
}

__global__ void rng_tick_k(unsigned long long* ctr) { ++(*ctr); }

// dst (bf16 flat-grad view) += src (fp32 reduction result): one pass
// replacing the fp32->bf16 cast kernel + autograd's separate add kernel
// per parameter (the norm/bias/embedding grads were ~150 such kernel
// pairs per step)
__global__ void accum_f32_bf16_k(ushort* __restrict__ dst,
                                 const float* __restrict__ src, int64_t n) {
  int64_t i = (int64_t(blockIdx.x) * blockDim.x + threadIdx.x) * 8;
  const int64_t stride = int64_t(gridDim.x) * blockDim.x * 8;
  for (; i + 8 <= n; i += stride) {
    s16x8 vd = *reinterpret_cast<const s16x8*>(dst + i);
    f32x4 sa = *reinterpret_cast<const f32x4*>(src + i);
    f32x4 sb = *reinterpret_cast<const f32x4*>(src + i + 4);
    s16x8 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      o[j] = f2bf(bf2f(ushort(vd[j])) + sa[j]);
      o[j + 4] = f2bf(bf2f(ushort(vd[j + 4])) + sb[j]);
    }
    *reinterpret_cast<s16x8*>(dst + i) = o;
  }
  if (i < n)
    for (; i < n; ++i) dst[i] = f2bf(bf2f(dst[i]) + src[i]);
}

}  // namespace

#define LAUNCH_EW(kernel, n, ...)                                          \
  do {                                                                     \
    const int block_ = 256;                                                \
    kernel<<<elementwise_grid(n, block_, 8), block_, 0, s>>>(__VA_ARGS__); \
  } while (0)

void launch_gelu_fwd(const bf16_t* x, bf16_t* y, int64_t n, hipStream_t s) {
  LAUNCH_EW(map_bf16_k<gelu_f>, n, x, y, n);
}
void launch_gelu_bwd(const bf16_t* dy, const bf16_t* x, bf16_t* dx, int64_t n,
                     hipStream_t s) {
  LAUNCH_EW(map_grad_bf16_k<gelu_df>, n, dy, x, dx, n);
}
void launch_gelu_fwd_f32(const float* x, float* y, int64_t n, hipStream_t s) {
  LAUNCH_EW(map_f32_k<gelu_f>, n, x, y, n);
}
void launch_gelu_bwd_f32(const float* dy, const float* x, float* dx,
                         int64_t n, hipStream_t s) {
  LAUNCH_EW(map_grad_f32_k<gelu_df>, n, dy, x, dx, n);
}
void launch_swiglu_fwd(const bf16_t* g, const bf16_t* u, bf16_t* y, int64_t n,
                       hipStream_t s) {
  LAUNCH_EW(swiglu_fwd_k, n, g, u, y, n);
}
void launch_swiglu_bwd(const bf16_t* dy, const bf16_t* g, const bf16_t* u,
                       bf16_t* dg, bf16_t* du, int64_t n, hipStream_t s) {
  LAUNCH_EW(swiglu_bwd_k, n, dy, g, u, dg, du, n);
}
void launch_delta_sub(const float* w, const float* base, float* out,
                      int64_t n, hipStream_t s) {
  LAUNCH_EW(delta_sub_k, n, w, base, out, n);
}
void launch_axpy(float* w, const float* x, float alpha, int64_t n,
                 hipStream_t s) {
  LAUNCH_EW(axpy_k, n, w, x, alpha, n);
}
void launch_nan_any(const float* x, int64_t n, int* flag, hipStream_t s) {
  LAUNCH_EW(nan_any_k, n, x, n, flag);
}
void launch_l2norm_sq(const float* x, int64_t n, float* out, hipStream_t s) {
  LAUNCH_EW(l2norm_sq_k, n, x, n, out);
}
void launch_f32_to_bf16(const float* x, bf16_t* y, int64_t n, hipStream_t s) {
  LAUNCH_EW(f32_to_bf16_k, n, x, y, n);
}
void launch_colsum(const bf16_t* x, float* part, float* out, int64_t rows,
                   int cols, int stripes, hipStream_t s) {
  const ColRedCfg cfg = dta_colred_cfg(rows, cols);
  dim3 g1(cfg.gx, unsigned(stripes));
  colsum_part_k<<<g1, cfg.threads, 0, s>>>(x, part, rows, cols);
  const int g2 = (cols / 4 + 255) / 256;
  const int ry = stripes < 32 ? stripes : 32;
  colsum_reduce_k<<<dim3(g2, ry), 256, 0, s>>>(part, out, stripes, cols);
}
void launch_dropout(const bf16_t* x, bf16_t* y, int64_t n,
                    const unsigned long long* rng, unsigned long long site,
                    unsigned int thr16, float inv_keep, hipStream_t s) {
  LAUNCH_EW(dropout_k, n, x, y, n, rng, site, thr16, inv_keep);
}
void launch_rng_tick(unsigned long long* ctr, hipStream_t s) {
  rng_tick_k<<<1, 1, 0, s>>>(ctr);
}
void launch_accum_f32_bf16(bf16_t* dst, const float* src, int64_t n,
                           hipStream_t s) {
  LAUNCH_EW(accum_f32_bf16_k, n, dst, src, n);
}
