// Common device helpers for the gfx950 (CDNA4/MI355X) kernel library.
// Wave size is 64 on CDNA — every cross-lane idiom below assumes it.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#define DEV __device__ __forceinline__

constexpr int WAVE = 64;

// ---- vector types ---------------------------------------------------------
typedef __attribute__((ext_vector_type(2))) float    f32x2;
typedef __attribute__((ext_vector_type(4))) float    f32x4;
typedef __attribute__((ext_vector_type(16))) float   f32x16;
typedef __attribute__((ext_vector_type(4))) short    s16x4;
typedef __attribute__((ext_vector_type(8))) short    s16x8;   // 8 bf16 = 16 B
typedef __attribute__((ext_vector_type(8))) __bf16   bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16   bf16x4;

// ---- bf16 <-> f32 (hardware v_cvt; never bit-twiddle these in software) ---
DEV float bf2f(ushort u) {
  union { ushort s; __bf16 b; } x;
  x.s = u;
  return static_cast<float>(x.b);
}
DEV ushort f2bf(float f) {
  union { ushort s; __bf16 b; } x;
  x.b = static_cast<__bf16>(f);
  return x.s;
}

// ---- fast transcendentals (exp2/log2 are single hardware instructions) ----
#define LOG2E 1.4426950408889634f
#define LN2 0.6931471805599453f
DEV float fast_exp(float x) { return __builtin_exp2f(x * LOG2E); }
DEV float fast_log(float x) { return __builtin_log2f(x) * LN2; }
DEV float fast_tanh(float x) {
  // tanh(x) = (e-1)/(e+1), e = exp2(2x*log2e); clamp so e never overflows
  float xc = fminf(fmaxf(x, -10.0f), 10.0f);
  float e = __builtin_exp2f(2.0f * LOG2E * xc);
  return (e - 1.0f) / (e + 1.0f);
}

// ---- wave reductions (64-lane) --------------------------------------------
DEV float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off);
  return v;
}
DEV float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off));
  return v;
}
// reduce within contiguous groups of G lanes (G power of two <= 64)
template <int G> DEV float group_sum(float v) {
#pragma unroll
  for (int off = G / 2; off > 0; off >>= 1) v += __shfl_xor(v, off);
  return v;
}
template <int G> DEV float group_max(float v) {
#pragma unroll
  for (int off = G / 2; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off));
  return v;
}

// block reduce via LDS (expects <=16 waves); every thread returns the result
template <int MAXW = 16>
DEV float block_sum(float v, float* lds_scratch) {
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  const int nw = (blockDim.x + 63) >> 6;
  v = wave_sum(v);
  if (lane == 0) lds_scratch[wid] = v;
  __syncthreads();
  float r = 0.f;
#pragma unroll
  for (int i = 0; i < MAXW; ++i)
    if (i < nw) r += lds_scratch[i];
  __syncthreads();
  return r;
}
template <int MAXW = 16>
DEV float block_max(float v, float* lds_scratch) {
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  const int nw = (blockDim.x + 63) >> 6;
  v = wave_max(v);
  if (lane == 0) lds_scratch[wid] = v;
  __syncthreads();
  float r = -INFINITY;
#pragma unroll
  for (int i = 0; i < MAXW; ++i)
    if (i < nw) r = fmaxf(r, lds_scratch[i]);
  __syncthreads();
  return r;
}

// ---- counter-based dropout RNG (splitmix64 finalizer) ---------------------
// Graph-capturable, storage-free dropout: masks are a pure function of a
// device-resident step counter, a per-call-site salt and the element
// coordinates, so forward and both backward kernels regenerate identical
// masks with zero mask memory, and a hipGraph replay gets fresh masks from
// the ticked counter. One 64-bit hash yields FOUR 16-bit keep-draws
// (elements 4j..4j+3 of the innermost axis). The exact chain (mirrored
// bit-for-bit by the host gold in ops/droprng.py):
//   s1 = sm64(*ctr + site * 0xA24BAED4963EE407)
//   attention (per b,h head): s2 = sm64(s1 ^ (bh * 0x9E3779B97F4A7C15))
//       h  = sm64(s2 + ((q << 24) | (k >> 2)) * 0xD1B54A32D192ED03)
//       draw = (h >> (16 * (k & 3))) & 0xFFFF
//   elementwise: h = sm64(s1 + (i >> 2) * 0xD1B54A32D192ED03)
//       draw = (h >> (16 * (i & 3))) & 0xFFFF
// keep iff draw >= thr16, thr16 = ceil(p * 65536) => the realized keep
// probability is exactly 1 - thr16/65536 and scale = 65536/(65536-thr16)
// keeps the expectation unbiased.
DEV uint64_t sm64(uint64_t z) {
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
  return z ^ (z >> 31);
}
#define DTA_RNG_SITE_K 0xA24BAED4963EE407ULL
#define DTA_RNG_HEAD_K 0x9E3779B97F4A7C15ULL
#define DTA_RNG_IDX_K 0xD1B54A32D192ED03ULL

// ---- grid sizing (Guideline 11: cap + grid-stride for memory-bound) -------
constexpr int MAX_RESIDENT_BLOCKS = 2048;  // 256 CU x 8 blocks
inline int elementwise_grid(int64_t n, int block, int vec) {
  int64_t want = (n + int64_t(block) * vec - 1) / (int64_t(block) * vec);
  return int(want < MAX_RESIDENT_BLOCKS ? (want > 0 ? want : 1)
                                        : MAX_RESIDENT_BLOCKS);
}

#define HIP_CHECK_LAST()                                                     \
  do {                                                                       \
    hipError_t e_ = hipGetLastError();                                       \
    TORCH_CHECK(e_ == hipSuccess, "HIP kernel launch failed: ",              \
                hipGetErrorString(e_));                                      \
  } while (0)
