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

// ---- bf16 <-> f32 ---------------------------------------------------------
DEV float bf2f(ushort u) {
  union { uint32_t u32; float f; } x;
  x.u32 = uint32_t(u) << 16;
  return x.f;
}
DEV ushort f2bf(float f) {
  union { float f; uint32_t u32; } x;
  x.f = f;
  uint32_t u = x.u32;
  uint32_t lsb = (u >> 16) & 1;          // round to nearest even
  u += 0x7fff + lsb;
  if ((x.u32 & 0x7fffffff) > 0x7f800000) return ushort((x.u32 >> 16) | 0x40);
  return ushort(u >> 16);
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
