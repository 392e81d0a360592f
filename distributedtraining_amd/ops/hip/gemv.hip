// Weight-streaming GEMV / skinny GEMM for serving decode (M <= 4).
// (Serving is beyond-parity: the reference trains only; this backs the
// deployed-model inference endpoint, utils/serve.py.)
//
// y[M,N] = x[M,K] @ W[N,K]^T (+ bias) — the decode-step projection
// shapes (M = batch). hipBLASLt's small-M path measured only ~1.7 TB/s
// of weight stream on the Llama-3-8B decode step (9.4 ms/token against
// a ~2.5 ms HBM floor); this kernel is a pure streaming design:
//   * x (tiny) staged fp32 in LDS ONCE per block, transposed [k][m];
//     each block then walks MANY output columns (grid-stride) with x
//     resident — a stage-per-column variant serialized on the staging
//     latency and measured slower than hipBLASLt;
//   * per column, the block's 4 waves (256 lanes) stride the
//     K-contiguous weight row in 16 B chunks with a 2-deep load
//     pipeline; fp32 accumulators, wave-reduce + LDS cross-wave
//     combine, fused bias;
//   * K > KC (Llama down-proj K=14336) falls back to chunked staging
//     inside the column loop.
// Weights are read exactly once per token. M > 4 falls back to
// hipBLASLt in ops.linear (its tile machinery wins again from M ~ 16).
#include "dta_common.h"
#include "dta_kernels.h"

namespace {

// LDS budget 32 KiB fp32 for the x panel: KC * MT = 8192
template <int MT>
__global__ void gemv_k(const ushort* __restrict__ x,
                       const ushort* __restrict__ w,
                       const ushort* __restrict__ bias,
                       ushort* __restrict__ y, int M, int64_t N, int K) {
  constexpr int KC = 8192 / MT;
  __shared__ float xs[8192];
  __shared__ float red[4][4];
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;

  const auto stage = [&](int kc, int kn) {
    __syncthreads();
    for (int i = threadIdx.x; i < kn * MT; i += 256) {
      const int k = i / MT, mm = i % MT;
      xs[i] = (mm < M) ? bf2f(x[int64_t(mm) * K + kc + k]) : 0.f;
    }
    __syncthreads();
  };

  // one weight-row sweep over xs[0..kn) against W row n at offset kc
  const auto sweep = [&](float* acc, const ushort* wr, int kn) {
    int k8 = threadIdx.x * 8;
    const int step = 256 * 8;
    bool have = k8 + 8 <= kn;
    s16x8 wv{};
    if (have) wv = *reinterpret_cast<const s16x8*>(wr + k8);
    while (have) {
      const int nk8 = k8 + step;
      const bool hn = nk8 + 8 <= kn;
      s16x8 nxt{};
      if (hn) nxt = *reinterpret_cast<const s16x8*>(wr + nk8);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float wf = bf2f(ushort(wv[j]));
        const float* xk = xs + (k8 + j) * MT;
#pragma unroll
        for (int mm = 0; mm < MT; ++mm)
          acc[mm] = fmaf(wf, xk[mm], acc[mm]);
      }
      k8 = nk8;
      wv = nxt;
      have = hn;
    }
    for (int k = (kn & ~7) + threadIdx.x; k < kn; k += 256) {
      const float wf = bf2f(wr[k]);
#pragma unroll
      for (int mm = 0; mm < MT; ++mm)
        acc[mm] = fmaf(wf, xs[k * MT + mm], acc[mm]);
    }
  };

  const auto emit = [&](float* acc, int64_t n) {
#pragma unroll
    for (int mm = 0; mm < MT; ++mm) acc[mm] = wave_sum(acc[mm]);
    if (lane == 0) {
#pragma unroll
      for (int mm = 0; mm < MT; ++mm) red[wave][mm] = acc[mm];
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      const float bv = bias ? bf2f(bias[n]) : 0.f;
      for (int mm = 0; mm < M; ++mm)
        y[int64_t(mm) * N + n] = f2bf(red[0][mm] + red[1][mm] +
                                      red[2][mm] + red[3][mm] + bv);
    }
    __syncthreads();
  };

  if (K <= KC) {
    stage(0, K);
    for (int64_t n = blockIdx.x; n < N; n += gridDim.x) {
      float acc[MT];
#pragma unroll
      for (int mm = 0; mm < MT; ++mm) acc[mm] = 0.f;
      sweep(acc, w + n * int64_t(K), K);
      emit(acc, n);
    }
    return;
  }
  for (int64_t n = blockIdx.x; n < N; n += gridDim.x) {
    float acc[MT];
#pragma unroll
    for (int mm = 0; mm < MT; ++mm) acc[mm] = 0.f;
    for (int kc = 0; kc < K; kc += KC) {
      const int kn = (K - kc) < KC ? (K - kc) : KC;
      stage(kc, kn);
      sweep(acc, w + n * int64_t(K) + kc, kn);
    }
    emit(acc, n);
  }
}

}  // namespace

void launch_gemv(const bf16_t* x, const bf16_t* w, const bf16_t* bias,
                 bf16_t* y, int M, int64_t N, int K, hipStream_t s) {
  const int grid = int(N < 2048 ? N : 2048);
#define GEMV(MT)                                                          \
  gemv_k<MT><<<grid, 256, 0, s>>>(x, w, bias, y, M, N, K)
  if (M == 1) GEMV(1);
  else if (M <= 2) GEMV(2);
  else GEMV(4);
#undef GEMV
}
