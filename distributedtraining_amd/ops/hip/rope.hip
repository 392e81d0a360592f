// Rotary position embedding (Llama family — beyond the reference's GPT-2
// learned positions, neurons/miner.py:60-62), fwd and bwd (bwd = inverse
// rotation).
// Host-precomputed cos/sin tables [S, D/2] fp32 (Appendix B: never call
// trig per element on-device — it turns a memory-bound op VALU-bound).
// Layout: x [BH, S, D] with the half-split convention
// (x1=x[..,:D/2], x2=x[..,D/2:]): y1 = x1*c - x2*s ; y2 = x1*s + x2*c.
#include "dta_common.h"
#include "dta_kernels.h"

namespace {

constexpr int ROW_WAVES = 4;

// pos_p: device position offset (graph-replayable decode — the table row
// is *pos_p + in-sequence index, so one captured graph serves every step)
template <bool BWD>
__global__ void rope_k(const ushort* __restrict__ x,
                       const float* __restrict__ cost,
                       const float* __restrict__ sint,
                       ushort* __restrict__ y, int64_t bh, int seq, int hd,
                       const int* __restrict__ pos_p) {
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  const int d2 = hd >> 1;
  const int pos0 = pos_p ? *pos_p : 0;
  const int64_t rows = bh * seq;
  for (int64_t r = int64_t(blockIdx.x) * ROW_WAVES + wid; r < rows;
       r += int64_t(gridDim.x) * ROW_WAVES) {
    const int pos = pos0 + int(r % seq);
    const ushort* xr = x + r * hd;
    ushort* yr = y + r * hd;
    const float* c = cost + int64_t(pos) * d2;
    const float* sn = sint + int64_t(pos) * d2;
    for (int i = lane * 2; i + 2 <= d2; i += 128) {
      float x1a = bf2f(xr[i]),      x1b = bf2f(xr[i + 1]);
      float x2a = bf2f(xr[d2 + i]), x2b = bf2f(xr[d2 + i + 1]);
      f32x2 cv = *reinterpret_cast<const f32x2*>(c + i);
      f32x2 sv = *reinterpret_cast<const f32x2*>(sn + i);
      float sa = BWD ? -sv[0] : sv[0];
      float sb = BWD ? -sv[1] : sv[1];
      yr[i]          = f2bf(x1a * cv[0] - x2a * sa);
      yr[i + 1]      = f2bf(x1b * cv[1] - x2b * sb);
      yr[d2 + i]     = f2bf(x1a * sa + x2a * cv[0]);
      yr[d2 + i + 1] = f2bf(x1b * sb + x2b * cv[1]);
    }
    if (d2 & 1) {  // odd half-dim tail (tiny test models)
      if (lane == 0) {
        int i = d2 - 1;
        float x1 = bf2f(xr[i]), x2 = bf2f(xr[d2 + i]);
        float cc = c[i], ss = BWD ? -sn[i] : sn[i];
        yr[i] = f2bf(x1 * cc - x2 * ss);
        yr[d2 + i] = f2bf(x1 * ss + x2 * cc);
      }
    }
  }
}

}  // namespace

void launch_rope(const bf16_t* x, const float* cos_t, const float* sin_t,
                 bf16_t* y, int64_t bh, int seq, int hd, bool backward,
                 const int* pos_p, hipStream_t s) {
  int64_t rows = bh * seq;
  int64_t want = (rows + ROW_WAVES - 1) / ROW_WAVES;
  const int grid = int(want < 4096 ? (want > 0 ? want : 1) : 4096);
  if (backward)
    rope_k<true><<<grid, 256, 0, s>>>(x, cos_t, sin_t, y, bh, seq, hd,
                                      pos_p);
  else
    rope_k<false><<<grid, 256, 0, s>>>(x, cos_t, sin_t, y, bh, seq, hd,
                                       pos_p);
}
