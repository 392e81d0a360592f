#include "hip/hip_runtime.h"
// Fused causal flash attention, forward + backward, MFMA
// (mfma_f32_16x16x32_bf16) with online softmax — the CDNA4 replacement for
// the reference's transformers attention (SURVEY.md §2.2 op table; miner
// seq 64, validator seq 512).
//
// Structure (v1, correctness-first):
//   * one 64-lane wave owns a 16-row Q tile; blocks of 4 independent waves
//     (no __syncthreads in the loop), grid = BH x ceil(S/64).
//   * K/V tiles are read directly from global memory — at these sequence
//     lengths a (b,h)'s K/V fit in L2 (guide §5 common-mistake 7: LDS
//     staging of L2-resident data is pure overhead).
//   * swapped QK^T: mfma(A=K_tile, B=Q^T) puts a query's scores in lanes
//     sharing (lane&15) so the softmax row-reduce is two shfl_xor ops.
//   * P is redistributed score->A-fragment via 16 shfls per 32-key tile
//     and fed to mfma(P, V) accumulating O in f32.
//
// Fragment maps (verified on-device by mfma_selftest, see bindings.cpp):
//   mfma_f32_16x16x32_bf16: A[i][k]: i=lane&15, k=8*(lane>>4)+j (j=0..7)
//                           B[k][n]: n=lane&15, k=8*(lane>>4)+j
//                           C/D[i][j]: col=lane&15, row=4*(lane>>4)+reg
//
// Backward is the standard flash recomputation split into a dQ kernel
// (grid over Q tiles) and a dK/dV kernel (grid over KV tiles), with
// delta_row = sum_d dO*O precomputed.
#include "dta_common.h"
#include "dta_kernels.h"

namespace {

using mfma16 = f32x4;

DEV f32x4 mfma_bf16(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// load an A/B fragment whose per-lane 8 elements are CONTIGUOUS in memory:
// row r(lane), starting col c0(lane) — one 16B read.
DEV bf16x8 load_frag_row(const ushort* base, int64_t row_stride, int row,
                         int col0) {
  const s16x8 v = *reinterpret_cast<const s16x8*>(base + int64_t(row) * row_stride + col0);
  union { s16x8 s; bf16x8 b; } u;
  u.s = v;
  return u.b;
}

// load a B fragment with per-lane elements STRIDED by rows (V-style):
// element j comes from row r0+j, fixed col. 8 scalar reads (L2-resident).
DEV bf16x8 load_frag_col(const ushort* base, int64_t row_stride, int row0,
                         int col, int row_max /*exclusive*/) {
  bf16x8 out;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int r = row0 + j;
    ushort u = (r < row_max) ? base[int64_t(r) * row_stride + col] : 0;
    union { ushort s; __bf16 b; } c;
    c.s = u;
    out[j] = c.b;
  }
  return out;
}

DEV bf16x8 pack_bf16x8(const float* f) {
  bf16x8 o;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    union { ushort s; __bf16 b; } c;
    c.s = f2bf(f[j]);
    o[j] = c.b;
  }
  return o;
}

// Redistribute a 32-key score tile from C layout (q=lane&15,
// key=16*ks+4*(lane>>4)+r held in p0/p1) into the A-fragment layout
// (q=lane&15, k=8*(lane>>4)+j) — two shfl rounds per j.
DEV bf16x8 scores_to_afrag(const f32x4& p0, const f32x4& p1, int lane) {
  const int g = lane >> 4;
  float pa[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int kt = 8 * g + j;                    // the key this lane wants
    const int src = (lane & 15) + 16 * ((kt >> 2) & 3);
    const float va = __shfl(p0[j & 3], src);     // keys 0..15 live in p0
    const float vb = __shfl(p1[j & 3], src);     // keys 16..31 live in p1
    pa[j] = (kt < 16) ? va : vb;
  }
  return pack_bf16x8(pa);
}

// ---------------- forward ----------------
template <int DTILES>  // D = 16*DTILES (4 -> 64, 8 -> 128)
__global__ void attn_fwd_k(const ushort* __restrict__ q,
                           const ushort* __restrict__ k,
                           const ushort* __restrict__ v,
                           ushort* __restrict__ o, float* __restrict__ lse,
                           int seq, float scale) {
  constexpr int D = 16 * DTILES;
  constexpr int DSL = DTILES / 2;  // 32-wide d slices
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int64_t bh = blockIdx.y;
  const int q0 = blockIdx.x * 64 + wid * 16;  // this wave's q-tile base
  if (q0 >= seq) return;
  const ushort* qp = q + bh * int64_t(seq) * D;
  const ushort* kp = k + bh * int64_t(seq) * D;
  const ushort* vp = v + bh * int64_t(seq) * D;

  // Q fragments (B operand of the swapped QK^T): 16B per lane per slice
  bf16x8 qb[DSL];
  const int qrow = q0 + (lane & 15);
  const int qr_ld = qrow < seq ? qrow : seq - 1;
#pragma unroll
  for (int sl = 0; sl < DSL; ++sl)
    qb[sl] = load_frag_row(qp, D, qr_ld, 32 * sl + 8 * (lane >> 4));

  f32x4 acc[DTILES];
#pragma unroll
  for (int t = 0; t < DTILES; ++t) acc[t] = f32x4{0, 0, 0, 0};
  float m_run = -INFINITY, l_run = 0.f;

  const int kv_end = min(seq, q0 + 16);  // causal: keys <= max q row
  for (int kv0 = 0; kv0 < kv_end; kv0 += 32) {
    // scores S^T subtiles: p0 = keys kv0..kv0+15, p1 = +16..31
    f32x4 p0 = {0, 0, 0, 0}, p1 = {0, 0, 0, 0};
#pragma unroll
    for (int sl = 0; sl < DSL; ++sl) {
      const int c0 = 32 * sl + 8 * (lane >> 4);
      const int kr0 = kv0 + (lane & 15);
      const int kr1 = kr0 + 16;
      bf16x8 ka = load_frag_row(kp, D, kr0 < seq ? kr0 : seq - 1, c0);
      p0 = mfma_bf16(ka, qb[sl], p0);
      bf16x8 kb = load_frag_row(kp, D, kr1 < seq ? kr1 : seq - 1, c0);
      p1 = mfma_bf16(kb, qb[sl], p1);
    }
    // causal + bounds mask, scale; layout: q=lane&15, key=16ks+4g+r
    const int g = lane >> 4;
    float mx = -INFINITY;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int k0a = kv0 + 4 * g + r, k1a = k0a + 16;
      p0[r] = (k0a <= qrow && k0a < seq) ? p0[r] * scale : -INFINITY;
      p1[r] = (k1a <= qrow && k1a < seq) ? p1[r] * scale : -INFINITY;
      mx = fmaxf(mx, fmaxf(p0[r], p1[r]));
    }
    mx = fmaxf(mx, __shfl_xor(mx, 16));
    mx = fmaxf(mx, __shfl_xor(mx, 32));   // all 4 groups now share row max
    const float m_new = fmaxf(m_run, mx);
    const float alpha = (m_run == -INFINITY) ? 0.f : expf(m_run - m_new);
    float psum = 0.f;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      p0[r] = (p0[r] == -INFINITY) ? 0.f : expf(p0[r] - m_new);
      p1[r] = (p1[r] == -INFINITY) ? 0.f : expf(p1[r] - m_new);
      psum += p0[r] + p1[r];
    }
    psum += __shfl_xor(psum, 16);
    psum += __shfl_xor(psum, 32);
    l_run = l_run * alpha + psum;
    m_run = m_new;

    // O rescale: lane holds rows q=4g+r -> pull alpha from lane (4g+r)
    bf16x8 pa = scores_to_afrag(p0, p1, lane);
#pragma unroll
    for (int t = 0; t < DTILES; ++t) {
      const float a_r0 = __shfl(alpha, 4 * g + 0);
      const float a_r1 = __shfl(alpha, 4 * g + 1);
      const float a_r2 = __shfl(alpha, 4 * g + 2);
      const float a_r3 = __shfl(alpha, 4 * g + 3);
      acc[t][0] *= a_r0; acc[t][1] *= a_r1;
      acc[t][2] *= a_r2; acc[t][3] *= a_r3;
      bf16x8 vb = load_frag_col(vp, D, kv0 + 8 * g, 16 * t + (lane & 15),
                                seq);
      acc[t] = mfma_bf16(pa, vb, acc[t]);
    }
  }

  // epilogue: O /= l, store; lse = m + log(l)
  const int g = lane >> 4;
  const float l_q = (lane < 16) ? l_run : 0.f;  // canonical copy at g=0
  const float m_q = m_run;
  if (lane < 16 && qrow < seq) lse[bh * seq + qrow] = m_q + logf(l_run);
#pragma unroll
  for (int t = 0; t < DTILES; ++t) {
    float inv[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float lr = __shfl(l_run, 4 * g + r);
      inv[r] = lr > 0.f ? 1.0f / lr : 0.f;
    }
    ushort* op = o + bh * int64_t(seq) * D;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int orow = q0 + 4 * g + r;
      if (orow < seq)
        op[int64_t(orow) * D + 16 * t + (lane & 15)] = f2bf(acc[t][r] * inv[r]);
    }
  }
}

// ---------------- delta = rowsum(dO * O) ----------------
__global__ void attn_delta_k(const ushort* __restrict__ dout,
                             const ushort* __restrict__ o,
                             float* __restrict__ delta, int64_t rows,
                             int hd) {
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  for (int64_t r = int64_t(blockIdx.x) * 4 + wid; r < rows;
       r += int64_t(gridDim.x) * 4) {
    const ushort* a = dout + r * hd;
    const ushort* b = o + r * hd;
    float acc = 0.f;
    for (int i = lane * 2; i + 2 <= hd; i += 128) {
      acc = fmaf(bf2f(a[i]), bf2f(b[i]), acc);
      acc = fmaf(bf2f(a[i + 1]), bf2f(b[i + 1]), acc);
    }
    acc = wave_sum(acc);
    if (lane == 0) delta[r] = acc;
  }
}

// ---------------- backward dQ ----------------
// dQ[q,d] = scale * sum_k P(q,k) * (dP(q,k) - delta_q) * K[k,d]
template <int DTILES>
__global__ void attn_bwd_dq_k(const ushort* __restrict__ dout,
                              const ushort* __restrict__ q,
                              const ushort* __restrict__ k,
                              const ushort* __restrict__ v,
                              const float* __restrict__ lse,
                              const float* __restrict__ delta,
                              ushort* __restrict__ dq, int seq, float scale) {
  constexpr int D = 16 * DTILES;
  constexpr int DSL = DTILES / 2;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int64_t bh = blockIdx.y;
  const int q0 = blockIdx.x * 64 + wid * 16;
  if (q0 >= seq) return;
  const ushort* qp = q + bh * int64_t(seq) * D;
  const ushort* kp = k + bh * int64_t(seq) * D;
  const ushort* vp = v + bh * int64_t(seq) * D;
  const ushort* dop = dout + bh * int64_t(seq) * D;

  const int qrow = q0 + (lane & 15);
  const int qr_ld = qrow < seq ? qrow : seq - 1;
  bf16x8 qb[DSL], dob[DSL];
#pragma unroll
  for (int sl = 0; sl < DSL; ++sl) {
    const int c0 = 32 * sl + 8 * (lane >> 4);
    qb[sl] = load_frag_row(qp, D, qr_ld, c0);
    dob[sl] = load_frag_row(dop, D, qr_ld, c0);
  }
  const float lse_q = lse[bh * seq + qr_ld];
  const float dlt_q = delta[bh * seq + qr_ld];

  f32x4 acc[DTILES];
#pragma unroll
  for (int t = 0; t < DTILES; ++t) acc[t] = f32x4{0, 0, 0, 0};

  const int kv_end = min(seq, q0 + 16);
  for (int kv0 = 0; kv0 < kv_end; kv0 += 32) {
    f32x4 s0 = {0, 0, 0, 0}, s1 = {0, 0, 0, 0};
    f32x4 dp0 = {0, 0, 0, 0}, dp1 = {0, 0, 0, 0};
#pragma unroll
    for (int sl = 0; sl < DSL; ++sl) {
      const int c0 = 32 * sl + 8 * (lane >> 4);
      const int kr0 = kv0 + (lane & 15), kr1 = kr0 + 16;
      bf16x8 ka = load_frag_row(kp, D, kr0 < seq ? kr0 : seq - 1, c0);
      bf16x8 kb = load_frag_row(kp, D, kr1 < seq ? kr1 : seq - 1, c0);
      bf16x8 va = load_frag_row(vp, D, kr0 < seq ? kr0 : seq - 1, c0);
      bf16x8 vb2 = load_frag_row(vp, D, kr1 < seq ? kr1 : seq - 1, c0);
      s0 = mfma_bf16(ka, qb[sl], s0);
      s1 = mfma_bf16(kb, qb[sl], s1);
      dp0 = mfma_bf16(va, dob[sl], dp0);
      dp1 = mfma_bf16(vb2, dob[sl], dp1);
    }
    const int g = lane >> 4;
    // lse/delta for this lane's q (= lane&15) are already per-lane
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int k0a = kv0 + 4 * g + r, k1a = k0a + 16;
      const bool v0 = (k0a <= qrow && k0a < seq);
      const bool v1 = (k1a <= qrow && k1a < seq);
      const float P0 = v0 ? expf(s0[r] * scale - lse_q) : 0.f;
      const float P1 = v1 ? expf(s1[r] * scale - lse_q) : 0.f;
      s0[r] = P0 * (dp0[r] - dlt_q);
      s1[r] = P1 * (dp1[r] - dlt_q);
    }
    bf16x8 dsa = scores_to_afrag(s0, s1, lane);
#pragma unroll
    for (int t = 0; t < DTILES; ++t) {
      bf16x8 kb2 = load_frag_col(kp, D, kv0 + 8 * g, 16 * t + (lane & 15),
                                 seq);
      acc[t] = mfma_bf16(dsa, kb2, acc[t]);
    }
  }
  const int g = lane >> 4;
  ushort* dqp = dq + bh * int64_t(seq) * D;
#pragma unroll
  for (int t = 0; t < DTILES; ++t)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int orow = q0 + 4 * g + r;
      if (orow < seq)
        dqp[int64_t(orow) * D + 16 * t + (lane & 15)] =
            f2bf(acc[t][r] * scale);
    }
}

// ---------------- backward dK/dV ----------------
// wave owns a 16-row KV tile; iterates q tiles of 32 (q >= kv0).
// dV[k,d] = sum_q P^T(k,q) dO[q,d];  dK[k,d] = scale*sum_q dS^T(k,q) Q[q,d]
// P^T from S = mfma(Q_sub, K^T) -> C: (k=lane&15, q=4g+r).
template <int DTILES>
__global__ void attn_bwd_dkv_k(const ushort* __restrict__ dout,
                               const ushort* __restrict__ q,
                               const ushort* __restrict__ k,
                               const ushort* __restrict__ v,
                               const float* __restrict__ lse,
                               const float* __restrict__ delta,
                               ushort* __restrict__ dk,
                               ushort* __restrict__ dv, int seq,
                               float scale) {
  constexpr int D = 16 * DTILES;
  constexpr int DSL = DTILES / 2;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int64_t bh = blockIdx.y;
  const int kv0 = blockIdx.x * 64 + wid * 16;
  if (kv0 >= seq) return;
  const ushort* qp = q + bh * int64_t(seq) * D;
  const ushort* kp = k + bh * int64_t(seq) * D;
  const ushort* vp = v + bh * int64_t(seq) * D;
  const ushort* dop = dout + bh * int64_t(seq) * D;

  // K fragments as B operand of mfma(Q, K^T): (k=lane&15, d=8g+j)
  const int krow = kv0 + (lane & 15);
  const int kr_ld = krow < seq ? krow : seq - 1;
  bf16x8 kb[DSL];
#pragma unroll
  for (int sl = 0; sl < DSL; ++sl)
    kb[sl] = load_frag_row(kp, D, kr_ld, 32 * sl + 8 * (lane >> 4));

  f32x4 acck[DTILES], accv[DTILES];
#pragma unroll
  for (int t = 0; t < DTILES; ++t) {
    acck[t] = f32x4{0, 0, 0, 0};
    accv[t] = f32x4{0, 0, 0, 0};
  }

  const int g = lane >> 4;
  const int q_start = (kv0 / 32) * 32;  // first q tile with q >= kv0
  for (int q0 = q_start; q0 < seq; q0 += 32) {
    f32x4 s0 = {0, 0, 0, 0}, s1 = {0, 0, 0, 0};
#pragma unroll
    for (int sl = 0; sl < DSL; ++sl) {
      const int c0 = 32 * sl + 8 * (lane >> 4);
      const int qr0 = q0 + (lane & 15), qr1 = qr0 + 16;
      bf16x8 qa = load_frag_row(qp, D, qr0 < seq ? qr0 : seq - 1, c0);
      bf16x8 qa1 = load_frag_row(qp, D, qr1 < seq ? qr1 : seq - 1, c0);
      // S^T? no: mfma(A=Q[16q][32d], B=K^T) gives (k=lane&15, q=4g+r)
      s0 = mfma_bf16(qa, kb[sl], s0);
      s1 = mfma_bf16(qa1, kb[sl], s1);
    }
    // lane holds (k=krow, q = q0 + 4g + r (+16 for s1))
    f32x4 p0, p1, ds0, ds1;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qa0 = q0 + 4 * g + r, qa1 = qa0 + 16;
      const bool v0 = (qa0 >= krow && qa0 < seq);
      const bool v1 = (qa1 >= krow && qa1 < seq);
      const float lse0 = lse[bh * seq + (v0 ? qa0 : 0)];
      const float lse1 = lse[bh * seq + (v1 ? qa1 : 0)];
      p0[r] = v0 ? expf(s0[r] * scale - lse0) : 0.f;
      p1[r] = v1 ? expf(s1[r] * scale - lse1) : 0.f;
    }
    // dP^T(k,q) = sum_d V[k,d] dO[q,d]: mfma(A=dO[16q][32d], B=V^T)
    f32x4 dp0 = {0, 0, 0, 0}, dp1 = {0, 0, 0, 0};
    bf16x8 vbf[DSL];
#pragma unroll
    for (int sl = 0; sl < DSL; ++sl)
      vbf[sl] = load_frag_row(vp, D, kr_ld, 32 * sl + 8 * (lane >> 4));
#pragma unroll
    for (int sl = 0; sl < DSL; ++sl) {
      const int c0 = 32 * sl + 8 * (lane >> 4);
      const int qr0 = q0 + (lane & 15), qr1 = qr0 + 16;
      bf16x8 doa = load_frag_row(dop, D, qr0 < seq ? qr0 : seq - 1, c0);
      bf16x8 doa1 = load_frag_row(dop, D, qr1 < seq ? qr1 : seq - 1, c0);
      dp0 = mfma_bf16(doa, vbf[sl], dp0);
      dp1 = mfma_bf16(doa1, vbf[sl], dp1);
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qa0 = q0 + 4 * g + r, qa1 = qa0 + 16;
      const float d0 = delta[bh * seq + ((qa0 < seq) ? qa0 : 0)];
      const float d1 = delta[bh * seq + ((qa1 < seq) ? qa1 : 0)];
      ds0[r] = p0[r] * (dp0[r] - d0);
      ds1[r] = p1[r] * (dp1[r] - d1);
    }
    // A fragments (k=lane&15, q=8g+j) from (k=lane&15, q=16ks+4g+r)
    bf16x8 pa = scores_to_afrag(p0, p1, lane);
    bf16x8 dsa = scores_to_afrag(ds0, ds1, lane);
#pragma unroll
    for (int t = 0; t < DTILES; ++t) {
      bf16x8 dob = load_frag_col(dop, D, q0 + 8 * g, 16 * t + (lane & 15),
                                 seq);
      accv[t] = mfma_bf16(pa, dob, accv[t]);
      bf16x8 qcb = load_frag_col(qp, D, q0 + 8 * g, 16 * t + (lane & 15),
                                 seq);
      acck[t] = mfma_bf16(dsa, qcb, acck[t]);
    }
  }
  ushort* dkp = dk + bh * int64_t(seq) * D;
  ushort* dvp = dv + bh * int64_t(seq) * D;
#pragma unroll
  for (int t = 0; t < DTILES; ++t)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int orow = kv0 + 4 * g + r;
      if (orow < seq) {
        dkp[int64_t(orow) * D + 16 * t + (lane & 15)] =
            f2bf(acck[t][r] * scale);
        dvp[int64_t(orow) * D + 16 * t + (lane & 15)] = f2bf(accv[t][r]);
      }
    }
}

// ---------------- mfma layout probes ----------------
__global__ void mfma_probe16_k(const ushort* A, const ushort* B, float* D) {
  const int lane = threadIdx.x & 63;
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    union { ushort s; __bf16 v; } ca, cb;
    ca.s = A[(lane & 15) * 32 + 8 * (lane >> 4) + j];   // A[16][32]
    cb.s = B[(8 * (lane >> 4) + j) * 16 + (lane & 15)]; // B[32][16]
    a[j] = ca.v;
    b[j] = cb.v;
  }
  f32x4 c = {0, 0, 0, 0};
  c = mfma_bf16(a, b, c);
#pragma unroll
  for (int r = 0; r < 4; ++r)
    D[(4 * (lane >> 4) + r) * 16 + (lane & 15)] = c[r];
}

__global__ void mfma_probe32_k(const ushort* A, const ushort* B, float* D) {
  const int lane = threadIdx.x & 63;
  typedef __attribute__((ext_vector_type(16))) float f32x16_t;
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    union { ushort s; __bf16 v; } ca, cb;
    ca.s = A[(lane & 31) * 16 + 8 * (lane >> 5) + j];   // A[32][16]
    cb.s = B[(8 * (lane >> 5) + j) * 32 + (lane & 31)]; // B[16][32]
    a[j] = ca.v;
    b[j] = cb.v;
  }
  f32x16_t c = {};
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    D[row * 32 + (lane & 31)] = c[r];
  }
}

}  // namespace

void launch_attn_fwd(const bf16_t* q, const bf16_t* k, const bf16_t* v,
                     bf16_t* o, float* lse, int64_t bh, int seq, int hd,
                     float scale, hipStream_t s) {
  dim3 grid((seq + 63) / 64, bh);
  if (hd == 64)
   hipLaunchKernelGGL(( attn_fwd_k<4>), dim3(grid), dim3(256), 0, s, q, k, v, o, lse, seq, scale);
  else if (hd == 128)
   hipLaunchKernelGGL(( attn_fwd_k<8>), dim3(grid), dim3(256), 0, s, q, k, v, o, lse, seq, scale);
  else if (hd == 32)
   hipLaunchKernelGGL(( attn_fwd_k<2>), dim3(grid), dim3(256), 0, s, q, k, v, o, lse, seq, scale);
}

void launch_attn_delta(const bf16_t* dout, const bf16_t* o, float* delta,
                       int64_t rows, int hd, hipStream_t s) {
  int64_t want = (rows + 3) / 4;
  const int grid = int(want < 4096 ? (want > 0 ? want : 1) : 4096);
 hipLaunchKernelGGL(( attn_delta_k), dim3(grid), dim3(256), 0, s, dout, o, delta, rows, hd);
}

void launch_attn_bwd_dq(const bf16_t* dout, const bf16_t* q, const bf16_t* k,
                        const bf16_t* v, const float* lse, const float* delta,
                        bf16_t* dq, int64_t bh, int seq, int hd, float scale,
                        hipStream_t s) {
  dim3 grid((seq + 63) / 64, bh);
  if (hd == 64)
   hipLaunchKernelGGL(( attn_bwd_dq_k<4>), dim3(grid), dim3(256), 0, s, dout, q, k, v, lse, delta, dq, seq,
                                          scale);
  else if (hd == 128)
   hipLaunchKernelGGL(( attn_bwd_dq_k<8>), dim3(grid), dim3(256), 0, s, dout, q, k, v, lse, delta, dq, seq,
                                          scale);
  else if (hd == 32)
   hipLaunchKernelGGL(( attn_bwd_dq_k<2>), dim3(grid), dim3(256), 0, s, dout, q, k, v, lse, delta, dq, seq,
                                          scale);
}

void launch_attn_bwd_dkv(const bf16_t* dout, const bf16_t* q,
                         const bf16_t* k, const bf16_t* v, const float* lse,
                         const float* delta, bf16_t* dk, bf16_t* dv,
                         int64_t bh, int seq, int hd, float scale,
                         hipStream_t s) {
  dim3 grid((seq + 63) / 64, bh);
  if (hd == 64)
   hipLaunchKernelGGL(( attn_bwd_dkv_k<4>), dim3(grid), dim3(256), 0, s, dout, q, k, v, lse, delta, dk, dv,
                                           seq, scale);
  else if (hd == 128)
   hipLaunchKernelGGL(( attn_bwd_dkv_k<8>), dim3(grid), dim3(256), 0, s, dout, q, k, v, lse, delta, dk, dv,
                                           seq, scale);
  else if (hd == 32)
   hipLaunchKernelGGL(( attn_bwd_dkv_k<2>), dim3(grid), dim3(256), 0, s, dout, q, k, v, lse, delta, dk, dv,
                                           seq, scale);
}

void launch_mfma_probe_16(const bf16_t* A, const bf16_t* B, float* D,
                          hipStream_t s) {
 hipLaunchKernelGGL(( mfma_probe16_k), dim3(1), dim3(64), 0, s, A, B, D);
}
void launch_mfma_probe_32(const bf16_t* A, const bf16_t* B, float* D,
                          hipStream_t s) {
 hipLaunchKernelGGL(( mfma_probe32_k), dim3(1), dim3(64), 0, s, A, B, D);
}
