// Fused token+position embedding gather (fwd) and scatter-add (bwd).
// Replaces the reference's transformers wte/wpe lookup.
// fwd: out[t,:] = wte[ids[t]] + wpe[t % seq]; one wave per token row.
// bwd: dwte = fp32 atomicAdd scatter (random token ids: low contention);
// dwpe is NOT scattered here — every batch row hits the same seq_len
// positions (B-way contention per element, measured dominant) — it's a
// plain batch-dim column reduction handled by launch_colsum in the
// binding. fp32 accumulate, cast after.
#include "dta_common.h"
#include "dta_kernels.h"

namespace {

constexpr int ROW_WAVES = 4;

template <bool HAS_WPE>
__global__ void emb_fwd_k(const int64_t* __restrict__ ids,
                          const ushort* __restrict__ wte,
                          const ushort* __restrict__ wpe,
                          ushort* __restrict__ out, int64_t n_tok,
                          int seq_len, int dim,
                          const int* __restrict__ pos_p) {
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  const int nchunk = dim >> 3;
  // pos_p: device position offset (graph-replayable decode: wpe row =
  // *pos_p + in-sequence index instead of a host-sliced wpe)
  const int pos0 = (HAS_WPE && pos_p) ? *pos_p : 0;
  for (int64_t t = int64_t(blockIdx.x) * ROW_WAVES + wid; t < n_tok;
       t += int64_t(gridDim.x) * ROW_WAVES) {
    const int64_t id = ids[t];
    const ushort* te = wte + id * dim;
    const ushort* pe =
        HAS_WPE ? wpe + int64_t(pos0 + t % seq_len) * dim : nullptr;
    ushort* o = out + t * dim;
    for (int c = lane; c < nchunk; c += 64) {
      s16x8 v = *reinterpret_cast<const s16x8*>(te + c * 8);
      if (HAS_WPE) {
        s16x8 p = *reinterpret_cast<const s16x8*>(pe + c * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          v[j] = f2bf(bf2f(ushort(v[j])) + bf2f(ushort(p[j])));
      }
      *reinterpret_cast<s16x8*>(o + c * 8) = v;
    }
  }
}

template <bool HAS_WPE>
__global__ void emb_bwd_k(const ushort* __restrict__ dy,
                          const int64_t* __restrict__ ids,
                          float* __restrict__ dwte, float* __restrict__ dwpe,
                          int64_t n_tok, int seq_len, int dim) {
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  for (int64_t t = int64_t(blockIdx.x) * ROW_WAVES + wid; t < n_tok;
       t += int64_t(gridDim.x) * ROW_WAVES) {
    const int64_t id = ids[t];
    const ushort* g = dy + t * dim;
    float* te = dwte + id * dim;
    (void)seq_len; (void)dwpe;
    for (int c = lane; c * 2 < dim; c += 64) {
      atomicAdd(te + c * 2, bf2f(g[c * 2]));
      atomicAdd(te + c * 2 + 1, bf2f(g[c * 2 + 1]));
    }
  }
}

// KV-cache append at a DEVICE position (graph-replayable decode): write
// the new key/value rows ([B, Hk, D], batch stride knb over a strided
// qkv-slice view) into the caches at row *pos_p.
__global__ void kv_append_k(const ushort* __restrict__ kn,
                            const ushort* __restrict__ vn, int64_t knb,
                            ushort* __restrict__ kc, ushort* __restrict__ vc,
                            const int* __restrict__ pos_p, int B, int Hk,
                            int D, int64_t cb, int64_t ch) {
  const int pos = *pos_p;
  const int64_t n = int64_t(B) * Hk * D;
  for (int64_t i = int64_t(blockIdx.x) * blockDim.x + threadIdx.x; i < n;
       i += int64_t(gridDim.x) * blockDim.x) {
    const int d = int(i % D);
    const int h = int((i / D) % Hk);
    const int b = int(i / (int64_t(D) * Hk));
    const int64_t src = int64_t(b) * knb + int64_t(h) * D + d;
    const int64_t dst = int64_t(b) * cb + int64_t(h) * ch +
                        int64_t(pos) * D + d;
    kc[dst] = kn[src];
    vc[dst] = vn[src];
  }
}

__global__ void i32_inc_k(int* p) { ++(*p); }

}  // namespace

void launch_kv_append(const bf16_t* kn, const bf16_t* vn, int64_t knb,
                      bf16_t* kc, bf16_t* vc, const int* pos_p, int B,
                      int Hk, int D, int64_t cb, int64_t ch,
                      hipStream_t s) {
  const int64_t n = int64_t(B) * Hk * D;
  const int grid = int((n + 255) / 256) < 1024 ? int((n + 255) / 256) : 1024;
  kv_append_k<<<grid > 0 ? grid : 1, 256, 0, s>>>(kn, vn, knb, kc, vc,
                                                  pos_p, B, Hk, D, cb, ch);
}

void launch_i32_inc(int* p, hipStream_t s) { i32_inc_k<<<1, 1, 0, s>>>(p); }

void launch_embedding_fwd(const int64_t* ids, const bf16_t* wte,
                          const bf16_t* wpe, bf16_t* out, int64_t n_tok,
                          int seq_len, int dim, bool has_wpe,
                          const int* pos_p, hipStream_t s) {
  int64_t want = (n_tok + ROW_WAVES - 1) / ROW_WAVES;
  const int grid = int(want < 4096 ? (want > 0 ? want : 1) : 4096);
  if (has_wpe)
    emb_fwd_k<true><<<grid, 256, 0, s>>>(ids, wte, wpe, out, n_tok, seq_len,
                                         dim, pos_p);
  else
    emb_fwd_k<false><<<grid, 256, 0, s>>>(ids, wte, nullptr, out, n_tok,
                                          seq_len, dim, nullptr);
}

void launch_embedding_bwd(const bf16_t* dy, const int64_t* ids,
                          float* dwte_f32, float* dwpe_f32, int64_t n_tok,
                          int seq_len, int dim, bool has_wpe, hipStream_t s) {
  int64_t want = (n_tok + ROW_WAVES - 1) / ROW_WAVES;
  const int grid = int(want < 2048 ? (want > 0 ? want : 1) : 2048);
  emb_bwd_k<false><<<grid, 256, 0, s>>>(dy, ids, dwte_f32, nullptr, n_tok,
                                        seq_len, dim);
  if (has_wpe) {
    // dwpe[s,e] = sum_b dy[b,s,e]: batch-dim column reduction over the
    // [B, seq_len*dim] view (workspace + config from the colsum machinery)
    (void)dwpe_f32;  // launched by the binding via launch_colsum
  }
}
