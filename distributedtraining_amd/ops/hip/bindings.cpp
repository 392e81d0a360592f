// Torch glue for the gfx950 kernel library (_dta_hip extension).
// All tensor-shape/dtype validation lives here; kernels get raw pointers.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <cmath>
#include <vector>

#include "dta_kernels.h"

namespace {

using torch::Tensor;

inline hipStream_t stream() {
  return at::hip::getCurrentHIPStream().stream();
}
inline const bf16_t* bfp(const Tensor& t) {
  return reinterpret_cast<const bf16_t*>(t.data_ptr());
}
inline bf16_t* bfp_mut(Tensor& t) {
  return reinterpret_cast<bf16_t*>(t.data_ptr());
}

void check_bf16(const Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}
void check_f32(const Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be fp32");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

// ---- norms ----------------------------------------------------------------
// drop params shared by the norm bindings: p > 0 enables the fused
// residual-branch dropout (counter RNG; thr16 quantization as elsewhere)
struct DropArgs {
  const unsigned long long* rng = nullptr;
  unsigned long long site = 0;
  unsigned int thr16 = 0;
  float ik = 1.0f;
};
static DropArgs drop_args(const Tensor& rng, int64_t site, double p) {
  DropArgs d;
  if (p > 0.0) {
    TORCH_CHECK(p < 1.0, "dropout p must be < 1");
    TORCH_CHECK(rng.numel() == 1 && rng.scalar_type() == torch::kInt64 &&
                    rng.is_cuda(), "rng must be an int64 [1] GPU counter");
    d.rng = reinterpret_cast<const unsigned long long*>(
        rng.data_ptr<int64_t>());
    d.site = (unsigned long long)site;
    unsigned int thr = (unsigned int)std::ceil(p * 65536.0);
    if (thr > 65535u) thr = 65535u;
    d.thr16 = thr;
    d.ik = float(65536.0 / (65536.0 - double(thr)));
  }
  return d;
}

// res: empty tensor => plain LN; else fused residual (returns the bf16
// sum as an extra output feeding the ongoing residual stream). p > 0:
// the branch is dropout-ed BEFORE the add (sum = x + drop(res)).
std::vector<Tensor> layernorm_fwd(Tensor x, Tensor res, Tensor w, Tensor b,
                                  double eps, Tensor rng, int64_t site,
                                  double p) {
  check_bf16(x, "x"); check_bf16(w, "w"); check_bf16(b, "b");
  const int64_t rows = x.size(0);
  const int cols = int(x.size(1));
  TORCH_CHECK(cols % 8 == 0, "cols must be a multiple of 8");
  TORCH_CHECK(cols <= 8192,
              "norm kernels support cols <= 8192 (register-resident row)");
  const bool has_res = res.numel() > 0;
  if (has_res) check_bf16(res, "res");
  TORCH_CHECK(p == 0.0 || has_res, "dropout needs the residual branch");
  const DropArgs da = drop_args(rng, site, p);
  auto y = torch::empty_like(x);
  auto sum = has_res ? torch::empty_like(x) : torch::empty({0}, x.options());
  auto mean = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  auto rstd = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  launch_layernorm_fwd(bfp(x), has_res ? bfp(res) : nullptr,
                       has_res ? bfp_mut(sum) : nullptr, bfp(w), bfp(b),
                       bfp_mut(y), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), rows, cols, float(eps),
                       da.rng, da.site, da.thr16, da.ik, stream());
  return {y, mean, rstd, sum};
}

// ds: empty tensor => plain; else the gradient arriving on the sum
// stream, folded into dx (dx = ds + dLN/dx). p > 0: also emits
// dres = dx ⊙ mask/(1-p) (4th output) for the dropped branch.
std::vector<Tensor> layernorm_bwd(Tensor dy, Tensor ds, Tensor x, Tensor w,
                                  Tensor mean, Tensor rstd, Tensor rng,
                                  int64_t site, double p) {
  check_bf16(dy, "dy"); check_bf16(x, "x"); check_bf16(w, "w");
  const int64_t rows = x.size(0);
  const int cols = int(x.size(1));
  TORCH_CHECK(cols % 8 == 0, "cols must be a multiple of 8");
  TORCH_CHECK(cols <= 8192,
              "norm kernels support cols <= 8192 (register-resident row)");
  const bool has_ds = ds.numel() > 0;
  if (has_ds) check_bf16(ds, "ds");
  const DropArgs da = drop_args(rng, site, p);
  auto dx = torch::empty_like(x);
  auto dres = da.thr16 ? torch::empty_like(x)
                       : torch::empty({0}, x.options());
  auto f32 = x.options().dtype(torch::kFloat32);
  auto dw32 = torch::zeros({cols}, f32);
  auto db32 = torch::zeros({cols}, f32);
  const int stripes = dta_norm_bwd_stripes(rows, cols);
  auto part = torch::empty({2, stripes, cols}, f32);
  launch_layernorm_bwd(bfp(dy), has_ds ? bfp(ds) : nullptr, bfp(x), bfp(w),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       bfp_mut(dx), da.thr16 ? bfp_mut(dres) : nullptr,
                       dw32.data_ptr<float>(), db32.data_ptr<float>(),
                       part[0].data_ptr<float>(), part[1].data_ptr<float>(),
                       stripes, rows, cols, da.rng, da.site, da.thr16,
                       da.ik, stream());
  return {dx, dw32, db32, dres};  // fp32 grads: ops layer accumulates or casts
}

std::vector<Tensor> rmsnorm_fwd(Tensor x, Tensor res, Tensor w,
                                double eps, Tensor rng, int64_t site,
                                double p) {
  check_bf16(x, "x"); check_bf16(w, "w");
  const int64_t rows = x.size(0);
  const int cols = int(x.size(1));
  TORCH_CHECK(cols % 8 == 0, "cols must be a multiple of 8");
  TORCH_CHECK(cols <= 8192,
              "norm kernels support cols <= 8192 (register-resident row)");
  const bool has_res = res.numel() > 0;
  if (has_res) check_bf16(res, "res");
  TORCH_CHECK(p == 0.0 || has_res, "dropout needs the residual branch");
  const DropArgs da = drop_args(rng, site, p);
  auto y = torch::empty_like(x);
  auto sum = has_res ? torch::empty_like(x) : torch::empty({0}, x.options());
  auto rstd = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  launch_rmsnorm_fwd(bfp(x), has_res ? bfp(res) : nullptr,
                     has_res ? bfp_mut(sum) : nullptr, bfp(w), bfp_mut(y),
                     rstd.data_ptr<float>(), rows, cols, float(eps),
                     da.rng, da.site, da.thr16, da.ik, stream());
  return {y, rstd, sum};
}

std::vector<Tensor> rmsnorm_bwd(Tensor dy, Tensor ds, Tensor x, Tensor w,
                                Tensor rstd, Tensor rng, int64_t site,
                                double p) {
  check_bf16(dy, "dy"); check_bf16(x, "x"); check_bf16(w, "w");
  const int64_t rows = x.size(0);
  const int cols = int(x.size(1));
  TORCH_CHECK(cols % 8 == 0, "cols must be a multiple of 8");
  TORCH_CHECK(cols <= 8192,
              "norm kernels support cols <= 8192 (register-resident row)");
  const bool has_ds = ds.numel() > 0;
  if (has_ds) check_bf16(ds, "ds");
  const DropArgs da = drop_args(rng, site, p);
  auto dx = torch::empty_like(x);
  auto dres = da.thr16 ? torch::empty_like(x)
                       : torch::empty({0}, x.options());
  auto f32 = x.options().dtype(torch::kFloat32);
  auto dw32 = torch::zeros({cols}, f32);
  const int stripes = dta_norm_bwd_stripes(rows, cols);
  auto part = torch::empty({stripes, cols}, f32);
  launch_rmsnorm_bwd(bfp(dy), has_ds ? bfp(ds) : nullptr, bfp(x), bfp(w),
                     rstd.data_ptr<float>(), bfp_mut(dx),
                     da.thr16 ? bfp_mut(dres) : nullptr,
                     dw32.data_ptr<float>(), part.data_ptr<float>(),
                     stripes, rows, cols, da.rng, da.site, da.thr16, da.ik,
                     stream());
  return {dx, dw32, dres};  // fp32 grad
}

// ---- elementwise ----------------------------------------------------------
Tensor gelu_fwd(Tensor x) {
  auto y = torch::empty_like(x);
  if (x.scalar_type() == torch::kBFloat16) {
    check_bf16(x, "x");
    launch_gelu_fwd(bfp(x), bfp_mut(y), x.numel(), stream());
  } else {
    check_f32(x, "x");
    launch_gelu_fwd_f32(x.data_ptr<float>(), y.data_ptr<float>(), x.numel(),
                        stream());
  }
  return y;
}
Tensor gelu_bwd(Tensor dy, Tensor x) {
  auto dx = torch::empty_like(x);
  if (x.scalar_type() == torch::kBFloat16) {
    check_bf16(x, "x"); check_bf16(dy, "dy");
    launch_gelu_bwd(bfp(dy), bfp(x), bfp_mut(dx), x.numel(), stream());
  } else {
    launch_gelu_bwd_f32(dy.data_ptr<float>(), x.data_ptr<float>(),
                        dx.data_ptr<float>(), x.numel(), stream());
  }
  return dx;
}
Tensor swiglu_fwd(Tensor g, Tensor u) {
  check_bf16(g, "gate"); check_bf16(u, "up");
  auto y = torch::empty_like(g);
  launch_swiglu_fwd(bfp(g), bfp(u), bfp_mut(y), g.numel(), stream());
  return y;
}
// counter-based dropout: y = x ⊙ mask / keep. The same call with the same
// (rng value, site) regenerates the identical mask — backward is this very
// function applied to dy (no stored mask).
Tensor dropout_apply(Tensor x, Tensor rng, int64_t site, double p) {
  check_bf16(x, "x");
  TORCH_CHECK(p > 0.0 && p < 1.0, "dropout p must be in (0,1)");
  TORCH_CHECK(rng.numel() == 1 && rng.scalar_type() == torch::kInt64 &&
                  rng.is_cuda(), "rng must be an int64 [1] GPU counter");
  unsigned int thr = (unsigned int)std::ceil(p * 65536.0);
  if (thr > 65535u) thr = 65535u;
  const float inv_keep = float(65536.0 / (65536.0 - double(thr)));
  auto y = torch::empty_like(x);
  launch_dropout(bfp(x), bfp_mut(y), x.numel(),
                 reinterpret_cast<const unsigned long long*>(
                     rng.data_ptr<int64_t>()),
                 (unsigned long long)site, thr, inv_keep, stream());
  return y;
}

// dst must be a CONTIGUOUS bf16 view (a flat-grad slice); src fp32
void accum_f32_into_bf16(Tensor dst, Tensor src) {
  check_bf16(dst, "dst"); check_f32(src, "src");
  TORCH_CHECK(dst.numel() == src.numel(), "size mismatch");
  launch_accum_f32_bf16(bfp_mut(dst), src.data_ptr<float>(), dst.numel(),
                        stream());
}

void rng_tick(Tensor ctr) {
  TORCH_CHECK(ctr.numel() == 1 && ctr.scalar_type() == torch::kInt64 &&
                  ctr.is_cuda(), "ctr must be an int64 [1] GPU counter");
  launch_rng_tick(reinterpret_cast<unsigned long long*>(
                      ctr.data_ptr<int64_t>()), stream());
}

std::vector<Tensor> swiglu_bwd(Tensor dy, Tensor g, Tensor u) {
  check_bf16(dy, "dy");
  auto dg = torch::empty_like(g);
  auto du = torch::empty_like(u);
  launch_swiglu_bwd(bfp(dy), bfp(g), bfp(u), bfp_mut(dg), bfp_mut(du),
                    g.numel(), stream());
  return {dg, du};
}

// ---- flat plane ------------------------------------------------------------
void delta_sub(Tensor w, Tensor base, Tensor out) {
  check_f32(w, "w"); check_f32(base, "base"); check_f32(out, "out");
  launch_delta_sub(w.data_ptr<float>(), base.data_ptr<float>(),
                   out.data_ptr<float>(), w.numel(), stream());
}
void axpy(Tensor w, Tensor x, double alpha) {
  check_f32(w, "w"); check_f32(x, "x");
  launch_axpy(w.data_ptr<float>(), x.data_ptr<float>(), float(alpha),
              w.numel(), stream());
}
bool has_nan(Tensor x) {
  check_f32(x, "x");
  auto flag = torch::zeros({1}, x.options().dtype(torch::kInt32));
  launch_nan_any(x.data_ptr<float>(), x.numel(), flag.data_ptr<int>(),
                 stream());
  return flag.item<int>() != 0;
}
double l2norm_sq(Tensor x) {
  check_f32(x, "x");
  auto out = torch::zeros({1}, x.options());
  launch_l2norm_sq(x.data_ptr<float>(), x.numel(), out.data_ptr<float>(),
                   stream());
  return out.item<float>();
}

// bc: empty tensor => host bias-correction from 'step'; else a fp32[2]
// device buffer maintained by adamw_tick (hipGraph-capturable path).
void adamw_step(Tensor master, Tensor grad, Tensor m, Tensor v,
                Tensor out_bf16, int64_t step, double lr, double beta1,
                double beta2, double eps, double wd, Tensor bc) {
  check_f32(master, "master"); check_f32(m, "m"); check_f32(v, "v");
  const bool gb = grad.scalar_type() == torch::kBFloat16;
  bf16_t* outp = out_bf16.numel() ? bfp_mut(out_bf16) : nullptr;
  const float* bcp = bc.numel() ? bc.data_ptr<float>() : nullptr;
  launch_adamw(master.data_ptr<float>(), grad.data_ptr(), gb,
               m.data_ptr<float>(), v.data_ptr<float>(), outp, int(step),
               bcp, float(lr), float(beta1), float(beta2), float(eps),
               float(wd), master.numel(), stream());
}

void adamw_tick(Tensor t, Tensor bc, double beta1, double beta2) {
  TORCH_CHECK(t.scalar_type() == torch::kInt32 && t.is_cuda(), "t int32 gpu");
  check_f32(bc, "bc");
  launch_adamw_tick(t.data_ptr<int>(), bc.data_ptr<float>(), float(beta1),
                    float(beta2), stream());
}

// column sum of a [rows, cols] bf16 matrix -> fp32 [cols] (bias gradient)
Tensor colsum(Tensor x) {
  check_bf16(x, "x");
  TORCH_CHECK(x.dim() == 2 && x.size(1) % 8 == 0,
              "x must be [rows, cols] with cols % 8 == 0");
  const int64_t rows = x.size(0);
  const int cols = int(x.size(1));
  const int stripes = dta_colred_stripes(rows, cols);
  auto f32 = x.options().dtype(torch::kFloat32);
  auto part = torch::empty({stripes, cols}, f32);
  auto out = torch::zeros({cols}, f32);
  launch_colsum(bfp(x), part.data_ptr<float>(), out.data_ptr<float>(), rows,
                cols, stripes, stream());
  return out;
}

// ---- merge plane -----------------------------------------------------------
void weighted_merge(Tensor base, Tensor deltas, Tensor W, Tensor offsets,
                    Tensor out) {
  check_f32(base, "base"); check_f32(deltas, "deltas"); check_f32(out, "out");
  auto Wc = W.to(base.device(), torch::kFloat32).contiguous();
  auto offs = offsets.to(base.device()).contiguous();
  const int n_models = int(deltas.size(0));
  const int n_segs = int(offsets.numel() - 1);
  launch_weighted_merge(base.data_ptr<float>(), deltas.data_ptr<float>(),
                        Wc.data_ptr<float>(), offs.data_ptr<int64_t>(),
                        n_models, n_segs, base.numel(),
                        out.data_ptr<float>(), stream());
}

Tensor grad_merge_weights(Tensor g, Tensor base, Tensor deltas, Tensor merged,
                          Tensor offsets) {
  check_f32(base, "base"); check_f32(deltas, "deltas");
  const int n_models = int(deltas.size(0));
  const int n_segs = int(offsets.numel() - 1);
  const int64_t P = base.numel();
  // build boundary-aligned chunks host-side (<=1M elements each)
  auto offs_cpu = offsets.to(torch::kCPU).contiguous();
  const int64_t* op = offs_cpu.data_ptr<int64_t>();
  std::vector<int64_t> chunks;
  constexpr int64_t CH = 1 << 20;
  for (int j = 0; j < n_segs; ++j)
    for (int64_t s0 = op[j]; s0 < op[j + 1]; s0 += CH) {
      chunks.push_back(s0);
      chunks.push_back(std::min(s0 + CH, op[j + 1]));
      chunks.push_back(j);
    }
  const int n_chunks = int(chunks.size() / 3);
  auto ch = torch::from_blob(chunks.data(), {int64_t(chunks.size())},
                             torch::kInt64)
                .to(base.device());
  auto gw = torch::zeros({n_models, n_segs},
                         base.options().dtype(torch::kFloat32));
  const bool gb = g.scalar_type() == torch::kBFloat16;
  launch_grad_merge_weights(g.data_ptr(), gb, base.data_ptr<float>(),
                            deltas.data_ptr<float>(),
                            merged.data_ptr<float>(),
                            ch.data_ptr<int64_t>(), n_models, n_chunks, P,
                            gw.data_ptr<float>(), n_segs, stream());
  return gw;
}

// ---- CE --------------------------------------------------------------------
std::vector<Tensor> ce_fwd(Tensor logits, Tensor targets,
                           int64_t ignore_index) {
  check_bf16(logits, "logits");
  TORCH_CHECK(targets.scalar_type() == torch::kInt64, "targets int64");
  const int64_t rows = logits.size(0), vocab = logits.size(1);
  auto lse = torch::empty({rows}, logits.options().dtype(torch::kFloat32));
  auto loss = torch::zeros({1}, logits.options().dtype(torch::kFloat32));
  auto count = torch::zeros({1}, logits.options().dtype(torch::kInt32));
  launch_ce_fwd(bfp(logits), targets.data_ptr<int64_t>(), rows, vocab,
                ignore_index, lse.data_ptr<float>(), loss.data_ptr<float>(),
                count.data_ptr<int>(), stream());
  return {loss.squeeze(0), lse, count.squeeze(0)};
}

// pipelined head-GEMM+CE forward helpers: ce_chunk folds one logits tile
// (a [rows, cols] view with leading dim ld, vocab offset v0) into per-row
// running (m, s) and records the target logit; ce_finalize emits
// lse / loss_sum / count from the state alone.
void ce_chunk(Tensor chunk, Tensor targets, int64_t v0, Tensor tlogit,
              Tensor m_run, Tensor s_run) {
  TORCH_CHECK(chunk.is_cuda() && chunk.scalar_type() == torch::kBFloat16 &&
                  chunk.dim() == 2 && chunk.stride(1) == 1,
              "chunk must be a bf16 [rows, cols] view, contiguous cols");
  check_f32(m_run, "m_run"); check_f32(s_run, "s_run");
  check_f32(tlogit, "tlogit");
  launch_ce_chunk(bfp(chunk), chunk.stride(0), chunk.size(0), chunk.size(1),
                  targets.data_ptr<int64_t>(), v0,
                  tlogit.data_ptr<float>(), m_run.data_ptr<float>(),
                  s_run.data_ptr<float>(), stream());
}

std::vector<Tensor> ce_finalize(Tensor targets, Tensor m_run, Tensor s_run,
                                Tensor tlogit, int64_t ignore_index) {
  const int64_t rows = m_run.numel();
  auto lse = torch::empty({rows}, m_run.options());
  auto loss = torch::zeros({1}, m_run.options());
  auto count = torch::zeros({1}, m_run.options().dtype(torch::kInt32));
  launch_ce_finalize(targets.data_ptr<int64_t>(), m_run.data_ptr<float>(),
                     s_run.data_ptr<float>(), tlogit.data_ptr<float>(),
                     rows, ignore_index, lse.data_ptr<float>(),
                     loss.data_ptr<float>(), count.data_ptr<int>(),
                     stream());
  return {loss.squeeze(0), lse, count.squeeze(0)};
}

// scale_dev: empty tensor => host 'scale' scalar; else a 0/1-dim fp32
// device scalar (dloss/count), keeping backward free of host syncs.
// v0/nt: vocab-tile offset + store policy for the tiled pipeline.
Tensor ce_bwd(Tensor logits, Tensor targets, Tensor lse, Tensor scale_dev,
              double scale, int64_t ignore_index, int64_t v0, bool nt) {
  check_bf16(logits, "logits");
  const int64_t rows = logits.size(0), vocab = logits.size(1);
  auto dl = torch::empty_like(logits);
  const float* scp = scale_dev.numel() ? scale_dev.data_ptr<float>() : nullptr;
  launch_ce_bwd(bfp(logits), targets.data_ptr<int64_t>(),
                lse.data_ptr<float>(), float(scale), scp, ignore_index, v0,
                nt, bfp_mut(dl), rows, vocab, stream());
  return dl;
}

// ---- embedding -------------------------------------------------------------
// pos: optional int32 [1] device offset for the wpe row (graph decode)
Tensor embedding_fwd(Tensor ids, Tensor wte, Tensor wpe, Tensor pos) {
  check_bf16(wte, "wte");
  TORCH_CHECK(ids.scalar_type() == torch::kInt64, "ids int64");
  const bool has_wpe = wpe.numel() > 0;
  const int dim = int(wte.size(1));
  const int seq = int(ids.size(-1));
  const int64_t n_tok = ids.numel();
  auto out_sizes = ids.sizes().vec();
  out_sizes.push_back(dim);
  auto out = torch::empty(out_sizes, wte.options());
  const int* posp = nullptr;
  if (pos.numel()) {
    TORCH_CHECK(pos.scalar_type() == torch::kInt32 && pos.is_cuda(),
                "pos must be an int32 GPU scalar");
    posp = pos.data_ptr<int>();
  }
  launch_embedding_fwd(ids.data_ptr<int64_t>(), bfp(wte),
                       has_wpe ? bfp(wpe) : nullptr, bfp_mut(out), n_tok,
                       seq, dim, has_wpe, posp, stream());
  return out;
}

// new k/v rows ([B, 1, Hk*D] strided slices of the qkv projection) into
// the caches at device row *pos; pos += 1 via i32_inc afterwards
void kv_append(Tensor kn, Tensor vn, Tensor kc, Tensor vc, Tensor pos) {
  TORCH_CHECK(kn.dim() == 3 && kn.size(1) == 1 && kn.stride(2) == 1,
              "kn must be [B,1,F] with contiguous F");
  TORCH_CHECK(vn.stride(0) == kn.stride(0) && vn.stride(2) == 1,
              "vn must share kn's layout (one qkv projection)");
  TORCH_CHECK(pos.scalar_type() == torch::kInt32 && pos.is_cuda(), "pos");
  const int B = int(kc.size(0)), Hk = int(kc.size(1)), D = int(kc.size(3));
  launch_kv_append(bfp(kn), bfp(vn), kn.stride(0), bfp_mut(kc), bfp_mut(vc),
                   pos.data_ptr<int>(), B, Hk, D, kc.stride(0),
                   kc.stride(1), stream());
}

void i32_inc(Tensor t) {
  TORCH_CHECK(t.scalar_type() == torch::kInt32 && t.is_cuda(), "int32 gpu");
  launch_i32_inc(t.data_ptr<int>(), stream());
}

std::vector<Tensor> embedding_bwd(Tensor dy, Tensor ids, int64_t vocab,
                                  int64_t npos) {
  check_bf16(dy, "dy");
  const int dim = int(dy.size(-1));
  const int seq = int(ids.size(-1));
  const int64_t n_tok = ids.numel();
  auto f32 = dy.options().dtype(torch::kFloat32);
  auto dwte = torch::zeros({vocab, dim}, f32);
  auto dwpe = npos ? torch::zeros({npos, dim}, f32) : torch::zeros({0}, f32);
  launch_embedding_bwd(bfp(dy), ids.data_ptr<int64_t>(),
                       dwte.data_ptr<float>(), nullptr, n_tok, seq, dim,
                       false, stream());
  if (npos) {
    // dwpe[s,:] = sum_b dy[b,s,:] — batch-dim column reduction over the
    // [B, seq*dim] view (B-way same-address atomics measured dominant in
    // the scatter version)
    const int64_t B = n_tok / seq;
    const int pcols = seq * dim;
    TORCH_CHECK(pcols % 8 == 0, "seq*dim must be a multiple of 8");
    const int stripes = dta_colred_stripes(B, pcols);
    auto part = torch::empty({stripes, pcols}, f32);
    launch_colsum(bfp(dy), part.data_ptr<float>(), dwpe.data_ptr<float>(),
                  B, pcols, stripes, stream());
  }
  return {dwte, dwpe};  // fp32: ops layer accumulates or casts
}

// ---- serving gemv ----------------------------------------------------------
Tensor gemv(Tensor x, Tensor w, Tensor bias) {
  check_bf16(x, "x"); check_bf16(w, "w");
  TORCH_CHECK(x.dim() == 2 && w.dim() == 2 && x.size(1) == w.size(1),
              "x [M,K], w [N,K]");
  const int M = int(x.size(0));
  TORCH_CHECK(M >= 1 && M <= 4, "gemv is for M <= 4");
  const int64_t N = w.size(0);
  const int K = int(x.size(1));
  TORCH_CHECK(K % 8 == 0, "gemv needs K % 8 == 0 (16 B row alignment)");
  const bf16_t* bp = nullptr;
  if (bias.numel()) {
    check_bf16(bias, "bias");
    bp = bfp(bias);
  }
  auto y = torch::empty({M, N}, x.options());
  launch_gemv(bfp(x), bfp(w), bp, bfp_mut(y), M, N, K, stream());
  return y;
}

// ---- rope ------------------------------------------------------------------
Tensor rope_apply(Tensor x, Tensor cos_t, Tensor sin_t, bool backward,
                  const Tensor& pos) {
  check_bf16(x, "x");
  // x: [B,H,S,D] or [BH,S,D]
  const int hd = int(x.size(-1));
  const int seq = int(x.size(-2));
  const int64_t bh = x.numel() / (int64_t(seq) * hd);
  const int* posp = nullptr;
  if (pos.numel()) {
    TORCH_CHECK(pos.scalar_type() == torch::kInt32 && pos.is_cuda(),
                "pos must be an int32 GPU scalar");
    posp = pos.data_ptr<int>();
  }
  auto y = torch::empty_like(x);
  auto c = cos_t.contiguous();
  auto sn = sin_t.contiguous();
  launch_rope(bfp(x), c.data_ptr<float>(), sn.data_ptr<float>(), bfp_mut(y),
              bh, seq, hd, backward, posp, stream());
  return y;
}
Tensor rope_fwd(Tensor x, Tensor c, Tensor s, Tensor pos) {
  return rope_apply(x, c, s, false, pos);
}
Tensor rope_bwd(Tensor x, Tensor c, Tensor s, Tensor pos) {
  return rope_apply(x, c, s, true, pos);
}

// ---- attention -------------------------------------------------------------
// q/k/v: [B,H(,Hk),S,D] views, innermost stride 1 (checked). Output o (and
// dq) are allocated [B,S,H,D] contiguous — the caller permutes back, so a
// transformer layer does zero transpose copies around attention.
static void check_attn_view(const Tensor& t, const char* n) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kBFloat16, n,
              " must be bf16 on GPU");
  TORCH_CHECK(t.dim() == 4 && t.stride(3) == 1, n,
              " must be a [B,H,S,D] view with contiguous D");
}

// kvlen: int32 [B] valid-key prefix per batch row (empty = no mask);
// rng: int64 [1] device step counter + site/p: attention-prob dropout
// (p == 0 disables). thr16 quantizes p to 1/65536 steps; inv_keep uses the
// REALIZED keep probability so the expectation stays unbiased.
static void set_mask_drop(AttnGeom& g, const Tensor& kvlen, const Tensor& rng,
                          int64_t site, double p, int64_t B) {
  if (kvlen.numel() > 0) {
    TORCH_CHECK(kvlen.scalar_type() == torch::kInt32 && kvlen.is_cuda() &&
                    kvlen.is_contiguous() && kvlen.numel() == B,
                "kvlen must be a contiguous int32 [B] GPU tensor");
    g.kvlen = kvlen.data_ptr<int>();
  }
  if (p > 0.0) {
    TORCH_CHECK(p < 1.0, "dropout p must be < 1");
    TORCH_CHECK(rng.numel() == 1 && rng.scalar_type() == torch::kInt64 &&
                    rng.is_cuda(),
                "rng must be an int64 [1] GPU counter");
    g.rng = reinterpret_cast<const unsigned long long*>(
        rng.data_ptr<int64_t>());
    g.site = (unsigned long long)site;
    unsigned int thr = (unsigned int)std::ceil(p * 65536.0);
    if (thr > 65535u) thr = 65535u;
    g.thr16 = thr;
    g.inv_keep = float(65536.0 / (65536.0 - double(thr)));
  }
}

static AttnGeom make_geom(const Tensor& q, const Tensor& k, const Tensor& v,
                          const Tensor& o_bshd, const Tensor& dout,
                          double scale) {
  AttnGeom g{};
  g.B = int(q.size(0)); g.H = int(q.size(1));
  g.seq = int(q.size(2)); g.hd = int(q.size(3));
  const int hk = int(k.size(1));
  TORCH_CHECK(g.H % hk == 0, "H must be a multiple of H_kv");
  g.grp = g.H / hk;
  g.scale = float(scale);
  g.qb = q.stride(0); g.qh = q.stride(1); g.qs = q.stride(2);
  g.kb = k.stride(0); g.kh = k.stride(1); g.ks = k.stride(2);
  g.vb = v.stride(0); g.vh = v.stride(1); g.vs = v.stride(2);
  // o_bshd is [B,S,H,D]: strides (batch, head, seq) = (0th, 2nd, 1st)
  g.ob = o_bshd.stride(0); g.oh = o_bshd.stride(2); g.os_ = o_bshd.stride(1);
  if (dout.defined()) {
    g.db_ = dout.stride(0); g.dh = dout.stride(1); g.ds = dout.stride(2);
  }
  return g;
}

std::vector<Tensor> attn_fwd(Tensor q, Tensor k, Tensor v, double scale,
                             Tensor kvlen, Tensor rng, int64_t site,
                             double p) {
  check_attn_view(q, "q"); check_attn_view(k, "k"); check_attn_view(v, "v");
  const int hd = int(q.size(3));
  TORCH_CHECK(hd == 32 || hd == 64 || hd == 128, "head dim must be 32/64/128");
  const int64_t B = q.size(0), H = q.size(1), S = q.size(2);
  auto o = torch::empty({B, S, H, hd}, q.options());
  auto lse = torch::empty({B * H, S}, q.options().dtype(torch::kFloat32));
  AttnGeom geo = make_geom(q, k, v, o, Tensor(), scale);
  set_mask_drop(geo, kvlen, rng, site, p, B);
  launch_attn_fwd(bfp(q), bfp(k), bfp(v), bfp_mut(o),
                  lse.data_ptr<float>(), geo, stream());
  return {o, lse};  // caller views o as [B,H,S,D] via permute(0,2,1,3)
}

std::vector<Tensor> attn_bwd(Tensor dout, Tensor q, Tensor k, Tensor v,
                             Tensor o_bshd, Tensor lse, double scale,
                             Tensor kvlen, Tensor rng, int64_t site,
                             double p) {
  check_attn_view(dout, "dout");
  const int64_t B = q.size(0), H = q.size(1), S = q.size(2);
  const int hd = int(q.size(3));
  const int64_t Hk = k.size(1);
  AttnGeom geo = make_geom(q, k, v, o_bshd, dout, scale);
  set_mask_drop(geo, kvlen, rng, site, p, B);
  auto delta = torch::empty({B * H, S}, lse.options());
  auto dq = torch::empty({B, S, H, hd}, q.options());
  launch_attn_bwd_dq(bfp(dout), bfp(q), bfp(k), bfp(v), bfp(o_bshd),
                     o_bshd.stride(0), o_bshd.stride(2), o_bshd.stride(1),
                     lse.data_ptr<float>(), delta.data_ptr<float>(),
                     bfp_mut(dq), geo, stream());
  if (geo.grp == 1) {
    // direct bf16 store into [B,S,Hk,D]; caller permutes to [B,Hk,S,D]
    auto dk = torch::empty({B, S, Hk, hd}, q.options());
    auto dv = torch::empty({B, S, Hk, hd}, q.options());
    geo.gkb = dk.stride(0); geo.gkh = dk.stride(2); geo.gks = dk.stride(1);
    launch_attn_bwd_dkv(bfp(dout), bfp(q), bfp(k), bfp(v),
                        lse.data_ptr<float>(), delta.data_ptr<float>(),
                        nullptr, nullptr, bfp_mut(dk), bfp_mut(dv), geo,
                        stream());
    return {dq, dk.permute({0, 2, 1, 3}), dv.permute({0, 2, 1, 3})};
  }
  // GQA: accumulate in fp32 [B,Hk,S,D] (q heads fold in), then cast
  auto dk32 = torch::zeros({B, Hk, S, hd},
                           q.options().dtype(torch::kFloat32));
  auto dv32 = torch::zeros_like(dk32);
  launch_attn_bwd_dkv(bfp(dout), bfp(q), bfp(k), bfp(v),
                      lse.data_ptr<float>(), delta.data_ptr<float>(),
                      dk32.data_ptr<float>(), dv32.data_ptr<float>(),
                      nullptr, nullptr, geo, stream());
  return {dq, dk32.to(torch::kBFloat16), dv32.to(torch::kBFloat16)};
}

// Packed-QKV backward: dq/dk/dv are stored directly into caller-provided
// [B,S,H,D]/[B,S,Hk,D] strided views of one dqkv buffer (innermost D
// contiguous) — eliminates the split-backward cat and all transpose copies.
void attn_bwd_packed(Tensor dout, Tensor q, Tensor k, Tensor v,
                     Tensor o_bshd, Tensor lse, double scale, Tensor dq_v,
                     Tensor dk_v, Tensor dv_v, Tensor kvlen, Tensor rng,
                     int64_t site, double p) {
  check_attn_view(dout, "dout");
  TORCH_CHECK(dq_v.stride(3) == 1 && dk_v.stride(3) == 1 &&
              dv_v.stride(3) == 1, "dqkv views need contiguous head_dim");
  const int64_t B = q.size(0), H = q.size(1), S = q.size(2);
  const int hd = int(q.size(3));
  const int64_t Hk = k.size(1);
  AttnGeom geo = make_geom(q, k, v, o_bshd, dout, scale);
  set_mask_drop(geo, kvlen, rng, site, p, B);
  auto delta = torch::empty({B * H, S}, lse.options());
  AttnGeom gq = geo;  // dq_v is [B,S,H,D]: (batch, head, seq) strides
  gq.ob = dq_v.stride(0); gq.oh = dq_v.stride(2); gq.os_ = dq_v.stride(1);
  launch_attn_bwd_dq(bfp(dout), bfp(q), bfp(k), bfp(v), bfp(o_bshd),
                     o_bshd.stride(0), o_bshd.stride(2), o_bshd.stride(1),
                     lse.data_ptr<float>(), delta.data_ptr<float>(),
                     bfp_mut(dq_v), gq, stream());
  if (geo.grp == 1) {
    geo.gkb = dk_v.stride(0); geo.gkh = dk_v.stride(2);
    geo.gks = dk_v.stride(1);
    launch_attn_bwd_dkv(bfp(dout), bfp(q), bfp(k), bfp(v),
                        lse.data_ptr<float>(), delta.data_ptr<float>(),
                        nullptr, nullptr, bfp_mut(dk_v), bfp_mut(dv_v), geo,
                        stream());
    return;
  }
  // GQA: fp32 accumulation (q-head fold-in), then cast into the views
  auto dk32 = torch::zeros({B, Hk, S, hd},
                           q.options().dtype(torch::kFloat32));
  auto dv32 = torch::zeros_like(dk32);
  launch_attn_bwd_dkv(bfp(dout), bfp(q), bfp(k), bfp(v),
                      lse.data_ptr<float>(), delta.data_ptr<float>(),
                      dk32.data_ptr<float>(), dv32.data_ptr<float>(),
                      nullptr, nullptr, geo, stream());
  dk_v.copy_(dk32.permute({0, 2, 1, 3}));
  dv_v.copy_(dv32.permute({0, 2, 1, 3}));
}

// Decode attention over a KV cache: q [B,H,D], k/v caches [B,Hk,Lmax,D]
// contiguous, first kv_len positions valid -> o [B,H,D].
Tensor attn_decode(Tensor q, Tensor kc, Tensor vc, int64_t kv_len,
                   double scale, Tensor len_dev) {
  check_bf16(q, "q"); check_bf16(kc, "kcache"); check_bf16(vc, "vcache");
  TORCH_CHECK(q.dim() == 3 && kc.dim() == 4, "q [B,H,D], cache [B,Hk,L,D]");
  const int B = int(q.size(0)), H = int(q.size(1)), hd = int(q.size(2));
  const int Hk = int(kc.size(1));
  TORCH_CHECK(hd == 64 || hd == 128, "decode head dim must be 64/128");
  TORCH_CHECK(H % Hk == 0 && kv_len <= kc.size(2), "bad cache geometry");
  const int* lp = nullptr;
  if (len_dev.numel()) {
    TORCH_CHECK(len_dev.scalar_type() == torch::kInt32 && len_dev.is_cuda(),
                "len_dev must be an int32 GPU scalar");
    lp = len_dev.data_ptr<int>();
  }
  auto o = torch::empty_like(q);
  launch_attn_decode(bfp(q), bfp(kc), bfp(vc), bfp_mut(o), B, H, H / Hk,
                     int(kv_len), lp, hd, kc.stride(0), kc.stride(1),
                     float(scale), stream());
  return o;
}

// ---- mfma self-test --------------------------------------------------------
Tensor mfma_selftest_16(Tensor A, Tensor B) {
  check_bf16(A, "A"); check_bf16(B, "B");
  auto D = torch::zeros({16, 16}, A.options().dtype(torch::kFloat32));
  launch_mfma_probe_16(bfp(A), bfp(B), D.data_ptr<float>(), stream());
  return D;
}
Tensor mfma_selftest_32(Tensor A, Tensor B) {
  check_bf16(A, "A"); check_bf16(B, "B");
  auto D = torch::zeros({32, 32}, A.options().dtype(torch::kFloat32));
  launch_mfma_probe_32(bfp(A), bfp(B), D.data_ptr<float>(), stream());
  return D;
}

}  // namespace

void register_lt_fused(pybind11::module& m);  // lt_fused.cpp

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  register_lt_fused(m);
  m.def("layernorm_fwd", &layernorm_fwd);
  m.def("layernorm_bwd", &layernorm_bwd);
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd);
  m.def("gelu_fwd", &gelu_fwd);
  m.def("gelu_bwd", &gelu_bwd);
  m.def("swiglu_fwd", &swiglu_fwd);
  m.def("swiglu_bwd", &swiglu_bwd);
  m.def("dropout_apply", &dropout_apply);
  m.def("rng_tick", &rng_tick);
  m.def("accum_f32_into_bf16", &accum_f32_into_bf16);
  m.def("delta_sub", &delta_sub);
  m.def("axpy", &axpy);
  m.def("has_nan", &has_nan);
  m.def("l2norm_sq", &l2norm_sq);
  m.def("adamw_step", &adamw_step);
  m.def("adamw_tick", &adamw_tick);
  m.def("colsum", &colsum);
  m.def("weighted_merge", &weighted_merge);
  m.def("grad_merge_weights", &grad_merge_weights);
  m.def("ce_fwd", &ce_fwd);
  m.def("ce_bwd", &ce_bwd);
  m.def("ce_chunk", &ce_chunk);
  m.def("ce_finalize", &ce_finalize);
  m.def("embedding_fwd", &embedding_fwd);
  m.def("embedding_bwd", &embedding_bwd);
  m.def("kv_append", &kv_append);
  m.def("i32_inc", &i32_inc);
  m.def("gemv", &gemv);
  m.def("rope_fwd", &rope_fwd);
  m.def("rope_bwd", &rope_bwd);
  m.def("attn_fwd", &attn_fwd);
  m.def("attn_bwd", &attn_bwd);
  m.def("attn_bwd_packed", &attn_bwd_packed);
  m.def("attn_decode", &attn_decode);
  m.def("mfma_selftest_16", &mfma_selftest_16);
  m.def("mfma_selftest_32", &mfma_selftest_32);
}
