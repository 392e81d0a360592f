// Fused decoupled AdamW over the flat parameter plane: one kernel updates
// fp32 master + m + v and writes the bf16 working copy (replaces the
// reference's torch.optim.AdamW step, training_manager.py:391).
// Memory-bound: ~(3 reads + 3 writes) x 4B + grad; float4-vectorized.
#include "dta_common.h"
#include "dta_kernels.h"

namespace {

// bc_p: optional device pointer {bc1, bc2} maintained by adamw_tick_k so
// the step is hipGraph-capturable (step counter lives on device); if null
// the host-computed bc1/bc2 scalars are used.
template <bool GRAD_BF16, bool WRITE_BF16>
__global__ void adamw_k(float* __restrict__ w, const void* __restrict__ gp,
                        float* __restrict__ m, float* __restrict__ v,
                        ushort* __restrict__ wout, float lr, float beta1,
                        float beta2, float eps, float wd, float bc1,
                        float bc2, const float* __restrict__ bc_p,
                        int64_t n) {
  int64_t i = (int64_t(blockIdx.x) * blockDim.x + threadIdx.x) * 4;
  const int64_t stride = int64_t(gridDim.x) * blockDim.x * 4;
  if (bc_p) { bc1 = bc_p[0]; bc2 = bc_p[1]; }
  const float decay = 1.0f - lr * wd;
  const float step_size = lr / bc1;
  const float inv_bc2 = 1.0f / bc2;
  for (; i + 4 <= n; i += stride) {
    f32x4 wv = *reinterpret_cast<const f32x4*>(w + i);
    f32x4 mv = *reinterpret_cast<const f32x4*>(m + i);
    f32x4 vv = *reinterpret_cast<const f32x4*>(v + i);
    f32x4 gv;
    if (GRAD_BF16) {
      s16x4 gr = *reinterpret_cast<const s16x4*>(
          reinterpret_cast<const ushort*>(gp) + i);
#pragma unroll
      for (int j = 0; j < 4; ++j) gv[j] = bf2f(ushort(gr[j]));
    } else {
      gv = *reinterpret_cast<const f32x4*>(
          reinterpret_cast<const float*>(gp) + i);
    }
    s16x4 bo;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float g = gv[j];
      float mm = fmaf(beta1, mv[j], (1.0f - beta1) * g);
      float vvj = fmaf(beta2, vv[j], (1.0f - beta2) * g * g);
      float ww = wv[j] * decay;
      ww -= step_size * mm / (sqrtf(vvj * inv_bc2) + eps);
      mv[j] = mm; vv[j] = vvj; wv[j] = ww;
      if (WRITE_BF16) bo[j] = f2bf(ww);
    }
    *reinterpret_cast<f32x4*>(w + i) = wv;
    *reinterpret_cast<f32x4*>(m + i) = mv;
    *reinterpret_cast<f32x4*>(v + i) = vv;
    if (WRITE_BF16) *reinterpret_cast<s16x4*>(wout + i) = bo;
  }
  if (i < n && i + 4 > n)
    for (; i < n; ++i) {
      float g = GRAD_BF16 ? bf2f(reinterpret_cast<const ushort*>(gp)[i])
                          : reinterpret_cast<const float*>(gp)[i];
      float mm = fmaf(beta1, m[i], (1.0f - beta1) * g);
      float vvj = fmaf(beta2, v[i], (1.0f - beta2) * g * g);
      float ww = w[i] * decay;
      ww -= step_size * mm / (sqrtf(vvj * inv_bc2) + eps);
      m[i] = mm; v[i] = vvj; w[i] = ww;
      if (WRITE_BF16) wout[i] = f2bf(ww);
    }
}

// Device-side step counter tick: t += 1; bc = {1-beta1^t, 1-beta2^t}.
// Launched (1,1) right before adamw_k inside a captured graph.
__global__ void adamw_tick_k(int* __restrict__ t, float* __restrict__ bc,
                             float beta1, float beta2) {
  const int nt = *t + 1;
  *t = nt;
  bc[0] = 1.0f - powf(beta1, float(nt));
  bc[1] = 1.0f - powf(beta2, float(nt));
}

}  // namespace

void launch_adamw_tick(int* t, float* bc, float beta1, float beta2,
                       hipStream_t s) {
  adamw_tick_k<<<1, 1, 0, s>>>(t, bc, beta1, beta2);
}

void launch_adamw(float* master, const void* grad, bool grad_is_bf16,
                  float* m, float* v, bf16_t* out_bf16, int step,
                  const float* bc_p, float lr, float beta1, float beta2,
                  float eps, float wd, int64_t n, hipStream_t s) {
  const float bc1 = bc_p ? 1.0f : 1.0f - powf(beta1, float(step));
  const float bc2 = bc_p ? 1.0f : 1.0f - powf(beta2, float(step));
  const int block = 256;
  const int grid = elementwise_grid(n, block, 4);
  if (grad_is_bf16) {
    if (out_bf16)
      adamw_k<true, true><<<grid, block, 0, s>>>(master, grad, m, v, out_bf16,
                                                 lr, beta1, beta2, eps, wd,
                                                 bc1, bc2, bc_p, n);
    else
      adamw_k<true, false><<<grid, block, 0, s>>>(master, grad, m, v, nullptr,
                                                  lr, beta1, beta2, eps, wd,
                                                  bc1, bc2, bc_p, n);
  } else {
    if (out_bf16)
      adamw_k<false, true><<<grid, block, 0, s>>>(master, grad, m, v, out_bf16,
                                                  lr, beta1, beta2, eps, wd,
                                                  bc1, bc2, bc_p, n);
    else
      adamw_k<false, false><<<grid, block, 0, s>>>(master, grad, m, v, nullptr,
                                                   lr, beta1, beta2, eps, wd,
                                                   bc1, bc2, bc_p, n);
  }
}
