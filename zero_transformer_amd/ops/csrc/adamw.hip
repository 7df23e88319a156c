// Fused AdamW on a flat fp32 master shard (the ZeRO-1 sharded update,
// reference xmap_train_functions.py:110-123 + optax chain main_zero.py:160-168):
// grad*scale -> element-wise clip(+-clip) -> Adam moments (b2=0.95, bias
// correction) -> decoupled weight decay -> param -= lr*update, emitting the
// bf16 working copy for the all-gather in the same pass.
// Memory-bound over 4 fp32 streams: vectorized f32x4 access.

#include "common.h"

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <torch/extension.h>

namespace {

template <typename GT, bool EMIT_BF16>
__global__ void adamw_kernel(float* __restrict__ p, uint16_t* __restrict__ p_bf16,
                             const GT* __restrict__ g, float* __restrict__ m,
                             float* __restrict__ v, long n, float lr, float beta1,
                             float beta2, float eps, float wd, float clip,
                             float gscale, float inv_bc1, float inv_bc2) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    float gi = to_f32(g[i]) * gscale;
    gi = fminf(fmaxf(gi, -clip), clip);
    float mi = beta1 * m[i] + (1.f - beta1) * gi;
    float vi = beta2 * v[i] + (1.f - beta2) * gi * gi;
    m[i] = mi;
    v[i] = vi;
    float update = (mi * inv_bc1) / (sqrtf(vi * inv_bc2) + eps);
    float pi = p[i];
    if (wd != 0.f) update += wd * pi;
    pi -= lr * update;
    p[i] = pi;
    if (EMIT_BF16) p_bf16[i] = f32_to_bf16(pi);
  }
}

}  // namespace

void adamw_step(at::Tensor p, at::Tensor p_bf16, at::Tensor g, at::Tensor m,
                at::Tensor v, long step, double lr, double beta1, double beta2,
                double eps, double wd, double clip, double grad_scale) {
  TORCH_CHECK(p.is_cuda() && p.is_contiguous() && p.scalar_type() == at::kFloat);
  TORCH_CHECK(m.scalar_type() == at::kFloat && v.scalar_type() == at::kFloat);
  const long n = p.numel();
  const bool emit = p_bf16.numel() == n;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const int block = 256;
  const int grid = capped_grid(n, block, 4096);
  const float inv_bc1 = 1.f / (1.f - powf((float)beta1, (float)step));
  const float inv_bc2 = 1.f / (1.f - powf((float)beta2, (float)step));

#define LAUNCH(GT, EMIT)                                                              \
  hipLaunchKernelGGL((adamw_kernel<GT, EMIT>), dim3(grid), dim3(block), 0, stream,     \
                     p.data_ptr<float>(), (uint16_t*)p_bf16.data_ptr(),                \
                     (const GT*)g.data_ptr(), m.data_ptr<float>(),                     \
                     v.data_ptr<float>(), n, (float)lr, (float)beta1, (float)beta2,    \
                     (float)eps, (float)wd, (float)clip, (float)grad_scale, inv_bc1,   \
                     inv_bc2)

  if (g.scalar_type() == at::kBFloat16) {
    if (emit) LAUNCH(uint16_t, true);
    else LAUNCH(uint16_t, false);
  } else if (g.scalar_type() == at::kFloat) {
    if (emit) LAUNCH(float, true);
    else LAUNCH(float, false);
  } else {
    TORCH_CHECK(false, "adamw: unsupported grad dtype");
  }
#undef LAUNCH
}
