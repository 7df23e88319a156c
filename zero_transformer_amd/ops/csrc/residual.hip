// Fused residual-add + dropout for gfx950:  y = x + dropout(h, p).
//
// Replaces the reference's dropout-then-residual-add pairs
// (src/models/layers.py:76,190 + GPT.py:43-49) — eager PyTorch runs these
// as a dropout kernel (which also materializes a mask tensor), a scale
// kernel in backward, and an add kernel. Here: ONE memory pass in forward
// (read x, h; write y) and one in backward (dh = dy * mask * inv_keep;
// dx aliases dy), with the mask regenerated from the counter RNG
// (common.h drop_bits32) instead of stored.
//
// Mask geometry: element i keeps iff byte (i & 3) of
// drop_bits32(seed, i >> 18, (i >> 2) & 0xffff) >= thr = round(p * 256).

#include "common.h"

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <torch/extension.h>

namespace {

ZTA_DEV uint32_t rd_bits(uint32_t seed, long i4) {
  // i4 = element_index / 4; split so the 32-bit hash inputs stay distinct
  return drop_bits32(seed, (int)(i4 >> 16), (int)(i4 & 0xffff));
}

__global__ __launch_bounds__(256) void res_drop_fwd(
    const uint16_t* __restrict__ x, const uint16_t* __restrict__ h,
    uint16_t* __restrict__ y, long n, uint32_t thr, float inv_keep,
    uint32_t seed) {
  const long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  const long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i = i0; i < n; i += stride) {
    s16x8 x8 = *reinterpret_cast<const s16x8*>(&x[i]);
    s16x8 h8 = *reinterpret_cast<const s16x8*>(&h[i]);
    const uint32_t b0 = rd_bits(seed, i >> 2);
    const uint32_t b1 = rd_bits(seed, (i >> 2) + 1);
    s16x8 y8;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const uint32_t bits = e < 4 ? b0 : b1;
      const bool keep = ((bits >> (8 * (e & 3))) & 0xffu) >= thr;
      const float hv = keep ? bf16_to_f32((uint16_t)h8[e]) * inv_keep : 0.f;
      y8[e] = (short)f32_to_bf16(bf16_to_f32((uint16_t)x8[e]) + hv);
    }
    *reinterpret_cast<s16x8*>(&y[i]) = y8;
  }
}

__global__ __launch_bounds__(256) void res_drop_bwd(
    const uint16_t* __restrict__ dy, uint16_t* __restrict__ dh, long n,
    uint32_t thr, float inv_keep, uint32_t seed) {
  const long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  const long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i = i0; i < n; i += stride) {
    s16x8 d8 = *reinterpret_cast<const s16x8*>(&dy[i]);
    const uint32_t b0 = rd_bits(seed, i >> 2);
    const uint32_t b1 = rd_bits(seed, (i >> 2) + 1);
    s16x8 o8;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const uint32_t bits = e < 4 ? b0 : b1;
      const bool keep = ((bits >> (8 * (e & 3))) & 0xffu) >= thr;
      o8[e] = keep ? (short)f32_to_bf16(bf16_to_f32((uint16_t)d8[e]) * inv_keep)
                   : (short)0;
    }
    *reinterpret_cast<s16x8*>(&dh[i]) = o8;
  }
}

}  // namespace

at::Tensor residual_dropout_fwd(at::Tensor x, at::Tensor h, double p, int64_t seed) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && h.is_contiguous());
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && h.scalar_type() == at::kBFloat16);
  TORCH_CHECK(x.numel() == h.numel() && x.numel() % 8 == 0);
  const long n = x.numel();
  auto y = at::empty_like(x);
  const uint32_t thr = (uint32_t)(p * 256.0 + 0.5);
  const float inv_keep = thr ? 256.0f / (256.0f - (float)thr) : 1.0f;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const int block = 256;
  const int grid = capped_grid(n / 8, block, 4096);
  hipLaunchKernelGGL(res_drop_fwd, dim3(grid), dim3(block), 0, stream,
                     (const uint16_t*)x.data_ptr(), (const uint16_t*)h.data_ptr(),
                     (uint16_t*)y.data_ptr(), n, thr, inv_keep, (uint32_t)seed);
  return y;
}

at::Tensor residual_dropout_bwd(at::Tensor dy, double p, int64_t seed) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous());
  const long n = dy.numel();
  auto dh = at::empty_like(dy);
  const uint32_t thr = (uint32_t)(p * 256.0 + 0.5);
  const float inv_keep = thr ? 256.0f / (256.0f - (float)thr) : 1.0f;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const int block = 256;
  const int grid = capped_grid(n / 8, block, 4096);
  hipLaunchKernelGGL(res_drop_bwd, dim3(grid), dim3(block), 0, stream,
                     (const uint16_t*)dy.data_ptr(), (uint16_t*)dh.data_ptr(), n, thr,
                     inv_keep, (uint32_t)seed);
  return dh;
}
