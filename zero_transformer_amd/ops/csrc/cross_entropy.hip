// Fused gather-based cross entropy for gfx950: never materializes the
// (N, 50304) one-hot of the reference (GPT.py:105-111, losses.py:10-23).
// fp32 log-softmax statistics over bf16 logits.
//
// fwd: one block per row, single pass over the logits (online softmax:
//      running max + rescaled sum), s16x8-vectorized loads (G13) ->
//      per-row loss (lse - logit[target]) + saved lse.
// bwd: dlogits = (softmax - onehot) * dloss / N in one vectorized pass.

#include "common.h"

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <torch/extension.h>

namespace {

// bf16 fast path: V % 8 == 0. Online max/sum in a single read of the row.
__global__ __launch_bounds__(256) void ce_fwd_vec(
    const uint16_t* __restrict__ logits, const long* __restrict__ targets,
    float* __restrict__ loss, float* __restrict__ lse_out, long rows, int V) {
  __shared__ float scratch[16];
  const int t = threadIdx.x;
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const uint16_t* lr = logits + row * V;
    float m = -3.4e38f, s = 0.f;
    for (int c = t * 8; c < V; c += 256 * 8) {
      s16x8 x8 = *reinterpret_cast<const s16x8*>(&lr[c]);
      float xv[8], lm = -3.4e38f;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        xv[e] = bf16_to_f32((uint16_t)x8[e]);
        lm = fmaxf(lm, xv[e]);
      }
      if (lm > m) {
        s *= __expf(m - lm);
        m = lm;
      }
#pragma unroll
      for (int e = 0; e < 8; ++e) s += __expf(xv[e] - m);
    }
    // cross-lane/wave combine of (m, s)
    {
#pragma unroll
      for (int off = 32; off > 0; off >>= 1) {
        float mo = __shfl_xor(m, off, WAVE);
        float so = __shfl_xor(s, off, WAVE);
        float mn = fmaxf(m, mo);
        s = s * __expf(m - mn) + so * __expf(mo - mn);
        m = mn;
      }
      const int lane = t & (WAVE - 1), wid = t / WAVE;
      __shared__ float sm[4], ss[4];
      if (lane == 0) {
        sm[wid] = m;
        ss[wid] = s;
      }
      __syncthreads();
      float mn = fmaxf(fmaxf(sm[0], sm[1]), fmaxf(sm[2], sm[3]));
      s = ss[0] * __expf(sm[0] - mn) + ss[1] * __expf(sm[1] - mn) +
          ss[2] * __expf(sm[2] - mn) + ss[3] * __expf(sm[3] - mn);
      m = mn;
      __syncthreads();
    }
    const float lse = m + __logf(s);
    if (t == 0) {
      lse_out[row] = lse;
      const long tg = targets[row];
      loss[row] = tg >= 0 ? lse - bf16_to_f32(lr[tg]) : 0.f;
    }
  }
}

__global__ __launch_bounds__(256) void ce_bwd_vec(
    const uint16_t* __restrict__ logits, const long* __restrict__ targets,
    const float* __restrict__ lse, const float* __restrict__ dloss,
    uint16_t* __restrict__ dlogits, long rows, long divisor, int V) {
  const float scale = dloss[0] / divisor;
  const int t = threadIdx.x;
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const uint16_t* lr = logits + row * V;
    uint16_t* dr = dlogits + row * V;
    const float l = lse[row];
    const long tg = targets[row];
    if (tg < 0) {  // ignore_index row: zero gradient
      for (int c = t * 8; c < V; c += 256 * 8)
        *reinterpret_cast<s16x8*>(&dr[c]) = s16x8{};
      continue;
    }
    for (int c = t * 8; c < V; c += 256 * 8) {
      s16x8 x8 = *reinterpret_cast<const s16x8*>(&lr[c]);
      s16x8 d8;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float p = __expf(bf16_to_f32((uint16_t)x8[e]) - l);
        d8[e] = (short)f32_to_bf16(scale * (p - (c + e == tg ? 1.f : 0.f)));
      }
      *reinterpret_cast<s16x8*>(&dr[c]) = d8;
    }
  }
}

// Generic fallback (fp32 or odd V)
template <typename T>
__global__ void ce_fwd_kernel(const T* __restrict__ logits,
                              const long* __restrict__ targets,
                              float* __restrict__ loss, float* __restrict__ lse_out,
                              long rows, int V) {
  __shared__ float scratch[16];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* lr = logits + row * V;
    float m = -3.4e38f;
    for (int c = threadIdx.x; c < V; c += blockDim.x) m = fmaxf(m, to_f32(lr[c]));
    m = wave_reduce_max(m);
    {  // cross-wave max via scratch
      const int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
      if (lane == 0) scratch[wid] = m;
      __syncthreads();
      const int nw = blockDim.x / WAVE;
      float v = (threadIdx.x < nw) ? scratch[threadIdx.x] : -3.4e38f;
      if (wid == 0) v = wave_reduce_max(v);
      if (threadIdx.x == 0) scratch[0] = v;
      __syncthreads();
      m = scratch[0];
      __syncthreads();
    }
    float s = 0.f;
    for (int c = threadIdx.x; c < V; c += blockDim.x) s += expf(to_f32(lr[c]) - m);
    s = block_reduce_sum(s, scratch);
    const float lse = m + logf(s);
    if (threadIdx.x == 0) {
      lse_out[row] = lse;
      const long tg = targets[row];
      loss[row] = tg >= 0 ? lse - to_f32(lr[tg]) : 0.f;
    }
    __syncthreads();
  }
}

template <typename T>
__global__ void ce_bwd_kernel(const T* __restrict__ logits,
                              const long* __restrict__ targets,
                              const float* __restrict__ lse,
                              const float* __restrict__ dloss,  // scalar
                              T* __restrict__ dlogits, long rows, long divisor,
                              int V) {
  const float scale = dloss[0] / divisor;  // mean over counted rows
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* lr = logits + row * V;
    T* dr = dlogits + row * V;
    const float l = lse[row];
    const long t = targets[row];
    if (t < 0) {
      for (int c = threadIdx.x; c < V; c += blockDim.x) dr[c] = from_f32<T>(0.f);
      continue;
    }
    for (int c = threadIdx.x; c < V; c += blockDim.x) {
      float p = expf(to_f32(lr[c]) - l);
      dr[c] = from_f32<T>(scale * (p - (c == t ? 1.f : 0.f)));
    }
  }
}

}  // namespace

// `divisor`: number of rows the mean is taken over (rows with target < 0 are
// ignored and contribute 0; the caller knows the valid count analytically —
// the shifted-CE path passes B*(T-1)). divisor <= 0 means all rows count.
std::vector<at::Tensor> cross_entropy_fwd(at::Tensor logits, at::Tensor targets,
                                          int64_t divisor) {
  TORCH_CHECK(logits.is_cuda() && logits.is_contiguous() && logits.dim() == 2);
  TORCH_CHECK(targets.scalar_type() == at::kLong);
  const long rows = logits.size(0);
  const int V = logits.size(1);
  auto loss = at::empty({rows}, logits.options().dtype(at::kFloat));
  auto lse = at::empty({rows}, logits.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  const int block = 256;
  const int grid = int(std::min<long>(rows, 4096));
  if (logits.scalar_type() == at::kBFloat16 && V % 8 == 0) {
    hipLaunchKernelGGL(ce_fwd_vec, dim3(grid), dim3(block), 0, stream,
                       (const uint16_t*)logits.data_ptr(), targets.data_ptr<long>(),
                       loss.data_ptr<float>(), lse.data_ptr<float>(), rows, V);
  } else if (logits.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(ce_fwd_kernel<uint16_t>, dim3(grid), dim3(block), 0, stream,
                       (const uint16_t*)logits.data_ptr(), targets.data_ptr<long>(),
                       loss.data_ptr<float>(), lse.data_ptr<float>(), rows, V);
  } else if (logits.scalar_type() == at::kFloat) {
    hipLaunchKernelGGL(ce_fwd_kernel<float>, dim3(grid), dim3(block), 0, stream,
                       logits.data_ptr<float>(), targets.data_ptr<long>(),
                       loss.data_ptr<float>(), lse.data_ptr<float>(), rows, V);
  } else {
    TORCH_CHECK(false, "cross_entropy: unsupported dtype");
  }
  const double div = divisor > 0 ? (double)divisor : (double)rows;
  return {loss.sum() / div, lse};
}

at::Tensor cross_entropy_bwd(at::Tensor logits, at::Tensor targets, at::Tensor lse,
                             at::Tensor dloss, int64_t divisor) {
  const long rows = logits.size(0);
  const long div = divisor > 0 ? divisor : rows;
  const int V = logits.size(1);
  auto dlogits = at::empty_like(logits);
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  auto dl = dloss.to(at::kFloat).contiguous();
  const int block = 256;
  const int grid = int(std::min<long>(rows, 4096));
  if (logits.scalar_type() == at::kBFloat16 && V % 8 == 0) {
    hipLaunchKernelGGL(ce_bwd_vec, dim3(grid), dim3(block), 0, stream,
                       (const uint16_t*)logits.data_ptr(), targets.data_ptr<long>(),
                       lse.data_ptr<float>(), dl.data_ptr<float>(),
                       (uint16_t*)dlogits.data_ptr(), rows, div, V);
  } else if (logits.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(ce_bwd_kernel<uint16_t>, dim3(grid), dim3(block), 0, stream,
                       (const uint16_t*)logits.data_ptr(), targets.data_ptr<long>(),
                       lse.data_ptr<float>(), dl.data_ptr<float>(),
                       (uint16_t*)dlogits.data_ptr(), rows, div, V);
  } else {
    hipLaunchKernelGGL(ce_bwd_kernel<float>, dim3(grid), dim3(block), 0, stream,
                       logits.data_ptr<float>(), targets.data_ptr<long>(),
                       lse.data_ptr<float>(), dl.data_ptr<float>(),
                       dlogits.data_ptr<float>(), rows, div, V);
  }
  return dlogits;
}
