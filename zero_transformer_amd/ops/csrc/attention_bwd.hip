// Flash attention backward for gfx950 (CDNA4 MFMA): FA2-style two-kernel
// scheme with P recomputed from the saved per-row LSE — no O(T^2) tensor is
// ever materialized (the reference's autograd attention backward
// re-materializes the dense probs; SURVEY.md §2.3 "Backward of all above").
//
//   delta[b,h,i] = sum_d dO * O                  (delta kernel)
//   dQ kernel : per 128-row Q block, loop KV tiles:
//       S = Q K^T, P = exp(S*sc + bias - lse), dP = dO V^T,
//       dS = sc * P (M∘dP/keep - delta), dQ += dS K
//   dKdV kernel: per 128-key KV block, loop Q tiles:
//       S^T = K Q^T, P^T, dP^T = V dO^T,
//       dV += (M∘P^T/keep) dO, dK += sc * (P^T(M∘dP^T/keep - delta)) Q
//
// The dropout mask M is regenerated from the counter RNG with the forward's
// seed. All softmax math in fp32; MFMA operand staging mirrors the forward
// (row-major XOR-swizzled tiles for k-contiguous B reads, explicitly
// transposed tiles where the reduction runs over rows).

#include "common.h"

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <torch/extension.h>

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

constexpr int QB = 128;
constexpr int KB = 32;
constexpr float NEG_INF = -3.0e38f;
constexpr int TSTRIDE = KB + 8;  // transposed-tile row stride (elements)

ZTA_DEV int swz(int row, int byte_off) { return byte_off ^ ((row & 7) << 4); }

// ---------------------------------------------------------------------------
__global__ void delta_kernel(const uint16_t* __restrict__ dout,
                             const uint16_t* __restrict__ o,
                             float* __restrict__ delta, long rows, int D) {
  const int wid = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int nw = gridDim.x * blockDim.x / WAVE;
  for (long row = wid; row < rows; row += nw) {
    float s = 0.f;
    for (int d = lane; d < D; d += WAVE)
      s += bf16_to_f32(dout[row * D + d]) * bf16_to_f32(o[row * D + d]);
    s = wave_reduce_sum(s);
    if (lane == 0) delta[row] = s;
  }
}

// ---------------------------------------------------------------------------
// dQ kernel: 4 waves x 32 q rows.
// LDS: k_lds [KB][128] swz | kt_lds [D][TSTRIDE] | v_lds [KB][128] swz |
//      ds_lds 4x[32][TSTRIDE]
template <int D>
__global__ __launch_bounds__(256) void flash_dq_kernel(
    const uint16_t* __restrict__ q, const uint16_t* __restrict__ k,
    const uint16_t* __restrict__ v, const uint16_t* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    const float* __restrict__ slopes, uint16_t* __restrict__ dq, int H, int T,
    float scale, float p_drop, uint64_t seed) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  uint16_t* k_lds = (uint16_t*)smem;
  uint16_t* kt_lds = k_lds + KB * 128;
  uint16_t* v_lds = kt_lds + D * TSTRIDE;
  uint16_t* ds_lds = v_lds + KB * 128;

  const int bh = blockIdx.y;
  const int h = bh % H;
  const long base = (long)bh * T * D;
  const int q0 = blockIdx.x * QB;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int li = lane & 31, hi = lane >> 5;
  const int qw = q0 + wave * 32;
  const float slope = slopes[h];
  const uint32_t drop_thr = (uint32_t)(p_drop * 65536.0f + 0.5f);
  const float inv_keep = drop_thr ? 65536.0f / (65536.0f - (float)drop_thr) : 1.0f;

  constexpr int KS = D / 16;
  constexpr int DB = D / 32;
  bf16x8 q_frag[KS], do_frag[KS];
  {
    const int qi = qw + li;
    const bool ok = qi < T;
#pragma unroll
    for (int s = 0; s < KS; ++s) {
      if (ok) {
        q_frag[s] = *reinterpret_cast<const bf16x8*>(&q[base + (long)qi * D + s * 16 + 8 * hi]);
        do_frag[s] = *reinterpret_cast<const bf16x8*>(&dout[base + (long)qi * D + s * 16 + 8 * hi]);
      } else {
        q_frag[s] = bf16x8{};
        do_frag[s] = bf16x8{};
      }
    }
  }
  // per-register row constants
  float lse_r[16], delta_r[16];
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int qi = qw + (r & 3) + 8 * (r >> 2) + 4 * hi;
    lse_r[r] = qi < T ? lse[(long)bh * T + qi] : 0.f;
    delta_r[r] = qi < T ? delta[(long)bh * T + qi] : 0.f;
  }

  f32x16 dq_acc[DB];
#pragma unroll
  for (int d = 0; d < DB; ++d) dq_acc[d] = f32x16{};

  const int kv_end = min(T, q0 + QB);
  for (int kt = 0; kt < kv_end; kt += KB) {
    for (int idx = threadIdx.x * 8; idx < KB * D; idx += 256 * 8) {
      const int key = idx / D, d = idx % D;
      const int kg = kt + key;
      s16x8 kv8{}, vv8{};
      if (kg < T) {
        kv8 = *reinterpret_cast<const s16x8*>(&k[base + (long)kg * D + d]);
        vv8 = *reinterpret_cast<const s16x8*>(&v[base + (long)kg * D + d]);
      }
      *reinterpret_cast<s16x8*>((char*)k_lds + swz(key, key * 256 + d * 2)) = kv8;
      *reinterpret_cast<s16x8*>((char*)v_lds + swz(key, key * 256 + d * 2)) = vv8;
#pragma unroll
      for (int e = 0; e < 8; ++e) kt_lds[(d + e) * TSTRIDE + key] = (uint16_t)kv8[e];
    }
    __syncthreads();

    f32x16 s_acc{}, dp_acc{};
#pragma unroll
    for (int s = 0; s < KS; ++s) {
      const int kk = s * 16 + 8 * hi;
      bf16x8 k_frag = *reinterpret_cast<const bf16x8*>((char*)k_lds + swz(li, li * 256 + kk * 2));
      bf16x8 v_frag = *reinterpret_cast<const bf16x8*>((char*)v_lds + swz(li, li * 256 + kk * 2));
      s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(q_frag[s], k_frag, s_acc, 0, 0, 0);
      dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(do_frag[s], v_frag, dp_acc, 0, 0, 0);
    }

    const int kj = kt + li;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
      const int qi = qw + row;
      float p = 0.f;
      if (!(kj > qi || kj >= T || qi >= T))
        p = __expf(s_acc[r] * scale + slope * (float)(kj - qi) - lse_r[r]);
      float dp = dp_acc[r];
      if (drop_thr) {
        const uint64_t bits = drop_bits(seed, bh * T + qi, kj >> 2);
        const bool keep = (uint16_t)(bits >> (16 * (kj & 3))) >= drop_thr;
        dp = keep ? dp * inv_keep : 0.f;
      }
      const float ds = scale * p * (dp - delta_r[r]);
      ds_lds[(wave * 32 + row) * TSTRIDE + li] = f32_to_bf16(ds);
    }

#pragma unroll
    for (int d = 0; d < DB; ++d) {
#pragma unroll
      for (int s2 = 0; s2 < KB / 16; ++s2) {
        const int kk = s2 * 16 + 8 * hi;
        bf16x8 a_frag = *reinterpret_cast<const bf16x8*>(&ds_lds[(wave * 32 + li) * TSTRIDE + kk]);
        bf16x8 b_frag = *reinterpret_cast<const bf16x8*>(&kt_lds[(d * 32 + li) * TSTRIDE + kk]);
        dq_acc[d] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_frag, b_frag, dq_acc[d], 0, 0, 0);
      }
    }
    __syncthreads();
  }

#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
    const int qi = qw + row;
    if (qi >= T) continue;
#pragma unroll
    for (int d = 0; d < DB; ++d)
      dq[base + (long)qi * D + d * 32 + li] = f32_to_bf16(dq_acc[d][r]);
  }
}

// ---------------------------------------------------------------------------
// dKdV kernel: 4 waves x 32 keys = 128-key block, loop q tiles of 32.
// LDS: q_lds [32][128] swz | qt_lds [D][TSTRIDE] | do_lds [32][128] swz |
//      dot_lds [D][TSTRIDE] | pT_lds 4x[32][TSTRIDE] | dsT_lds 4x[32][TSTRIDE]
template <int D>
__global__ __launch_bounds__(256) void flash_dkdv_kernel(
    const uint16_t* __restrict__ q, const uint16_t* __restrict__ k,
    const uint16_t* __restrict__ v, const uint16_t* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    const float* __restrict__ slopes, uint16_t* __restrict__ dk,
    uint16_t* __restrict__ dv, int H, int T, float scale, float p_drop,
    uint64_t seed) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  uint16_t* q_lds = (uint16_t*)smem;
  uint16_t* qt_lds = q_lds + 32 * 128;
  uint16_t* do_lds = qt_lds + D * TSTRIDE;
  uint16_t* dot_lds = do_lds + 32 * 128;
  uint16_t* pT_lds = dot_lds + D * TSTRIDE;
  uint16_t* dsT_lds = pT_lds + 4 * 32 * TSTRIDE;

  const int bh = blockIdx.y;
  const int h = bh % H;
  const long base = (long)bh * T * D;
  const int k0 = blockIdx.x * QB;  // 128 keys per block
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int li = lane & 31, hi = lane >> 5;
  const int kw = k0 + wave * 32;  // this wave's first key
  const float slope = slopes[h];
  const uint32_t drop_thr = (uint32_t)(p_drop * 65536.0f + 0.5f);
  const float inv_keep = drop_thr ? 65536.0f / (65536.0f - (float)drop_thr) : 1.0f;

  constexpr int KS = D / 16;
  constexpr int DB = D / 32;
  bf16x8 k_frag[KS], v_frag[KS];
  {
    const int kg = kw + li;
    const bool ok = kg < T;
#pragma unroll
    for (int s = 0; s < KS; ++s) {
      if (ok) {
        k_frag[s] = *reinterpret_cast<const bf16x8*>(&k[base + (long)kg * D + s * 16 + 8 * hi]);
        v_frag[s] = *reinterpret_cast<const bf16x8*>(&v[base + (long)kg * D + s * 16 + 8 * hi]);
      } else {
        k_frag[s] = bf16x8{};
        v_frag[s] = bf16x8{};
      }
    }
  }

  f32x16 dk_acc[DB], dv_acc[DB];
#pragma unroll
  for (int d = 0; d < DB; ++d) {
    dk_acc[d] = f32x16{};
    dv_acc[d] = f32x16{};
  }

  for (int qt = k0; qt < T; qt += 32) {
    // stage Q/dO tiles (row-major swizzled + transposed)
    for (int idx = threadIdx.x * 8; idx < 32 * D; idx += 256 * 8) {
      const int qr = idx / D, d = idx % D;
      const int qg = qt + qr;
      s16x8 qv8{}, dov8{};
      if (qg < T) {
        qv8 = *reinterpret_cast<const s16x8*>(&q[base + (long)qg * D + d]);
        dov8 = *reinterpret_cast<const s16x8*>(&dout[base + (long)qg * D + d]);
      }
      *reinterpret_cast<s16x8*>((char*)q_lds + swz(qr, qr * 256 + d * 2)) = qv8;
      *reinterpret_cast<s16x8*>((char*)do_lds + swz(qr, qr * 256 + d * 2)) = dov8;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        qt_lds[(d + e) * TSTRIDE + qr] = (uint16_t)qv8[e];
        dot_lds[(d + e) * TSTRIDE + qr] = (uint16_t)dov8[e];
      }
    }
    __syncthreads();

    // S^T = K Q^T ; dP^T = V dO^T
    f32x16 st_acc{}, dpt_acc{};
#pragma unroll
    for (int s = 0; s < KS; ++s) {
      const int kk = s * 16 + 8 * hi;
      bf16x8 qb = *reinterpret_cast<const bf16x8*>((char*)q_lds + swz(li, li * 256 + kk * 2));
      bf16x8 dob = *reinterpret_cast<const bf16x8*>((char*)do_lds + swz(li, li * 256 + kk * 2));
      st_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(k_frag[s], qb, st_acc, 0, 0, 0);
      dpt_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(v_frag[s], dob, dpt_acc, 0, 0, 0);
    }

    const int qi = qt + li;  // this lane's q column
    const float lse_q = qi < T ? lse[(long)bh * T + qi] : 0.f;
    const float delta_q = qi < T ? delta[(long)bh * T + qi] : 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
      const int kj = kw + row;
      float p = 0.f;
      if (!(kj > qi || kj >= T || qi >= T))
        p = __expf(st_acc[r] * scale + slope * (float)(kj - qi) - lse_q);
      float dp = dpt_acc[r];
      float p_pv = p;
      if (drop_thr) {
        const uint64_t bits = drop_bits(seed, bh * T + qi, kj >> 2);
        const bool keep = (uint16_t)(bits >> (16 * (kj & 3))) >= drop_thr;
        dp = keep ? dp * inv_keep : 0.f;
        p_pv = keep ? p * inv_keep : 0.f;
      }
      const float ds = scale * p * (dp - delta_q);
      dsT_lds[(wave * 32 + row) * TSTRIDE + li] = f32_to_bf16(ds);
      pT_lds[(wave * 32 + row) * TSTRIDE + li] = f32_to_bf16(p_pv);
    }

    // dK += dS^T Q ; dV += P^T dO   (reduction over the 32 q columns)
#pragma unroll
    for (int d = 0; d < DB; ++d) {
#pragma unroll
      for (int s2 = 0; s2 < 32 / 16; ++s2) {
        const int kk = s2 * 16 + 8 * hi;
        bf16x8 dsa = *reinterpret_cast<const bf16x8*>(&dsT_lds[(wave * 32 + li) * TSTRIDE + kk]);
        bf16x8 pa = *reinterpret_cast<const bf16x8*>(&pT_lds[(wave * 32 + li) * TSTRIDE + kk]);
        bf16x8 qtb = *reinterpret_cast<const bf16x8*>(&qt_lds[(d * 32 + li) * TSTRIDE + kk]);
        bf16x8 dotb = *reinterpret_cast<const bf16x8*>(&dot_lds[(d * 32 + li) * TSTRIDE + kk]);
        dk_acc[d] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dsa, qtb, dk_acc[d], 0, 0, 0);
        dv_acc[d] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa, dotb, dv_acc[d], 0, 0, 0);
      }
    }
    __syncthreads();
  }

#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
    const int kj = kw + row;
    if (kj >= T) continue;
#pragma unroll
    for (int d = 0; d < DB; ++d) {
      dk[base + (long)kj * D + d * 32 + li] = f32_to_bf16(dk_acc[d][r]);
      dv[base + (long)kj * D + d * 32 + li] = f32_to_bf16(dv_acc[d][r]);
    }
  }
}

template <int D>
void launch_bwd(const at::Tensor& q, const at::Tensor& k, const at::Tensor& v,
                const at::Tensor& dout, const at::Tensor& lse, const at::Tensor& delta,
                const at::Tensor& slopes, at::Tensor& dq, at::Tensor& dk,
                at::Tensor& dv, int B, int H, int T, float scale, float p_drop,
                uint64_t seed, hipStream_t stream) {
  dim3 grid((T + QB - 1) / QB, B * H);
  const size_t smem_dq =
      (KB * 128 + D * TSTRIDE + KB * 128 + 4 * 32 * TSTRIDE) * sizeof(uint16_t);
  hipLaunchKernelGGL(flash_dq_kernel<D>, grid, dim3(256), smem_dq, stream,
                     (const uint16_t*)q.data_ptr(), (const uint16_t*)k.data_ptr(),
                     (const uint16_t*)v.data_ptr(), (const uint16_t*)dout.data_ptr(),
                     lse.data_ptr<float>(), delta.data_ptr<float>(),
                     slopes.data_ptr<float>(), (uint16_t*)dq.data_ptr(), H, T, scale,
                     p_drop, seed);
  const size_t smem_kv =
      (2 * 32 * 128 + 2 * D * TSTRIDE + 8 * 32 * TSTRIDE) * sizeof(uint16_t);
  hipLaunchKernelGGL(flash_dkdv_kernel<D>, grid, dim3(256), smem_kv, stream,
                     (const uint16_t*)q.data_ptr(), (const uint16_t*)k.data_ptr(),
                     (const uint16_t*)v.data_ptr(), (const uint16_t*)dout.data_ptr(),
                     lse.data_ptr<float>(), delta.data_ptr<float>(),
                     slopes.data_ptr<float>(), (uint16_t*)dk.data_ptr(),
                     (uint16_t*)dv.data_ptr(), H, T, scale, p_drop, seed);
}

}  // namespace

std::vector<at::Tensor> attn_bwd(at::Tensor dout, at::Tensor q, at::Tensor k,
                                 at::Tensor v, at::Tensor slopes, at::Tensor o,
                                 at::Tensor lse, double p_drop, int64_t seed) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && q.dim() == 4);
  const int B = q.size(0), H = q.size(1), T = q.size(2), D = q.size(3);
  auto dq = at::empty_like(q);
  auto dk = at::empty_like(k);
  auto dv = at::empty_like(v);
  auto delta = at::empty({B, H, T}, q.options().dtype(at::kFloat));
  auto sl = slopes.to(at::kFloat).contiguous();
  const float scale = 1.0f / sqrtf((float)D);
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  {
    const long rows = (long)B * H * T;
    const int block = 256;
    hipLaunchKernelGGL(delta_kernel, dim3(capped_grid(rows * WAVE, block)), dim3(block),
                       0, stream, (const uint16_t*)dout.data_ptr(),
                       (const uint16_t*)o.data_ptr(), delta.data_ptr<float>(), rows, D);
  }
  switch (D) {
    case 32: launch_bwd<32>(q, k, v, dout, lse, delta, sl, dq, dk, dv, B, H, T, scale, (float)p_drop, (uint64_t)seed, stream); break;
    case 64: launch_bwd<64>(q, k, v, dout, lse, delta, sl, dq, dk, dv, B, H, T, scale, (float)p_drop, (uint64_t)seed, stream); break;
    case 96: launch_bwd<96>(q, k, v, dout, lse, delta, sl, dq, dk, dv, B, H, T, scale, (float)p_drop, (uint64_t)seed, stream); break;
    case 128: launch_bwd<128>(q, k, v, dout, lse, delta, sl, dq, dk, dv, B, H, T, scale, (float)p_drop, (uint64_t)seed, stream); break;
    default: TORCH_CHECK(false, "attn_bwd: head_dim must be 32/64/96/128, got ", D);
  }
  return {dq, dk, dv};
}
