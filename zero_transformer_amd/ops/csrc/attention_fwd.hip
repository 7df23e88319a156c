// Fused causal ALiBi flash attention forward for gfx950 (CDNA4 MFMA).
//
// Replaces the reference's dense O(T^2) materialized attention
// (layers.py:159-178: q@k^T, +alibi, tril mask, fp32 softmax, dropout, @v)
// with a flash-style online-softmax kernel; softmax statistics stay fp32 end
// to end (the bf16-softmax failure of logs/580.md:94-98 cannot happen by
// construction). ALiBi is the true bias slope*(j-i) (shift-equivalent to the
// reference's single-row mask, layers.py:33-44). Dropout uses a counter RNG
// so backward regenerates the identical mask without storing it.
//
// Structure (the CDNA4 guide's swapped-QK^T recipe, §B attention):
//   * 8 waves/block, each owning 32 q rows (QB = 256); K/V staged once per
//     64-key tile, two 32-key MFMA subtiles per staging.
//   * QK^T is computed SWAPPED — mfma(A=K, B=Q^T) — so the C column index
//     is the q row: each lane holds a full P-row segment in registers and
//     the softmax row-reduce is 15 in-lane ops + one shfl_xor(32), with
//     scalar (not per-reg) running m/l. No P round-trip through LDS:
//     P -> bf16 A-fragments via v_cvt_pk_bf16_f32 + v_permlane32_swap (T12).
//   * K staged row-major in XOR-swizzled LDS (T2, conflict-free
//     ds_read_b128); V transposed ([d][key]) so PV B-reads are k-contiguous.
//   * defer-max: the O rescale is skipped when no lane's running max grew
//     (exact: threshold 0), __expf (v_exp_f32) throughout.
//
// Fragment maps (gfx950, §3): A: lane l holds A[i=l%32][k=8*(l/32)+e];
// B: B[k=8*(l/32)+e][j=l%32]; C/D: col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5).

#include "common.h"

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <torch/extension.h>

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

constexpr int NW = 8;        // waves per block
constexpr int QB = NW * 32;  // q rows per block
constexpr int KB = 64;       // staged kv tile (2 x 32-key subtiles)
constexpr float NEG_INF = -3.0e38f;
constexpr int VT_STRIDE = KB + 8;  // transposed-V row stride (elements)

ZTA_DEV int swz(int row, int byte_off) { return byte_off ^ ((row & 7) << 4); }

template <int D>
__global__ __launch_bounds__(512) void flash_fwd_kernel(
    const uint16_t* __restrict__ q, const uint16_t* __restrict__ k,
    const uint16_t* __restrict__ v, const float* __restrict__ slopes,
    uint16_t* __restrict__ o, float* __restrict__ lse, int H, int T,
    float scale, float p_drop, uint64_t seed) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  uint16_t* k_lds = (uint16_t*)smem;         // KB*128 (row stride 256 B, swz)
  uint16_t* vt_lds = k_lds + KB * 128;       // D*VT_STRIDE

  const int bh = blockIdx.y;
  const int h = bh % H;
  const long base = (long)bh * T * D;
  const int q0 = blockIdx.x * QB;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int li = lane & 31;
  const int hi = lane >> 5;
  const int qw = q0 + wave * 32;
  const int qi = qw + li;  // THIS lane's q row (swapped layout)
  const float slope = slopes[h];
  const uint32_t drop_thr = (uint32_t)(p_drop * 256.0f + 0.5f);
  const float inv_keep = drop_thr ? 256.0f / (256.0f - (float)drop_thr) : 1.0f;

  constexpr int KS = D / 16;
  constexpr int DB = D / 32;

  // Q fragments (B operand of the swapped QK^T): lane l holds
  // Q[qw + l%32][8*(l/32) + e] per k-step — a 16 B row-major load.
  bf16x8 q_frag[KS];
  {
    const bool ok = qi < T;
#pragma unroll
    for (int s = 0; s < KS; ++s)
      q_frag[s] = ok ? *reinterpret_cast<const bf16x8*>(
                           &q[base + (long)qi * D + s * 16 + 8 * hi])
                     : bf16x8{};
  }

  float m_run = NEG_INF, l_run = 0.f;
  f32x16 o_acc[DB];
#pragma unroll
  for (int d = 0; d < DB; ++d) o_acc[d] = f32x16{};

  const int kv_end = min(T, q0 + QB);
  for (int kt = 0; kt < kv_end; kt += KB) {
    // ---- stage K row-major (swizzled) + V transposed, all 8 waves ----
    for (int idx = threadIdx.x * 8; idx < KB * D; idx += 512 * 8) {
      const int key = idx / D, d = idx % D;
      const int kg = kt + key;
      s16x8 kv8{}, vv8{};
      if (kg < T) {
        kv8 = *reinterpret_cast<const s16x8*>(&k[base + (long)kg * D + d]);
        vv8 = *reinterpret_cast<const s16x8*>(&v[base + (long)kg * D + d]);
      }
      *reinterpret_cast<s16x8*>((char*)k_lds + swz(key, key * 256 + d * 2)) = kv8;
#pragma unroll
      for (int e = 0; e < 8; ++e) vt_lds[(d + e) * VT_STRIDE + key] = (uint16_t)vv8[e];
    }
    __syncthreads();

#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      const int kt32 = kt + sub * 32;
      if (kt32 > qw + 31 || kt32 >= T) continue;  // fully masked for this wave

      // ---- S^T tile: C[i=key][j=q] = mfma(A=K, B=Q^T) ----
      f32x16 s_acc{};
#pragma unroll
      for (int s = 0; s < KS; ++s) {
        const int kk = s * 16 + 8 * hi;
        const int krow = sub * 32 + li;
        bf16x8 a_frag = *reinterpret_cast<const bf16x8*>(
            (char*)k_lds + swz(krow, krow * 256 + kk * 2));
        s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_frag, q_frag[s], s_acc, 0, 0, 0);
      }

      // ---- in-lane softmax for q row `qi`; reg r holds key kt32+crow(r,hi) ----
      float sv[16];
      float tile_max = NEG_INF;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kj = kt32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        float x = s_acc[r] * scale + slope * (float)(kj - qi);
        if (kj > qi || kj >= T || qi >= T) x = NEG_INF;
        sv[r] = x;
        tile_max = fmaxf(tile_max, x);
      }
      tile_max = fmaxf(tile_max, __shfl_xor(tile_max, 32, 64));

      float alpha = 1.f;
      const bool valid = tile_max > 0.5f * NEG_INF;
      if (!__all(tile_max <= m_run)) {  // defer-max, exact threshold 0
        const float mn = valid ? fmaxf(m_run, tile_max) : m_run;
        // alpha = exp(old_m - new_m); 0 when old_m was -inf (O, l still zero)
        alpha = (m_run > 0.5f * NEG_INF) ? __expf(m_run - mn) : (valid ? 0.f : 1.f);
        m_run = mn;
      }
      float p[16];
      float row_sum = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        p[r] = (valid && m_run > 0.5f * NEG_INF) ? __expf(sv[r] - m_run) : 0.f;
        row_sum += p[r];
      }
      row_sum += __shfl_xor(row_sum, 32, 64);
      l_run = l_run * alpha + row_sum;

      // ---- dropout on the PV path (denominator keeps the full softmax) ----
      if (drop_thr) {
        const int bhT_qi = bh * T + qi;
#pragma unroll
        for (int g = 0; g < 4; ++g) {
          const int kbase = kt32 + 8 * g + 4 * hi;  // keys kbase..kbase+3 = regs 4g..4g+3
          const uint32_t bits = drop_bits32((uint32_t)seed, bhT_qi, kbase >> 2);
#pragma unroll
          for (int e = 0; e < 4; ++e) {
            const bool keep = ((bits >> (8 * e)) & 0xffu) >= drop_thr;
            p[4 * g + e] = keep ? p[4 * g + e] * inv_keep : 0.f;
          }
        }
      }

      // ---- rescale O by alpha of each reg's q row (broadcast via shfl) ----
      if (!__all(alpha == 1.f)) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const float ar = __shfl(alpha, (r & 3) + 8 * (r >> 2) + 4 * hi, 64);
#pragma unroll
          for (int d = 0; d < DB; ++d) o_acc[d][r] *= ar;
        }
      }

      // ---- P -> bf16 A-fragments in-register (cvt_pk + permlane32_swap) ----
      // words w[j]: cvt_pk(p[2j], p[2j+1]); swap pairs (w0,w2),(w1,w3) and
      // (w4,w6),(w5,w7) -> lane-correct A[i=q][k=key] fragments.
      unsigned w[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(w[j]) : "v"(p[2 * j]), "v"(p[2 * j + 1]));
      }
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        auto r0 = __builtin_amdgcn_permlane32_swap(w[4 * j + 0], w[4 * j + 2], false, false);
        w[4 * j + 0] = r0[0];
        w[4 * j + 2] = r0[1];
        auto r1 = __builtin_amdgcn_permlane32_swap(w[4 * j + 1], w[4 * j + 3], false, false);
        w[4 * j + 1] = r1[0];
        w[4 * j + 3] = r1[1];
      }
      bf16x8 pa[2];
      {
        union { unsigned u[4]; bf16x8 v8; } cvt;
        cvt.u[0] = w[0]; cvt.u[1] = w[1]; cvt.u[2] = w[2]; cvt.u[3] = w[3];
        pa[0] = cvt.v8;
        cvt.u[0] = w[4]; cvt.u[1] = w[5]; cvt.u[2] = w[6]; cvt.u[3] = w[7];
        pa[1] = cvt.v8;
      }

      // ---- O += P @ V ----
#pragma unroll
      for (int d = 0; d < DB; ++d) {
#pragma unroll
        for (int s2 = 0; s2 < 2; ++s2) {
          bf16x8 b_frag = *reinterpret_cast<const bf16x8*>(
              &vt_lds[(d * 32 + li) * VT_STRIDE + sub * 32 + s2 * 16 + 8 * hi]);
          o_acc[d] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa[s2], b_frag, o_acc[d], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // ---- epilogue: O rows are crow(r,hi); l/m live in the row's lane ----
  const float inv_l = l_run > 0.f ? 1.f / l_run : 0.f;
  if (hi == 0 && qi < T)
    lse[(long)bh * T + qi] = l_run > 0.f ? m_run + __logf(l_run) : NEG_INF;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
    const int qr = qw + row;
    if (qr >= T) continue;
    const float il = __shfl(inv_l, row, 64);
#pragma unroll
    for (int d = 0; d < DB; ++d)
      o[base + (long)qr * D + d * 32 + li] = f32_to_bf16(o_acc[d][r] * il);
  }
}

template <int D>
void launch_fwd(const at::Tensor& q, const at::Tensor& k, const at::Tensor& v,
                const at::Tensor& slopes, at::Tensor& o, at::Tensor& lse, int B,
                int H, int T, float scale, float p_drop, uint64_t seed,
                hipStream_t stream) {
  dim3 grid((T + QB - 1) / QB, B * H);
  const size_t smem = (KB * 128 + D * VT_STRIDE) * sizeof(uint16_t);
  hipLaunchKernelGGL(flash_fwd_kernel<D>, grid, dim3(512), smem, stream,
                     (const uint16_t*)q.data_ptr(), (const uint16_t*)k.data_ptr(),
                     (const uint16_t*)v.data_ptr(), slopes.data_ptr<float>(),
                     (uint16_t*)o.data_ptr(), lse.data_ptr<float>(), H, T, scale,
                     p_drop, seed);
}

}  // namespace

std::vector<at::Tensor> attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                 at::Tensor slopes, double p_drop, int64_t seed) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && q.dim() == 4, "q must be (B,H,T,D) contiguous");
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "attn_fwd: bf16 only");
  const int B = q.size(0), H = q.size(1), T = q.size(2), D = q.size(3);
  auto o = at::empty_like(q);
  auto lse = at::empty({B, H, T}, q.options().dtype(at::kFloat));
  auto sl = slopes.to(at::kFloat).contiguous();
  const float scale = 1.0f / sqrtf((float)D);
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  switch (D) {
    case 32: launch_fwd<32>(q, k, v, sl, o, lse, B, H, T, scale, (float)p_drop, (uint64_t)seed, stream); break;
    case 64: launch_fwd<64>(q, k, v, sl, o, lse, B, H, T, scale, (float)p_drop, (uint64_t)seed, stream); break;
    case 96: launch_fwd<96>(q, k, v, sl, o, lse, B, H, T, scale, (float)p_drop, (uint64_t)seed, stream); break;
    case 128: launch_fwd<128>(q, k, v, sl, o, lse, B, H, T, scale, (float)p_drop, (uint64_t)seed, stream); break;
    default: TORCH_CHECK(false, "attn_fwd: head_dim must be one of 32/64/96/128, got ", D);
  }
  return {o, lse};
}
