// Shared helpers for zero_transformer_amd CDNA4 (gfx950) kernels.
// Native HIP only — no CUDA compatibility paths.
#pragma once

#include <hip/hip_runtime.h>
#include <stdint.h>

#define ZTA_DEV __device__ __forceinline__

// CDNA wavefront is 64 lanes (not 32).
constexpr int WAVE = 64;

// ---------------------------------------------------------------------------
// bf16 <-> f32 (bf16 carried as uint16 bits; RNE conversion)
// ---------------------------------------------------------------------------
ZTA_DEV float bf16_to_f32(uint16_t h) {
  union {
    uint32_t u;
    float f;
  } c;
  c.u = uint32_t(h) << 16;
  return c.f;
}

ZTA_DEV uint16_t f32_to_bf16(float f) {
  union {
    float f;
    uint32_t u;
  } c;
  c.f = f;
  // round-to-nearest-even; NaN-safe
  uint32_t u = c.u;
  if ((u & 0x7fffffffu) > 0x7f800000u) return uint16_t((u >> 16) | 0x0040u);  // quiet NaN
  uint32_t round = ((u >> 16) & 1u) + 0x7fffu;
  return uint16_t((u + round) >> 16);
}

// Generic scalar load/store helpers so kernels template over bf16 / f32.
template <typename T>
ZTA_DEV float to_f32(T v);
template <>
ZTA_DEV float to_f32<float>(float v) { return v; }
template <>
ZTA_DEV float to_f32<uint16_t>(uint16_t v) { return bf16_to_f32(v); }

template <typename T>
ZTA_DEV T from_f32(float v);
template <>
ZTA_DEV float from_f32<float>(float v) { return v; }
template <>
ZTA_DEV uint16_t from_f32<uint16_t>(float v) { return f32_to_bf16(v); }

// Vector types for wide loads (G13: always vectorize bf16 loads).
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(2))) float f32x2;
typedef __attribute__((ext_vector_type(4))) short s16x4;
typedef __attribute__((ext_vector_type(8))) short s16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

// ---------------------------------------------------------------------------
// Wave reductions (64-wide butterfly; __shfl_xor covers all 64 lanes on CDNA)
// ---------------------------------------------------------------------------
ZTA_DEV float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

ZTA_DEV float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// Reduce across a 32-lane half (used for 32x32 MFMA row stats).
ZTA_DEV float half_reduce_sum(float v) {
#pragma unroll
  for (int off = 16; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

ZTA_DEV float half_reduce_max(float v) {
#pragma unroll
  for (int off = 16; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// Block reduction via LDS: `scratch` must hold >= blockDim.x/WAVE floats.
ZTA_DEV float block_reduce_sum(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  const int nw = blockDim.x / WAVE;
  v = (threadIdx.x < nw) ? scratch[threadIdx.x] : 0.0f;
  if (wid == 0) v = wave_reduce_sum(v);
  if (threadIdx.x == 0) scratch[0] = v;
  __syncthreads();
  v = scratch[0];
  __syncthreads();
  return v;
}

// ---------------------------------------------------------------------------
// Counter-based RNG (splitmix64 finalizer) for in-kernel dropout: the same
// (seed, index) always produces the same uniform — forward and backward
// regenerate identical masks without storing them.
// ---------------------------------------------------------------------------
ZTA_DEV uint64_t mix64(uint64_t x) {
  x += 0x9e3779b97f4a7c15ull;
  x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ull;
  x = (x ^ (x >> 27)) * 0x94d049bb133111ebull;
  return x ^ (x >> 31);
}

ZTA_DEV float uniform01(uint64_t seed, uint64_t idx) {
  // top 24 bits -> [0, 1)
  return float(mix64(seed ^ mix64(idx)) >> 40) * (1.0f / 16777216.0f);
}

// ---------------------------------------------------------------------------
// Attention-dropout RNG: murmur3-finalizer 32-bit hash, 4x 8-bit thresholds
// per call, shared by the fwd / dq / dkdv kernels so the mask regenerates
// identically in every orientation (and cheaply: ~8 VALU per 4 elements —
// the dkdv kernel evaluates it with q varying per register, 16 calls per
// subtile, so hash cost matters). Element (qi, kj) keeps iff byte
// (bits(seed, bh*T+qi, kj>>2) >> 8*(kj&3)) & 0xff >= thr, thr = round(p*256);
// rescale by 256/(256-thr). Mirrored in Python by ops.reference.drop_mask.
// ---------------------------------------------------------------------------
ZTA_DEV uint32_t mix32(uint32_t x) {
  x ^= x >> 16;
  x *= 0x85ebca6bu;
  x ^= x >> 13;
  x *= 0xc2b2ae35u;
  x ^= x >> 16;
  return x;
}

ZTA_DEV uint32_t drop_bits32(uint32_t seed, int bhT_qi, int kgroup) {
  return mix32(seed ^ ((uint32_t)bhT_qi * 0x9e3779b9u) ^
               ((uint32_t)kgroup * 0x85ebca6bu));
}

// ---------------------------------------------------------------------------
// Grid sizing helper (G11): cap memory-bound grids, grid-stride the rest.
// ---------------------------------------------------------------------------
inline int capped_grid(long total_threads, int block, int cap = 2048) {
  long g = (total_threads + block - 1) / block;
  return int(g < cap ? (g > 1 ? g : 1) : cap);
}
