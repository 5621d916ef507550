// Shared helpers for the gfx950 (CDNA4, MI355X) kernels.
//
// Design rules (from the CDNA4 programming guide, applied throughout):
//  - wave64: every wave-width constant is 64, masks are 64-bit.
//  - bf16 memory traffic is vectorized: 8 bf16 = one uint4 = 16 B/lane.
//  - f32 accumulation everywhere; bf16 only at the memory boundary.
//  - memory-bound kernels use grid-stride loops capped near 2048 blocks.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#define DEV_INLINE __device__ __forceinline__

constexpr int kWave = 64;

using bf16_t = __hip_bfloat16;

// 8 bf16 in one 16-byte register quad — the coalescing sweet spot.
union bf16x8 {
  uint4 u;
  ushort s[8];
};

DEV_INLINE float bf16_to_f32(uint16_t x) {
  union { float f; uint32_t u; } cvt;
  cvt.u = uint32_t(x) << 16;
  return cvt.f;
}

DEV_INLINE uint16_t f32_to_bf16(float x) {
  union { float f; uint32_t u; } cvt;
  cvt.f = x;
  uint32_t u = cvt.u;
  // round-to-nearest-even
  uint32_t rounding = 0x7fffu + ((u >> 16) & 1u);
  u += rounding;
  return uint16_t(u >> 16);
}

// ---- KV-cache element codecs --------------------------------------
// The paged cache stores bf16 (default) or OCP fp8-e4m3 (opt-in,
// `kv_dtype: fp8`): half the HBM traffic per token and double the
// resident KV capacity of the same budget.  gfx950 converts natively
// (v_cvt_pk_f32_fp8 / v_cvt_pk_fp8_f32 — 2 elements per VALU op); all
// attention math stays f32.  Kernels touching the cache are templated
// on one of these codecs; a "vec8" is always 8 cache ELEMENTS (16 B
// bf16 / 8 B fp8), so per-lane geometry is unchanged.
typedef float f32x2_cvt __attribute__((ext_vector_type(2)));

struct CacheBF16 {
  using elem = uint16_t;
  using vec8 = uint4;
  static DEV_INLINE void to_f32(const vec8& raw, float* out8) {
    const bf16x8* p = reinterpret_cast<const bf16x8*>(&raw);
#pragma unroll
    for (int i = 0; i < 8; ++i) out8[i] = bf16_to_f32(p->s[i]);
  }
  static DEV_INLINE vec8 from_f32(const float* in8) {
    bf16x8 o;
#pragma unroll
    for (int i = 0; i < 8; ++i) o.s[i] = f32_to_bf16(in8[i]);
    return o.u;
  }
};

struct CacheFP8 {
  using elem = uint8_t;
  using vec8 = uint2;
  static DEV_INLINE void to_f32(const vec8& raw, float* out8) {
    const f32x2_cvt a = __builtin_amdgcn_cvt_pk_f32_fp8(raw.x, false);
    const f32x2_cvt b = __builtin_amdgcn_cvt_pk_f32_fp8(raw.x, true);
    const f32x2_cvt c = __builtin_amdgcn_cvt_pk_f32_fp8(raw.y, false);
    const f32x2_cvt d = __builtin_amdgcn_cvt_pk_f32_fp8(raw.y, true);
    out8[0] = a.x; out8[1] = a.y; out8[2] = b.x; out8[3] = b.y;
    out8[4] = c.x; out8[5] = c.y; out8[6] = d.x; out8[7] = d.y;
  }
  static DEV_INLINE vec8 from_f32(const float* in8) {
    vec8 o;
    o.x = __builtin_amdgcn_cvt_pk_fp8_f32(in8[0], in8[1], 0u, false);
    o.x = __builtin_amdgcn_cvt_pk_fp8_f32(in8[2], in8[3], o.x, true);
    o.y = __builtin_amdgcn_cvt_pk_fp8_f32(in8[4], in8[5], 0u, false);
    o.y = __builtin_amdgcn_cvt_pk_fp8_f32(in8[6], in8[7], o.y, true);
    return o;
  }
};

// Wave-wide reductions (64 lanes).
DEV_INLINE float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, kWave);
  return v;
}

DEV_INLINE float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, kWave));
  return v;
}

// Sum within a contiguous lane group of width W (power of two).
template <int W>
DEV_INLINE float group_sum(float v) {
#pragma unroll
  for (int off = W / 2; off > 0; off >>= 1) v += __shfl_xor(v, off, kWave);
  return v;
}

// Same reduction on the VALU pipe via DPP row ops — no ds_swizzle, no
// lgkmcnt stall chain (__shfl_xor issues on the DS pipe: the decode-attn
// hot loop measured 120 ds_swizzles + 94 lgkmcnt waits per iteration
// group before this).  Only valid for REDUCTIONS (W <= 16): after stage
// k every lane of a 2^k sub-group holds that sub-group's total, so a
// mirror — any representative of the complementary sub-group — replaces
// the exact xor partner that DPP cannot express.
template <int CTRL>
DEV_INLINE float dpp_mov_f32(float v) {
  return __builtin_bit_cast(
      float, __builtin_amdgcn_update_dpp(
                 0, __builtin_bit_cast(int, v), CTRL, 0xf, 0xf, true));
}

template <int W>
DEV_INLINE float group_sum_dpp(float v) {
  static_assert(W == 2 || W == 4 || W == 8 || W == 16);
  if constexpr (W >= 2) v += dpp_mov_f32<0xB1>(v);    // quad_perm ^1
  if constexpr (W >= 4) v += dpp_mov_f32<0x4E>(v);    // quad_perm ^2
  if constexpr (W >= 8) v += dpp_mov_f32<0x141>(v);   // row_half_mirror
  if constexpr (W >= 16) v += dpp_mov_f32<0x140>(v);  // row_mirror
  return v;
}

