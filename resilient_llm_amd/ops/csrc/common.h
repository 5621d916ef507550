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

