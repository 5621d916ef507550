// Fused rotary embedding + paged-KV-cache append.
//
// One pass does what three would: rotate q in place, rotate k and write
// it straight into its cache slot, copy v into its slot — the k tensor is
// also updated in place (the prefill kernel reads it from the in-batch
// layout).  GPT-NeoX half-rotation (Llama style): pair (i, i + D/2).
//
// cos_sin layout: [max_pos, D] f32, row = [cos(0..D/2) | sin(0..D/2)].
// Cache layout: [num_blocks, n_kv_heads, block_size, D] bf16; slot s ->
// block s / block_size, row s % block_size.  slot_mapping[t] == -1 skips
// the cache write (tokens beyond the allocated budget never occur; -1 is
// used by tests).

#include "common.h"

namespace rlli {

namespace {

template <typename KC>
__global__ void rope_kv_kernel(uint16_t* __restrict__ q,
                               uint16_t* __restrict__ k,
                               uint16_t* __restrict__ v,
                               const int32_t* __restrict__ positions,
                               const float* __restrict__ cos_sin,
                               typename KC::elem* __restrict__ k_cache,
                               typename KC::elem* __restrict__ v_cache,
                               const int32_t* __restrict__ slot_mapping,
                               int n_q_heads, int n_kv_heads, int head_dim,
                               int block_size, int q_stride, int kv_stride) {
  const int t = blockIdx.x;
  const int pos = positions[t];
  const int half = head_dim / 2;
  const int vecs_per_head = half / 8;          // pair-vectors per head
  const float* cs_row = cos_sin + int64_t(pos) * head_dim;

  const int32_t slot = slot_mapping[t];
  const int64_t cache_base =
      slot >= 0 ? (int64_t(slot / block_size) * n_kv_heads * block_size +
                   int64_t(slot % block_size)) * head_dim
                : 0;
  // per-head cache offset: ((block*n_kv + h)*block_size + row)*D
  //  = cache_base + h*block_size*head_dim

  // ---- rotate q and k (and write k into cache) ----
  const int total_rot = (n_q_heads + n_kv_heads) * vecs_per_head;
  for (int w = threadIdx.x; w < total_rot; w += blockDim.x) {
    const int h = w / vecs_per_head;
    const int p8 = (w % vecs_per_head) * 8;    // first pair index of this vec
    const bool is_q = h < n_q_heads;
    uint16_t* base = is_q
        ? q + int64_t(t) * q_stride + int64_t(h) * head_dim
        : k + int64_t(t) * kv_stride + int64_t(h - n_q_heads) * head_dim;
    bf16x8 lo, hi, olo, ohi;
    lo.u = *reinterpret_cast<const uint4*>(base + p8);
    hi.u = *reinterpret_cast<const uint4*>(base + half + p8);
    float4 c01 = *reinterpret_cast<const float4*>(cs_row + p8);
    float4 c23 = *reinterpret_cast<const float4*>(cs_row + p8 + 4);
    float4 s01 = *reinterpret_cast<const float4*>(cs_row + half + p8);
    float4 s23 = *reinterpret_cast<const float4*>(cs_row + half + p8 + 4);
    const float cv[8] = {c01.x, c01.y, c01.z, c01.w, c23.x, c23.y, c23.z, c23.w};
    const float sv[8] = {s01.x, s01.y, s01.z, s01.w, s23.x, s23.y, s23.z, s23.w};
    float rlo[8], rhi[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const float x1 = bf16_to_f32(lo.s[i]);
      const float x2 = bf16_to_f32(hi.s[i]);
      rlo[i] = x1 * cv[i] - x2 * sv[i];
      rhi[i] = x2 * cv[i] + x1 * sv[i];
      olo.s[i] = f32_to_bf16(rlo[i]);
      ohi.s[i] = f32_to_bf16(rhi[i]);
    }
    *reinterpret_cast<uint4*>(base + p8) = olo.u;
    *reinterpret_cast<uint4*>(base + half + p8) = ohi.u;
    if (!is_q && slot >= 0) {
      typename KC::elem* dst = k_cache + cache_base +
                      int64_t(h - n_q_heads) * block_size * head_dim;
      *reinterpret_cast<typename KC::vec8*>(dst + p8) = KC::from_f32(rlo);
      *reinterpret_cast<typename KC::vec8*>(dst + half + p8) =
          KC::from_f32(rhi);
    }
  }

  // ---- copy v into cache ----
  if (slot >= 0) {
    const int total_v = n_kv_heads * (head_dim / 8);
    for (int w = threadIdx.x; w < total_v; w += blockDim.x) {
      const int h = w / (head_dim / 8);
      const int d8 = (w % (head_dim / 8)) * 8;
      bf16x8 vv;
      vv.u = *reinterpret_cast<const uint4*>(
          v + int64_t(t) * kv_stride + int64_t(h) * head_dim + d8);
      float vf[8];
#pragma unroll
      for (int i = 0; i < 8; ++i) vf[i] = bf16_to_f32(vv.s[i]);
      typename KC::elem* dst = v_cache + cache_base +
                      int64_t(h) * block_size * head_dim + d8;
      *reinterpret_cast<typename KC::vec8*>(dst) = KC::from_f32(vf);
    }
  }
}

}  // namespace

void launch_rope_kv_append(
    uint16_t* q, uint16_t* k, uint16_t* v,
    const int32_t* positions, const float* cos_sin,
    void* k_cache, void* v_cache,
    const int32_t* slot_mapping,
    int tokens, int n_q_heads, int n_kv_heads, int head_dim,
    int block_size, int q_stride, int kv_stride, bool cache_fp8,
    hipStream_t stream) {
  if (tokens == 0) return;
  const int threads = 256;
  if (cache_fp8)
    hipLaunchKernelGGL(rope_kv_kernel<CacheFP8>, dim3(tokens), dim3(threads),
                       0, stream, q, k, v, positions, cos_sin,
                       static_cast<CacheFP8::elem*>(k_cache),
                       static_cast<CacheFP8::elem*>(v_cache),
                       slot_mapping, n_q_heads, n_kv_heads, head_dim,
                       block_size, q_stride, kv_stride);
  else
    hipLaunchKernelGGL(rope_kv_kernel<CacheBF16>, dim3(tokens), dim3(threads),
                       0, stream, q, k, v, positions, cos_sin,
                       static_cast<CacheBF16::elem*>(k_cache),
                       static_cast<CacheBF16::elem*>(v_cache),
                       slot_mapping, n_q_heads, n_kv_heads, head_dim,
                       block_size, q_stride, kv_stride);
}

}  // namespace rlli
