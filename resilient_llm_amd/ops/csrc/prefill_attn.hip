// Varlen causal prefill attention (flash-style online softmax, VALU path).
//
// v1 geometry mirrors decode_attn: one workgroup per (query token,
// kv-head); the GROUP query heads sharing the kv-head ride in registers;
// 4 waves x (64/GW) lane-groups stride over the causal prefix, each
// lane-group keeping an online-softmax partial merged through LDS.
//
// At serving prompt lengths (<= a few hundred tokens per chunk) the
// prefix K/V is L2/L3-resident, so the O(S^2) re-read stays on-die; the
// MFMA-tiled prefill kernel (32x32x16_bf16, LDS-staged K/V — guide §B)
// is the planned upgrade for long-context chunked prefill.  Attention is
// a small fraction of prefill FLOPs at these shapes (the GEMMs dominate).

#include "common.h"

namespace rlli {

namespace {

constexpr float kNegInf = -1e30f;

template <int GW, int GROUP>
__global__ __launch_bounds__(256)
void prefill_attn_kernel(const uint16_t* __restrict__ q,
                         const uint16_t* __restrict__ k,
                         const uint16_t* __restrict__ v,
                         const int32_t* __restrict__ cu_seqlens,
                         uint16_t* __restrict__ out,
                         int n_seqs, int n_kv_heads, float scale,
                         int q_stride, int kv_stride) {
  constexpr int D = GW * 8;
  constexpr int GPW = 64 / GW;
  const int t = blockIdx.x / n_kv_heads;       // global query token
  const int kvh = blockIdx.x % n_kv_heads;
  const int lane = threadIdx.x & (kWave - 1);
  const int wave = threadIdx.x >> 6;
  const int n_waves = blockDim.x >> 6;
  const int group = lane / GW;
  const int gl = lane % GW;
  const int d0 = gl * 8;
  const int n_part = n_waves * GPW;
  const int pid = wave * GPW + group;

  // binary search: seq s.t. cu[s] <= t < cu[s+1]
  int lo = 0, hi = n_seqs - 1;
  while (lo < hi) {
    const int mid = (lo + hi + 1) >> 1;
    if (cu_seqlens[mid] <= t) lo = mid; else hi = mid - 1;
  }
  const int seq_start = cu_seqlens[lo];
  const int kv_len = t - seq_start + 1;        // causal prefix incl. self

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* accs = reinterpret_cast<float*>(smem_raw);       // [n_part][GROUP][D]
  float* ml = accs + n_part * GROUP * D;                  // [n_part][GROUP][2]

  const int n_q_heads = n_kv_heads * GROUP;
  float qv[GROUP][8];
#pragma unroll
  for (int h = 0; h < GROUP; ++h) {
    bf16x8 qh;
    qh.u = *reinterpret_cast<const uint4*>(
        q + int64_t(t) * q_stride + (kvh * GROUP + h) * D + d0);
#pragma unroll
    for (int i = 0; i < 8; ++i) qv[h][i] = bf16_to_f32(qh.s[i]);
  }

  float m[GROUP], l[GROUP], acc[GROUP][8];
#pragma unroll
  for (int h = 0; h < GROUP; ++h) {
    m[h] = kNegInf;
    l[h] = 0.f;
#pragma unroll
    for (int i = 0; i < 8; ++i) acc[h][i] = 0.f;
  }

  // lane-group pid takes kv tokens pid, pid + n_part, ...
  for (int j = pid; j < kv_len; j += n_part) {
    const int64_t row = int64_t(seq_start + j) * kv_stride + kvh * D;
    bf16x8 kv8;
    kv8.u = *reinterpret_cast<const uint4*>(k + row + d0);
    float kf[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) kf[i] = bf16_to_f32(kv8.s[i]);
    bf16x8 vv8;
    vv8.u = *reinterpret_cast<const uint4*>(v + row + d0);
    float vf[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) vf[i] = bf16_to_f32(vv8.s[i]);

#pragma unroll
    for (int h = 0; h < GROUP; ++h) {
      float s = 0.f;
#pragma unroll
      for (int i = 0; i < 8; ++i) s += qv[h][i] * kf[i];
      s = group_sum_dpp<GW>(s);
      s *= scale;
      const float m_new = fmaxf(m[h], s);
      const float corr = __expf(m[h] - m_new);
      const float p = __expf(s - m_new);
      m[h] = m_new;
      l[h] = l[h] * corr + p;
#pragma unroll
      for (int i = 0; i < 8; ++i) acc[h][i] = acc[h][i] * corr + p * vf[i];
    }
  }

#pragma unroll
  for (int h = 0; h < GROUP; ++h) {
    float* dst = accs + ((int64_t(pid) * GROUP + h) * D) + d0;
#pragma unroll
    for (int i = 0; i < 8; ++i) dst[i] = acc[h][i];
    if (gl == 0) {
      ml[(pid * GROUP + h) * 2 + 0] = m[h];
      ml[(pid * GROUP + h) * 2 + 1] = l[h];
    }
  }
  __syncthreads();

  for (int idx = threadIdx.x; idx < GROUP * D; idx += blockDim.x) {
    const int h = idx / D;
    const int d = idx % D;
    float M = kNegInf;
    for (int p = 0; p < n_part; ++p)
      M = fmaxf(M, ml[(p * GROUP + h) * 2 + 0]);
    float num = 0.f, den = 0.f;
    for (int p = 0; p < n_part; ++p) {
      const float w = __expf(ml[(p * GROUP + h) * 2 + 0] - M);
      num += w * accs[(int64_t(p) * GROUP + h) * D + d];
      den += w * ml[(p * GROUP + h) * 2 + 1];
    }
    out[(int64_t(t) * n_q_heads + kvh * GROUP + h) * D + d] =
        f32_to_bf16(den > 0.f ? num / den : 0.f);
  }
}

template <int GW, int GROUP>
void dispatch_prefill(const uint16_t* q, const uint16_t* k, const uint16_t* v,
                      const int32_t* cu_seqlens, uint16_t* out, int n_seqs,
                      int total_tokens, int n_kv_heads, float scale,
                      int q_stride, int kv_stride, hipStream_t stream) {
  constexpr int D = GW * 8;
  constexpr int GPW = 64 / GW;
  const int n_part = 4 * GPW;
  const size_t smem = size_t(n_part) * GROUP * (D + 2) * sizeof(float);
  hipLaunchKernelGGL((prefill_attn_kernel<GW, GROUP>),
                     dim3(total_tokens * n_kv_heads), dim3(256), smem, stream,
                     q, k, v, cu_seqlens, out, n_seqs, n_kv_heads, scale,
                     q_stride, kv_stride);
}

}  // namespace

void launch_prefill_attn(const uint16_t* q, const uint16_t* k,
                         const uint16_t* v, const int32_t* cu_seqlens,
                         uint16_t* out, int n_seqs, int total_tokens,
                         int n_q_heads, int n_kv_heads, int head_dim,
                         float scale, int q_stride, int kv_stride,
                         hipStream_t stream) {
  if (total_tokens == 0) return;
  const int group = n_q_heads / n_kv_heads;
  auto run = [&](auto gw_tag, auto group_tag) {
    dispatch_prefill<decltype(gw_tag)::value, decltype(group_tag)::value>(
        q, k, v, cu_seqlens, out, n_seqs, total_tokens, n_kv_heads, scale,
        q_stride, kv_stride, stream);
  };
  using I8 = std::integral_constant<int, 8>;
  using I16 = std::integral_constant<int, 16>;
  using G1 = std::integral_constant<int, 1>;
  using G2 = std::integral_constant<int, 2>;
  using G4 = std::integral_constant<int, 4>;
  using G8 = std::integral_constant<int, 8>;
  if (head_dim == 128) {
    if (group == 1) run(I16{}, G1{});
    else if (group == 2) run(I16{}, G2{});
    else if (group == 4) run(I16{}, G4{});
    else if (group == 8) run(I16{}, G8{});
  } else if (head_dim == 64) {
    if (group == 1) run(I8{}, G1{});
    else if (group == 2) run(I8{}, G2{});
    else if (group == 4) run(I8{}, G4{});
    else if (group == 8) run(I8{}, G8{});
  }
}

}  // namespace rlli
