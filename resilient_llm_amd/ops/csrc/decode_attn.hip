// Paged GQA decode attention (one new query token per sequence).
//
// Memory-bound: the cost is streaming each sequence's K/V exactly once
// from HBM.  Geometry per workgroup = (sequence, kv-head) — times a
// split-K factor when batch*n_kv_heads alone cannot fill 256 CUs
// (flash-decode: each of n_split WGs walks an interleaved share of the
// block chain, emits an f32 partial, and decode_combine_kernel reduces;
// 2.9-12.7x at batch<=8, bit-identical path at n_split==1).
// Within a workgroup, 256 threads = 4 waves:
//   - the GROUP (= n_q / n_kv, Llama-8B: 4) query heads that share this
//     kv-head ride along in registers, so K/V bytes are read ONCE for
//     all of them;
//   - a wave64 splits into 64/GW lane-groups of GW = head_dim/8 lanes;
//     each lane holds 8 dims (one bf16x8 = 16 B load — G13), each
//     lane-group walks its share of the tokens of a cache block, each
//     wave walks every 4th cache block (split-KV);
//   - each lane-group keeps a private online-softmax partial
//     (m, l, acc[GROUP][8]); the 4*(64/GW) partials merge through LDS at
//     the end (log-sum-exp combine).
// Lane-group token validity is group-uniform, so the tail branch never
// diverges within a lane-group's reduction.
// Reference parity: no kernel exists in aws-samples/sample-resilient-llm-inference
// (model execution lived in Bedrock) — greenfield
// against the BASELINE.json north-star kernel list.

#include "common.h"

#include <type_traits>

namespace rlli {

namespace {

constexpr float kNegInf = -1e30f;

// FUSED variant: rope(q,k) + cache append of the new token + its
// self-attention term live INSIDE this kernel (decode fuses 2 kernels
// into 1; the self token is handled in registers by partial 0, so no
// in-kernel cache write->read ordering exists — future steps read the
// appended k/v after the kernel boundary).
// BS: compile-time cache block size (0 = runtime).  The engine's paged
// cache uses 16; baking it makes tok_per_grp a constant, so the K/V
// issue loop fully unrolls with no per-load `i < tok_per_grp` branch
// (the runtime form emitted s_cbranch between every load pair,
// fragmenting the batch and forcing early vmcnt waits).
// ABL: perf-ablation variants (RLLI_ATTN_ABLATE, results intentionally
// WRONG, never used for real inference): 1 = skip LDS publish/merge,
// 2 = skip online softmax, 3 = loads only.  Isolates which phase owns
// the wall time (guide pitfall 8: ablate before optimizing).
template <int GW, int GROUP, bool FUSED, int BS = 0, int ABL = 0,
          typename KC = CacheBF16>
__global__ __launch_bounds__(256)
void decode_attn_kernel(const uint16_t* __restrict__ q,
                        const typename KC::elem* __restrict__ k_cache,
                        const typename KC::elem* __restrict__ v_cache,
                        const int32_t* __restrict__ block_table,
                        const int32_t* __restrict__ seq_lens,
                        uint16_t* __restrict__ out,
                        int n_kv_heads, int block_size_rt, int max_blocks,
                        float scale, int q_stride,
                        const uint16_t* __restrict__ k_src,
                        const uint16_t* __restrict__ v_src,
                        const int32_t* __restrict__ positions,
                        const float* __restrict__ cos_sin,
                        const int32_t* __restrict__ slot_mapping,
                        typename KC::elem* __restrict__ k_cache_w,
                        typename KC::elem* __restrict__ v_cache_w,
                        int n_split,
                        float* __restrict__ part_out,
                        float* __restrict__ part_ml) {
  constexpr int D = GW * 8;
  constexpr int GPW = 64 / GW;                 // lane-groups per wave
  const int block_size = BS ? BS : block_size_rt;
  // split-K (flash-decode): when batch*n_kv_heads can't fill the chip,
  // n_split workgroups share one (sequence, kv-head) — each walks an
  // interleaved subset of the cache blocks and emits an f32 partial
  // (unnormalized acc + running max/denominator); a tiny combine
  // kernel reduces them.  n_split==1 is the classic single-WG path.
  const int split = blockIdx.x % n_split;
  const int sk = blockIdx.x / n_split;
  const int seq = sk / n_kv_heads;
  const int kvh = sk % n_kv_heads;
  const int len = seq_lens[seq];
  const int lane = threadIdx.x & (kWave - 1);
  const int wave = threadIdx.x >> 6;
  const int n_waves = blockDim.x >> 6;
  const int group = lane / GW;
  const int gl = lane % GW;
  const int d0 = gl * 8;
  const int n_part = n_waves;    // one partial per wave (in-wave merged)

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* accs = reinterpret_cast<float*>(smem_raw);       // [n_part][GROUP][D]
  float* ml = accs + n_part * GROUP * D;                  // [n_part][GROUP][2]

  // ---- load the GROUP query heads that map to this kv head ----
  float qv[GROUP][8];
#pragma unroll
  for (int h = 0; h < GROUP; ++h) {
    bf16x8 qh;
    qh.u = *reinterpret_cast<const uint4*>(
        q + int64_t(seq) * q_stride + (kvh * GROUP + h) * D + d0);
#pragma unroll
    for (int i = 0; i < 8; ++i) qv[h][i] = bf16_to_f32(qh.s[i]);
  }

  // rope coefficients for this lane's 8 dims (pair index = d mod D/2);
  // the rotation partner's raw value sits GW/2 lanes away in the group
  float cs_c[FUSED ? 8 : 1], cs_s[FUSED ? 8 : 1];
  bool first_half = true;
  if constexpr (FUSED) {
    constexpr int HALF = D / 2;
    const int pos = positions[seq];
    first_half = d0 < HALF;
    const int pidx = d0 % HALF;
    const float* cs = cos_sin + int64_t(pos) * D;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      cs_c[i] = cs[pidx + i];
      cs_s[i] = cs[HALF + pidx + i];
    }
    // rotate every query head in place
#pragma unroll
    for (int h = 0; h < GROUP; ++h) {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const float own = qv[h][i];
        const float other = __shfl_xor(own, GW / 2, kWave);
        qv[h][i] = first_half ? own * cs_c[i] - other * cs_s[i]
                              : own * cs_c[i] + other * cs_s[i];
      }
    }
  }

  float m[GROUP], l[GROUP], acc[GROUP][8];
#pragma unroll
  for (int h = 0; h < GROUP; ++h) {
    m[h] = kNegInf;
    l[h] = 0.f;
#pragma unroll
    for (int i = 0; i < 8; ++i) acc[h][i] = 0.f;
  }

  if constexpr (FUSED) {
    // group 0 of wave 0: rope k, append k/v to the cache, and fold the
    // NEW token's self-attention term into partial 0
    if (wave == 0 && group == 0 && split == 0) {
      bf16x8 kr, vr;
      kr.u = *reinterpret_cast<const uint4*>(
          k_src + int64_t(seq) * q_stride + kvh * D + d0);
      vr.u = *reinterpret_cast<const uint4*>(
          v_src + int64_t(seq) * q_stride + kvh * D + d0);
      float kf[8], vf[8];
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const float own = bf16_to_f32(kr.s[i]);
        const float other = __shfl_xor(own, GW / 2, kWave);
        kf[i] = first_half ? own * cs_c[i] - other * cs_s[i]
                           : own * cs_c[i] + other * cs_s[i];
        vf[i] = bf16_to_f32(vr.s[i]);
      }
      const int32_t slot = slot_mapping[seq];
      const int64_t cbase =
          (int64_t(slot / block_size) * n_kv_heads + kvh) * block_size * D +
          int64_t(slot % block_size) * D + d0;
      // append at cache precision (bf16 identity / fp8 cvt_pk); the
      // self-attention term below uses the STORED values so future
      // steps reproduce this step's math exactly
      const typename KC::vec8 kw = KC::from_f32(kf);
      const typename KC::vec8 vw = KC::from_f32(vf);
      *reinterpret_cast<typename KC::vec8*>(k_cache_w + cbase) = kw;
      *reinterpret_cast<typename KC::vec8*>(v_cache_w + cbase) = vw;
      KC::to_f32(kw, kf);
      KC::to_f32(vw, vf);
#pragma unroll
      for (int h = 0; h < GROUP; ++h) {
        float s = 0.f;
#pragma unroll
        for (int i = 0; i < 8; ++i) s += qv[h][i] * kf[i];
        s = group_sum_dpp<GW>(s) * scale;
        m[h] = s;
        l[h] = 1.f;
#pragma unroll
        for (int i = 0; i < 8; ++i) acc[h][i] = vf[i];
      }
    }
  }

  // Latency discipline: batch-issue ALL of a cache block's K/V vectors
  // for this lane-group (TPG k-loads + TPG v-loads back to back, plus
  // the NEXT block's K prefetched) before any arithmetic touches them —
  // one lane then has ~2*TPG+ loads in flight instead of 2, which is
  // what moves this kernel from latency-bound to bandwidth-bound
  // (guide Guideline 7/15: waits belong at the first consumer).
  const int len_cache = FUSED ? len - 1 : len;   // FUSED: prefix only
  const int n_blocks = (len_cache + block_size - 1) / block_size;
  constexpr int TPG = 16 / GPW >= 1 ? 16 / GPW : 1;   // tokens per group @bs16
  const int tok_per_grp = block_size / GPW;

  // phys index fetched one pipeline stage AHEAD of its block's K/V
  // issue: with the load inside issue_block the address of every K/V
  // load depended on a just-issued block_table load and the compiler
  // had to drain vmcnt before computing it, serializing the double
  // buffer.
  auto fetch_phys = [&](int b) -> int {
    return block_table[int64_t(seq) * max_blocks + b];
  };
  using cvec8 = typename KC::vec8;
  auto issue_block = [&](int phys, cvec8 (&kraw)[TPG], cvec8 (&vraw)[TPG]) {
    const int64_t kv_base =
        (int64_t(phys) * n_kv_heads + kvh) * block_size * D;
#pragma unroll
    for (int i = 0; i < TPG; ++i) {
      if (i < tok_per_grp) {
        const int tok = group + GPW * i;
        kraw[i] = *reinterpret_cast<const cvec8*>(
            k_cache + kv_base + int64_t(tok) * D + d0);
        vraw[i] = *reinterpret_cast<const cvec8*>(
            v_cache + kv_base + int64_t(tok) * D + d0);
      }
    }
  };
  auto consume_block = [&](int b, const cvec8 (&kraw)[TPG],
                           const cvec8 (&vraw)[TPG]) {
#pragma unroll
    for (int i = 0; i < TPG; ++i) {
      if (i >= tok_per_grp) break;
      const int tok = group + GPW * i;         // token within the block
      if (b * block_size + tok >= len_cache) continue;   // group-uniform tail
      float kf[8], vf[8];
      KC::to_f32(kraw[i], kf);
      KC::to_f32(vraw[i], vf);

      if constexpr (ABL == 3) {              // loads only
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[0][j] += kf[j] + vf[j];
        continue;
      }
#pragma unroll
      for (int h = 0; h < GROUP; ++h) {
        float s = 0.f;
#pragma unroll
        for (int j = 0; j < 8; ++j) s += qv[h][j] * kf[j];
        s = group_sum_dpp<GW>(s);
        s *= scale;
        if constexpr (ABL == 2) {            // no online softmax
          l[h] += s;
#pragma unroll
          for (int j = 0; j < 8; ++j) acc[h][j] += s * vf[j];
          continue;
        }
        const float m_new = fmaxf(m[h], s);
        const float corr = __expf(m[h] - m_new);
        const float p = __expf(s - m_new);
        m[h] = m_new;
        l[h] = l[h] * corr + p;
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[h][j] = acc[h][j] * corr + p * vf[j];
      }
    }
  };

  // Block-batched softmax consume: all TPG scores per head are computed
  // before ONE running-max update for the whole block — the per-token
  // form's serial corr/exp chain was 4.6 us of the 20.8 us kernel at
  // batch 64 (RLLI_ATTN_ABLATE=2 vs 1); batching makes the TPG exps
  // independent and pays one corr per (block, head).  Invalid tail
  // tokens contribute exp(kNegInf - m_new) == 0 (kNegInf is a finite
  // -1e30, so no -inf minus -inf NaN is reachable; every walked block
  // holds at least one valid token, so m_new is always finite).
  auto consume_block_batched = [&](int bb, const cvec8 (&kraw)[TPG],
                                   const cvec8 (&vraw)[TPG]) {
    float vfs[TPG][8];
    float svals[GROUP][TPG];
#pragma unroll
    for (int i = 0; i < TPG; ++i) {
      if (i >= tok_per_grp) break;
      const int tok = group + GPW * i;
      const bool valid = bb * block_size + tok < len_cache;
      float kf[8];
      KC::to_f32(kraw[i], kf);
      KC::to_f32(vraw[i], vfs[i]);
#pragma unroll
      for (int h = 0; h < GROUP; ++h) {
        float s = 0.f;
#pragma unroll
        for (int j = 0; j < 8; ++j) s += qv[h][j] * kf[j];
        s = group_sum_dpp<GW>(s) * scale;
        svals[h][i] = valid ? s : kNegInf;
      }
    }
#pragma unroll
    for (int h = 0; h < GROUP; ++h) {
      float m_new = m[h];
#pragma unroll
      for (int i = 0; i < TPG; ++i) {
        if (i >= tok_per_grp) break;
        m_new = fmaxf(m_new, svals[h][i]);
      }
      const float corr = __expf(m[h] - m_new);
      float p[TPG], psum = 0.f;
#pragma unroll
      for (int i = 0; i < TPG; ++i) {
        if (i >= tok_per_grp) break;
        p[i] = __expf(svals[h][i] - m_new);
        psum += p[i];
      }
      m[h] = m_new;
      l[h] = l[h] * corr + psum;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float t = acc[h][j] * corr;
#pragma unroll
        for (int i = 0; i < TPG; ++i) {
          if (i >= tok_per_grp) break;
          t += p[i] * vfs[i][j];
        }
        acc[h][j] = t;
      }
    }
  };

  const int bstride = n_waves * n_split;     // global block-chain count
  if (tok_per_grp <= TPG) {
    // depth-2 software pipeline over cache blocks: the NEXT block's
    // K/V loads are in flight while this block computes (named A/B
    // buffers, statically indexed — guide §5.4 rule 20)
    cvec8 kA[TPG], vA[TPG], kB[TPG], vB[TPG];
    int b = split * n_waves + wave;
    int physA = 0, physB = 0;
    if (b < n_blocks) {
      physA = fetch_phys(b);
      if (b + bstride < n_blocks) physB = fetch_phys(b + bstride);
      issue_block(physA, kA, vA);
    }
    auto consume = [&](int bb, const cvec8 (&kraw)[TPG],
                       const cvec8 (&vraw)[TPG]) {
      if constexpr (ABL == 0)
        consume_block_batched(bb, kraw, vraw);
      else
        consume_block(bb, kraw, vraw);
    };
    while (b < n_blocks) {
      if (b + bstride < n_blocks) {
        issue_block(physB, kB, vB);
        if (b + 2 * bstride < n_blocks) physA = fetch_phys(b + 2 * bstride);
      }
      consume(b, kA, vA);
      b += bstride;
      if (b >= n_blocks) break;
      if (b + bstride < n_blocks) {
        issue_block(physA, kA, vA);
        if (b + 2 * bstride < n_blocks) physB = fetch_phys(b + 2 * bstride);
      }
      consume(b, kB, vB);
      b += bstride;
    }
  } else {
    // oversized cache blocks: plain per-sub-batch loop
    for (int b = split * n_waves + wave; b < n_blocks; b += bstride) {
      const int phys = block_table[int64_t(seq) * max_blocks + b];
      const int64_t kv_base =
          (int64_t(phys) * n_kv_heads + kvh) * block_size * D;
      for (int bi = 0; bi < tok_per_grp; ++bi) {
        const int tok = group + GPW * bi;
        if (b * block_size + tok >= len_cache) continue;
        float kf[8], vf[8];
        KC::to_f32(*reinterpret_cast<const cvec8*>(
            k_cache + kv_base + int64_t(tok) * D + d0), kf);
        KC::to_f32(*reinterpret_cast<const cvec8*>(
            v_cache + kv_base + int64_t(tok) * D + d0), vf);
#pragma unroll
        for (int h = 0; h < GROUP; ++h) {
          float s = 0.f;
#pragma unroll
          for (int j = 0; j < 8; ++j) s += qv[h][j] * kf[j];
          s = group_sum_dpp<GW>(s);
          s *= scale;
          const float m_new = fmaxf(m[h], s);
          const float corr = __expf(m[h] - m_new);
          const float p = __expf(s - m_new);
          m[h] = m_new;
          l[h] = l[h] * corr + p;
#pragma unroll
          for (int j = 0; j < 8; ++j)
            acc[h][j] = acc[h][j] * corr + p * vf[j];
        }
      }
    }
  }

  if constexpr (ABL != 0) {
    // ablation: no LDS publish/merge — wave 0's lane-groups dump their
    // raw partials straight to out (wrong values, comparable traffic)
    if (wave == 0) {
#pragma unroll
      for (int h = 0; h < GROUP; ++h) {
        bf16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j) o.s[j] = f32_to_bf16(acc[h][j] + l[h]);
        if (group == 0)
          *reinterpret_cast<uint4*>(
              out + (int64_t(seq) * n_kv_heads * GROUP + kvh * GROUP + h) * D +
              d0) = o.u;
      }
    }
    return;
  }

  // ---- in-wave cross-group merge (log-sum-exp): lanes off apart hold
  // the same d0 slice for sibling lane-groups, so GPW partials combine
  // with shuffles instead of LDS round-trips — the LDS image shrinks
  // GPWx (16 -> 4 partials at GW=16) and the final scan with it (the
  // full publish+merge was 4.1 us of the 20.8 us kernel at batch 64,
  // RLLI_ATTN_ABLATE=1 vs 0).  kNegInf is finite, so idle lane-groups
  // (m = kNegInf, l = 0) combine as weight-0 with no -inf-inf NaN.
#pragma unroll
  for (int off = GW; off < 64; off <<= 1) {
#pragma unroll
    for (int h = 0; h < GROUP; ++h) {
      const float m_o = __shfl_xor(m[h], off, kWave);
      const float l_o = __shfl_xor(l[h], off, kWave);
      const float M = fmaxf(m[h], m_o);
      const float wa = __expf(m[h] - M);
      const float wb = __expf(m_o - M);
      l[h] = l[h] * wa + l_o * wb;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        acc[h][j] = acc[h][j] * wa + __shfl_xor(acc[h][j], off, kWave) * wb;
      m[h] = M;
    }
  }

  // ---- publish one partial per WAVE (group 0 lanes carry the merged
  // value for their d0 slice) ----
  if (group == 0) {
#pragma unroll
    for (int h = 0; h < GROUP; ++h) {
      float* dst = accs + ((int64_t(wave) * GROUP + h) * D) + d0;
#pragma unroll
      for (int j = 0; j < 8; ++j) dst[j] = acc[h][j];
      if (gl == 0) {
        ml[(wave * GROUP + h) * 2 + 0] = m[h];
        ml[(wave * GROUP + h) * 2 + 1] = l[h];
      }
    }
  }
  __syncthreads();

  // ---- merge: each thread owns (head, dim) output elements ----
  const int n_wpart = n_waves;
  for (int idx = threadIdx.x; idx < GROUP * D; idx += blockDim.x) {
    const int h = idx / D;
    const int d = idx % D;
    float M = kNegInf;
    for (int p = 0; p < n_wpart; ++p)
      M = fmaxf(M, ml[(p * GROUP + h) * 2 + 0]);
    float num = 0.f, den = 0.f;
    for (int p = 0; p < n_wpart; ++p) {
      const float w = __expf(ml[(p * GROUP + h) * 2 + 0] - M);
      num += w * accs[(int64_t(p) * GROUP + h) * D + d];
      den += w * ml[(p * GROUP + h) * 2 + 1];
    }
    if (n_split == 1) {
      out[(int64_t(seq) * n_kv_heads * GROUP + kvh * GROUP + h) * D + d] =
          f32_to_bf16(den > 0.f ? num / den : 0.f);
    } else {
      // unnormalized partial in this split's max-frame; a split whose
      // block range was empty publishes M=-inf/den=0 and combines to 0
      const int64_t pb = (int64_t(sk) * n_split + split) * GROUP + h;
      part_out[pb * D + d] = num;
      if (d == 0) {
        part_ml[pb * 2 + 0] = M;
        part_ml[pb * 2 + 1] = den;
      }
    }
  }
}

// Reduce n_split partials per (sequence, kv-head): one WG per (seq,
// kv-head), threads stride over GROUP*D output elements.
__global__ __launch_bounds__(256)
void decode_combine_kernel(const float* __restrict__ part,
                           const float* __restrict__ ml,
                           uint16_t* __restrict__ out,
                           int n_split, int group, int D) {
  const int sk = blockIdx.x;
  const int ghd = group * D;
  for (int idx = threadIdx.x; idx < ghd; idx += blockDim.x) {
    const int h = idx / D;
    float M = kNegInf;
    for (int s = 0; s < n_split; ++s)
      M = fmaxf(M, ml[((int64_t(sk) * n_split + s) * group + h) * 2]);
    float num = 0.f, den = 0.f;
    if (M > kNegInf) {
      for (int s = 0; s < n_split; ++s) {
        const int64_t b = (int64_t(sk) * n_split + s) * group + h;
        const float w = __expf(ml[b * 2] - M);
        num += w * part[(int64_t(sk) * n_split + s) * ghd + idx];
        den += w * ml[b * 2 + 1];
      }
    }
    out[int64_t(sk) * ghd + idx] = f32_to_bf16(den > 0.f ? num / den : 0.f);
  }
}

struct FusedArgs {
  const uint16_t* k_src = nullptr;
  const uint16_t* v_src = nullptr;
  const int32_t* positions = nullptr;
  const float* cos_sin = nullptr;
  const int32_t* slot_mapping = nullptr;
  void* k_cache_w = nullptr;     // cache-codec elem type
  void* v_cache_w = nullptr;
};

template <int GW, int GROUP, typename KC>
void dispatch_decode(const uint16_t* q, const void* k_cache_v,
                     const void* v_cache_v, const int32_t* block_table,
                     const int32_t* seq_lens, uint16_t* out, int batch,
                     int n_kv_heads, int block_size, int max_blocks,
                     float scale, int q_stride, const FusedArgs* fa,
                     int n_split, float* part_out, float* part_ml,
                     hipStream_t stream) {
  using CElem = typename KC::elem;
  const CElem* k_cache = static_cast<const CElem*>(k_cache_v);
  const CElem* v_cache = static_cast<const CElem*>(v_cache_v);
  constexpr int D = GW * 8;
  const int n_part = 4;          // one in-wave-merged partial per wave
  const size_t smem = size_t(n_part) * GROUP * (D + 2) * sizeof(float);
  const dim3 grid(batch * n_kv_heads * n_split);
  auto launch = [&](auto bs_tag, auto abl_tag) {
    constexpr int BS = decltype(bs_tag)::value;
    constexpr int ABL = decltype(abl_tag)::value;
    if (fa != nullptr) {
      hipLaunchKernelGGL((decode_attn_kernel<GW, GROUP, true, BS, ABL, KC>),
                         grid, dim3(256), smem, stream,
                         q, k_cache, v_cache, block_table, seq_lens, out,
                         n_kv_heads, block_size, max_blocks, scale, q_stride,
                         fa->k_src, fa->v_src, fa->positions, fa->cos_sin,
                         fa->slot_mapping,
                         static_cast<CElem*>(fa->k_cache_w),
                         static_cast<CElem*>(fa->v_cache_w),
                         n_split, part_out, part_ml);
    } else {
      hipLaunchKernelGGL((decode_attn_kernel<GW, GROUP, false, BS, ABL, KC>),
                         grid, dim3(256), smem, stream,
                         q, k_cache, v_cache, block_table, seq_lens, out,
                         n_kv_heads, block_size, max_blocks, scale, q_stride,
                         nullptr, nullptr, nullptr, nullptr, nullptr,
                         nullptr, nullptr, n_split, part_out, part_ml);
    }
  };
  using I0 = std::integral_constant<int, 0>;
  // RLLI_ATTN_ABLATE: perf-ablation kernels (WRONG results — bench
  // only; instantiated for the bf16 cache only)
  static const int abl = [] {
    const char* e = std::getenv("RLLI_ATTN_ABLATE");
    return e ? atoi(e) : 0;
  }();
  if (block_size == 16) {
    using B16 = std::integral_constant<int, 16>;
    if constexpr (std::is_same_v<KC, CacheBF16>) {
      switch (abl) {
        case 1: launch(B16{}, std::integral_constant<int, 1>{}); break;
        case 2: launch(B16{}, std::integral_constant<int, 2>{}); break;
        case 3: launch(B16{}, std::integral_constant<int, 3>{}); break;
        default: launch(B16{}, I0{});
      }
    } else {
      launch(B16{}, I0{});
    }
  } else {
    launch(I0{}, I0{});
  }
  if (n_split > 1) {
    hipLaunchKernelGGL(decode_combine_kernel, dim3(batch * n_kv_heads),
                       dim3(256), 0, stream, part_out, part_ml, out,
                       n_split, GROUP, D);
  }
}

}  // namespace

// Split factor so decode fills the chip: 256 CUs want >=512 workgroups
// in flight; small batches (interactive / long-context) get their
// sequences split across block-chains instead of idling CUs.
int decode_attn_n_split(int batch, int n_kv_heads) {
  // RLLI_ATTN_SPLIT forces the factor (A/B tuning); 0/unset = heuristic
  static const int forced = [] {
    const char* e = std::getenv("RLLI_ATTN_SPLIT");
    return e ? atoi(e) : 0;
  }();
  if (forced > 0) return forced > 16 ? 16 : forced;
  const int wgs = batch * n_kv_heads;
  if (wgs >= 384) return 1;
  int n = (512 + wgs - 1) / wgs;
  return n > 16 ? 16 : n;
}

void launch_decode_attn_impl(const uint16_t* q, const void* k_cache,
                             const void* v_cache,
                             const int32_t* block_table,
                             const int32_t* seq_lens, uint16_t* out,
                             int batch, int n_q_heads, int n_kv_heads,
                             int head_dim, int block_size, int max_blocks,
                             float scale, int q_stride, const FusedArgs* fa,
                             int n_split, float* part_out, float* part_ml,
                             bool cache_fp8, hipStream_t stream) {
  if (batch == 0) return;
  if (part_out == nullptr) n_split = 1;
  const int group = n_q_heads / n_kv_heads;
  auto run = [&](auto gw_tag, auto group_tag) {
    if (cache_fp8)
      dispatch_decode<decltype(gw_tag)::value, decltype(group_tag)::value,
                      CacheFP8>(
          q, k_cache, v_cache, block_table, seq_lens, out, batch, n_kv_heads,
          block_size, max_blocks, scale, q_stride, fa, n_split, part_out,
          part_ml, stream);
    else
      dispatch_decode<decltype(gw_tag)::value, decltype(group_tag)::value,
                      CacheBF16>(
          q, k_cache, v_cache, block_table, seq_lens, out, batch, n_kv_heads,
          block_size, max_blocks, scale, q_stride, fa, n_split, part_out,
          part_ml, stream);
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
    else return;   // caller validates
  } else if (head_dim == 64) {
    if (group == 1) run(I8{}, G1{});
    else if (group == 2) run(I8{}, G2{});
    else if (group == 4) run(I8{}, G4{});
    else if (group == 8) run(I8{}, G8{});
    else return;
  }
}

void launch_decode_attn(const uint16_t* q, const void* k_cache,
                        const void* v_cache, const int32_t* block_table,
                        const int32_t* seq_lens, uint16_t* out, int batch,
                        int n_q_heads, int n_kv_heads, int head_dim,
                        int block_size, int max_blocks, float scale,
                        int q_stride, int n_split, float* part_out,
                        float* part_ml, bool cache_fp8, hipStream_t stream) {
  launch_decode_attn_impl(q, k_cache, v_cache, block_table, seq_lens, out,
                          batch, n_q_heads, n_kv_heads, head_dim, block_size,
                          max_blocks, scale, q_stride, nullptr,
                          n_split, part_out, part_ml, cache_fp8, stream);
}

void launch_decode_attn_fused(
    const uint16_t* qkv, void* k_cache, void* v_cache,
    const int32_t* block_table, const int32_t* seq_lens,
    const int32_t* positions, const float* cos_sin,
    const int32_t* slot_mapping, uint16_t* out, int batch, int n_q_heads,
    int n_kv_heads, int head_dim, int block_size, int max_blocks,
    float scale, int qkv_stride, int n_split, float* part_out,
    float* part_ml, bool cache_fp8, hipStream_t stream) {
  FusedArgs fa;
  fa.k_src = qkv + n_q_heads * head_dim;
  fa.v_src = qkv + (n_q_heads + n_kv_heads) * head_dim;
  fa.positions = positions;
  fa.cos_sin = cos_sin;
  fa.slot_mapping = slot_mapping;
  fa.k_cache_w = k_cache;
  fa.v_cache_w = v_cache;
  launch_decode_attn_impl(qkv, k_cache, v_cache, block_table, seq_lens, out,
                          batch, n_q_heads, n_kv_heads, head_dim, block_size,
                          max_blocks, scale, qkv_stride, &fa,
                          n_split, part_out, part_ml, cache_fp8, stream);
}

}  // namespace rlli
