// MFMA chunked-prefill attention over the PAGED KV cache.
//
// Same tile structure as prefill_mfma.hip (32-row q-chunks, KVBLK=32,
// mfma_f32_16x16x32_bf16, online softmax on C fragments, LDS P bounce,
// cooperative V^T staging), but K/V come from the paged cache via block
// tables instead of the in-batch qkv tensor — so a chunk of Q attends
// over its sequence's ENTIRE cached prefix plus the chunk itself (whose
// K/V rope_kv_append just wrote).  This is what makes chunked prefill
// and mixed prefill/decode batches possible: per-chunk metadata
// (batch-row offset, position of the first row, row count, block-table
// row) comes from the engine.
//
// Cache block row [phys, kvh, tok, 0:D] is contiguous, so the QK^T
// B fragment is still a single 16-B load per lane, one block-table
// lookup per 16 kv positions (a KVBLK=64 tile spans four cache blocks).

#include "common.h"

#include <type_traits>

namespace rlli {

namespace {

using bf16x8_vec = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr float kNegInf = -1e30f;
constexpr int QBLK = 32;
constexpr int KVBLK = 64;
constexpr int PPAD = KVBLK + 8;

template <int D, typename KC = CacheBF16>
__global__ __launch_bounds__(256, 2)
void prefill_paged_kernel(const uint16_t* __restrict__ qkv,
                          const typename KC::elem* __restrict__ k_cache,
                          const typename KC::elem* __restrict__ v_cache,
                          const int32_t* __restrict__ chunk_row0,
                          const int32_t* __restrict__ chunk_pos0,
                          const int32_t* __restrict__ chunk_nrows,
                          const int32_t* __restrict__ chunk_btrow,
                          const int32_t* __restrict__ block_tables,
                          uint16_t* __restrict__ out,
                          int n_kv_heads, int group, int n_hw,
                          int qkv_stride, int max_blocks, int block_size,
                          float scale) {
  constexpr int CT = D / 16;
  constexpr int DC = D / 32;
  const int chunk = blockIdx.x / (n_kv_heads * n_hw);
  const int kvh = (blockIdx.x / n_hw) % n_kv_heads;
  const int hw = blockIdx.x % n_hw;
  const int heads_per_wg = group / n_hw;

  const int row0 = chunk_row0[chunk];          // first batch row of chunk
  const int pos0 = chunk_pos0[chunk];          // seq position of that row
  const int nrows = chunk_nrows[chunk];
  const int32_t* bt = block_tables + int64_t(chunk_btrow[chunk]) * max_blocks;
  const int n_q_heads = n_kv_heads * group;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int jcol = lane & 15;
  const int koct = lane >> 4;
  const bool active = wave < heads_per_wg;
  const int head = kvh * group + hw * heads_per_wg + wave;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  uint16_t* vt = reinterpret_cast<uint16_t*>(smem_raw);
  uint16_t* ks = vt + D * PPAD;                                  // KVBLK*D, swizzled
  uint16_t* p_lds = ks + KVBLK * D + wave * QBLK * PPAD;

  bf16x8_vec qf[2][DC];
#pragma unroll
  for (int qt = 0; qt < 2; ++qt) {
#pragma unroll
    for (int dc = 0; dc < DC; ++dc) {
      const int r = qt * 16 + jcol;
      uint4 raw = {0, 0, 0, 0};
      if (active && r < nrows)
        raw = *reinterpret_cast<const uint4*>(
            qkv + int64_t(row0 + r) * qkv_stride + head * D + dc * 32 +
            koct * 8);
      qf[qt][dc] = *reinterpret_cast<bf16x8_vec*>(&raw);
    }
  }

  f32x4 o_acc[2][CT];
#pragma unroll
  for (int qt = 0; qt < 2; ++qt)
#pragma unroll
    for (int ct = 0; ct < CT; ++ct) o_acc[qt][ct] = f32x4{0.f, 0.f, 0.f, 0.f};
  float m_st[2][4], l_st[2][4];
#pragma unroll
  for (int qt = 0; qt < 2; ++qt)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      m_st[qt][r] = kNegInf;
      l_st[qt][r] = 0.f;
    }

  // cache row address for kv position p (this kv head)
  auto kv_addr = [&](int p) -> int64_t {
    const int phys = bt[p / block_size];
    return (int64_t(phys) * n_kv_heads + kvh) * block_size * D +
           int64_t(p % block_size) * D;
  };

  const int kv_hi = pos0 + nrows;              // causal bound of the chunk

  for (int kv0 = 0; kv0 < kv_hi; kv0 += KVBLK) {
    __syncthreads();
    {
      constexpr int PIECES = KVBLK * D / 8 / 256;
#pragma unroll
      for (int pc = 0; pc < PIECES; ++pc) {
        const int idx = pc * 256 + threadIdx.x;
        const int kv = idx / (D / 8);
        const int d8 = (idx % (D / 8)) * 8;
        const int p = kv0 + kv;
        // stage through the cache codec: fp8 converts to bf16 HERE, so
        // the LDS images and every MFMA fragment below are unchanged
        typename KC::vec8 raw = {};
        typename KC::vec8 kraw = {};
        if (p < kv_hi) {
          const int64_t a = kv_addr(p);
          raw = *reinterpret_cast<const typename KC::vec8*>(v_cache + a + d8);
          kraw = *reinterpret_cast<const typename KC::vec8*>(k_cache + a + d8);
        }
        float vf8[8], kf8[8];
        KC::to_f32(raw, vf8);
        KC::to_f32(kraw, kf8);
        bf16x8 piece, kpiece;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          piece.s[j] = f32_to_bf16(vf8[j]);
          kpiece.s[j] = f32_to_bf16(kf8[j]);
          vt[(d8 + j) * PPAD + kv] = piece.s[j];
        }
        *reinterpret_cast<uint4*>(
            reinterpret_cast<char*>(ks) +
            (kv * D * 2 + ((d8 * 2) ^ ((kv & (D / 8 - 1)) << 4)))) = kpiece.u;
      }
    }
    __syncthreads();

    constexpr int KT = KVBLK / 16;
    f32x4 s_acc[2][KT];
#pragma unroll
    for (int qt = 0; qt < 2; ++qt)
#pragma unroll
      for (int kt = 0; kt < KT; ++kt) s_acc[qt][kt] = f32x4{0.f, 0.f, 0.f, 0.f};
    if (active) {
#pragma unroll
      for (int kt = 0; kt < KT; ++kt) {
        bf16x8_vec kf[DC];
        const int krow = kt * 16 + jcol;
#pragma unroll
        for (int dc = 0; dc < DC; ++dc) {
          uint4 raw = *reinterpret_cast<const uint4*>(
              reinterpret_cast<const char*>(ks) +
              (krow * D * 2 +
               (((dc * 32 + koct * 8) * 2) ^ ((krow & (D / 8 - 1)) << 4))));
          kf[dc] = *reinterpret_cast<bf16x8_vec*>(&raw);
        }
#pragma unroll
        for (int qt = 0; qt < 2; ++qt)
#pragma unroll
          for (int dc = 0; dc < DC; ++dc)
            s_acc[qt][kt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                qf[qt][dc], kf[dc], s_acc[qt][kt], 0, 0, 0);
      }

#pragma unroll
      for (int qt = 0; qt < 2; ++qt) {
        float corr[4];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int qpos = pos0 + qt * 16 + koct * 4 + r;   // seq position
          float sv[KT];
          float mx = kNegInf;
#pragma unroll
          for (int kt = 0; kt < KT; ++kt) {
            float s = s_acc[qt][kt][r] * scale;
            const int c = kv0 + kt * 16 + jcol;
            if (c > qpos || c >= kv_hi) s = kNegInf;
            sv[kt] = s;
            mx = fmaxf(mx, s);
          }
#pragma unroll
          for (int off = 8; off > 0; off >>= 1)
            mx = fmaxf(mx, __shfl_xor(mx, off, kWave));
          const float m_new = fmaxf(m_st[qt][r], mx);
          corr[r] = __expf(m_st[qt][r] - m_new);
          m_st[qt][r] = m_new;
          float rs = 0.f;
          const int prow = qt * 16 + koct * 4 + r;
#pragma unroll
          for (int kt = 0; kt < KT; ++kt) {
            const float p = __expf(sv[kt] - m_new);
            rs += p;
            p_lds[prow * PPAD + kt * 16 + jcol] = f32_to_bf16(p);
          }
#pragma unroll
          for (int off = 8; off > 0; off >>= 1)
            rs += __shfl_xor(rs, off, kWave);
          l_st[qt][r] = l_st[qt][r] * corr[r] + rs;
        }
#pragma unroll
        for (int ct = 0; ct < CT; ++ct)
#pragma unroll
          for (int r = 0; r < 4; ++r) o_acc[qt][ct][r] *= corr[r];
      }

#pragma unroll
      for (int qt = 0; qt < 2; ++qt) {
#pragma unroll
        for (int kb = 0; kb < KVBLK / 32; ++kb) {
          uint4 praw = *reinterpret_cast<const uint4*>(
              p_lds + (qt * 16 + jcol) * PPAD + kb * 32 + koct * 8);
          bf16x8_vec pfrag = *reinterpret_cast<bf16x8_vec*>(&praw);
#pragma unroll
          for (int ct = 0; ct < CT; ++ct) {
            uint4 vraw = *reinterpret_cast<const uint4*>(
                vt + (ct * 16 + jcol) * PPAD + kb * 32 + koct * 8);
            bf16x8_vec vfrag = *reinterpret_cast<bf16x8_vec*>(&vraw);
            o_acc[qt][ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                pfrag, vfrag, o_acc[qt][ct], 0, 0, 0);
          }
        }
      }
    }
  }

  if (active) {
#pragma unroll
    for (int qt = 0; qt < 2; ++qt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int rr = qt * 16 + koct * 4 + r;
        if (rr >= nrows) continue;
        const float inv_l = l_st[qt][r] > 0.f ? 1.f / l_st[qt][r] : 0.f;
#pragma unroll
        for (int ct = 0; ct < CT; ++ct)
          out[int64_t(row0 + rr) * n_q_heads * D + head * D + ct * 16 + jcol] =
              f32_to_bf16(o_acc[qt][ct][r] * inv_l);
      }
    }
  }
}

}  // namespace

void launch_prefill_paged(const uint16_t* qkv, const void* k_cache,
                          const void* v_cache, const int32_t* chunk_row0,
                          const int32_t* chunk_pos0,
                          const int32_t* chunk_nrows,
                          const int32_t* chunk_btrow,
                          const int32_t* block_tables, uint16_t* out,
                          int n_chunks, int n_kv_heads, int group,
                          int head_dim, int qkv_stride, int max_blocks,
                          int block_size, float scale, bool cache_fp8,
                          hipStream_t stream) {
  const int n_hw = group > 4 ? group / 4 : 1;
  const int blocks = n_chunks * n_kv_heads * n_hw;
  const size_t smem = size_t(head_dim) * PPAD * 2 +
                      size_t(KVBLK) * head_dim * 2 +
                      size_t(4) * QBLK * PPAD * 2;
  auto go = [&](auto d_tag, auto kc_tag) {
    using KC = decltype(kc_tag);
    hipLaunchKernelGGL((prefill_paged_kernel<decltype(d_tag)::value, KC>),
                       dim3(blocks), dim3(256), smem, stream, qkv,
                       static_cast<const typename KC::elem*>(k_cache),
                       static_cast<const typename KC::elem*>(v_cache),
                       chunk_row0, chunk_pos0, chunk_nrows, chunk_btrow,
                       block_tables, out, n_kv_heads, group, n_hw,
                       qkv_stride, max_blocks, block_size, scale);
  };
  using D128 = std::integral_constant<int, 128>;
  using D64 = std::integral_constant<int, 64>;
  if (head_dim == 128) {
    if (cache_fp8) go(D128{}, CacheFP8{}); else go(D128{}, CacheBF16{});
  } else if (head_dim == 64) {
    if (cache_fp8) go(D64{}, CacheFP8{}); else go(D64{}, CacheBF16{});
  }
}

}  // namespace rlli
