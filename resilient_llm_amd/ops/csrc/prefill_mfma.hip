// MFMA varlen causal prefill attention (flash-style online softmax).
//
// Replaces the VALU v1 (prefill_attn.hip) on the matrix cores: ~37 TF/s
// VALU -> MFMA tiles.  Geometry per workgroup = (32-row q-chunk of one
// sequence, kv-head), 4 waves:
//   - each wave owns ONE of the (up to 4) query heads of the GQA group
//     for the SAME 32 q rows, so all waves share the same causal range
//     and the same K/V tiles (Llama-8B GROUP=4 uses all 4; GROUP=8
//     splits across 2 workgroups);
//   - per KV tile (KVBLK=64): K and V are staged cooperatively ONCE per
//     workgroup in one barrier pair — K row-major but XOR-swizzled
//     (byte ^= (row & (D/8-1)) << 4) so the QK^T B-fragment read (16
//     lanes, 16 rows, same columns) is bank-conflict-free, V as padded
//     V^T so the PV B fragment is a contiguous ds_read_b128;
//     QK^T = 2x4 S-tiles of mfma_f32_16x16x32 accumulated over D
//     (B fragment = 16 contiguous bf16 of a K row — no transpose
//     needed); online softmax on the C-layout fragments (row r lives in
//     one 16-lane shfl group); P goes to bf16 through a per-wave padded
//     LDS buffer and comes back in A-fragment layout.
//   - O accumulates in registers ([2 q-tiles][D/16][4] f32), normalized
//     and stored at the end.
//
// Chunk tables (t0 / seq_start / seq_end per chunk) are built host-side
// from cu_seqlens.  Out-of-range K/V lanes load zeros and are masked to
// -inf before the softmax, so partial tail tiles never touch another
// sequence's rows.

#include "common.h"

namespace rlli {

namespace {

using bf16x8_vec = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr float kNegInf = -1e30f;
constexpr int QBLK = 32;
constexpr int KVBLK = 64;
constexpr int PPAD = KVBLK + 8;   // padded row stride (elements) for P / V^T

template <int D>
__global__ __launch_bounds__(256)
void prefill_mfma_kernel(const uint16_t* __restrict__ qkv,
                         const int32_t* __restrict__ chunk_t0,
                         const int32_t* __restrict__ chunk_seq_start,
                         const int32_t* __restrict__ chunk_seq_end,
                         uint16_t* __restrict__ out,
                         int n_kv_heads, int group, int n_hw,
                         int qkv_stride, float scale) {
  constexpr int CT = D / 16;            // output column tiles
  constexpr int DC = D / 32;            // k-dim chunks per S mfma
  const int chunk = blockIdx.x / (n_kv_heads * n_hw);
  const int kvh = (blockIdx.x / n_hw) % n_kv_heads;
  const int hw = blockIdx.x % n_hw;
  const int heads_per_wg = group / n_hw;       // <= 4

  const int t0 = chunk_t0[chunk];
  const int seq_start = chunk_seq_start[chunk];
  const int seq_end = chunk_seq_end[chunk];
  const int n_q_heads = n_kv_heads * group;
  const int kv_off = n_q_heads * D;            // k slice offset in qkv row
  const int v_off = (n_q_heads + n_kv_heads) * D;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int jcol = lane & 15;
  const int koct = lane >> 4;
  const bool active = wave < heads_per_wg;
  const int head = kvh * group + hw * heads_per_wg + wave;   // if active

  // LDS: V^T [D][PPAD] shared + per-wave P [QBLK][PPAD]
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  uint16_t* vt = reinterpret_cast<uint16_t*>(smem_raw);          // D*PPAD
  uint16_t* ks = vt + D * PPAD;                                  // KVBLK*D, swizzled
  uint16_t* p_lds = ks + KVBLK * D + wave * QBLK * PPAD;         // per wave

  // ---- load Q fragments: [2 q-tiles][DC] x 4 VGPR ----
  bf16x8_vec qf[2][DC];
#pragma unroll
  for (int qt = 0; qt < 2; ++qt) {
#pragma unroll
    for (int dc = 0; dc < DC; ++dc) {
      const int row = t0 + qt * 16 + jcol;
      uint4 raw = {0, 0, 0, 0};
      if (active && row < seq_end)
        raw = *reinterpret_cast<const uint4*>(
            qkv + int64_t(row) * qkv_stride + head * D + dc * 32 + koct * 8);
      qf[qt][dc] = *reinterpret_cast<bf16x8_vec*>(&raw);
    }
  }

  f32x4 o_acc[2][CT];
#pragma unroll
  for (int qt = 0; qt < 2; ++qt)
#pragma unroll
    for (int ct = 0; ct < CT; ++ct) o_acc[qt][ct] = f32x4{0.f, 0.f, 0.f, 0.f};
  // softmax state: this lane's 4 rows per q-tile (row = koct*4 + reg)
  float m_st[2][4], l_st[2][4];
#pragma unroll
  for (int qt = 0; qt < 2; ++qt)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      m_st[qt][r] = kNegInf;
      l_st[qt][r] = 0.f;
    }

  // causal bound: last q row of this chunk attends up to t0+QBLK-1
  const int kv_hi = min(t0 + QBLK, seq_end);

  for (int kv0 = seq_start; kv0 < kv_hi; kv0 += KVBLK) {
    // ---- cooperative V^T staging: V[kv0..kv0+32][D] -> vt[D][PPAD] ----
    __syncthreads();   // previous tile's reads done before overwrite
    {
      constexpr int PIECES = KVBLK * D / 8 / 256;
#pragma unroll
      for (int p = 0; p < PIECES; ++p) {
        const int idx = p * 256 + threadIdx.x;
        const int kv = idx / (D / 8);
        const int d8 = (idx % (D / 8)) * 8;
        const int row = kv0 + kv;
        uint4 raw = {0, 0, 0, 0};
        uint4 kraw = {0, 0, 0, 0};
        if (row < seq_end) {
          raw = *reinterpret_cast<const uint4*>(
              qkv + int64_t(row) * qkv_stride + v_off + kvh * D + d8);
          kraw = *reinterpret_cast<const uint4*>(
              qkv + int64_t(row) * qkv_stride + kv_off + kvh * D + d8);
        }
        bf16x8 piece;
        piece.u = raw;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          vt[(d8 + j) * PPAD + kv] = piece.s[j];
        // K stays row-major but XOR-swizzled so the B-fragment read
        // (16 lanes, 16 different rows, same column range) is
        // bank-conflict-free (guide T2 / G4); mask by row length so the
        // XOR never escapes the row (D=64 rows are only 128 B)
        *reinterpret_cast<uint4*>(
            reinterpret_cast<char*>(ks) +
            (kv * D * 2 + ((d8 * 2) ^ ((kv & (D / 8 - 1)) << 4)))) = kraw;
      }
    }
    __syncthreads();

    // ---- S = Q K^T over this tile (2 x KVBLK/16 16x16 tiles) ----
    constexpr int KT = KVBLK / 16;
    f32x4 s_acc[2][KT];
#pragma unroll
    for (int qt = 0; qt < 2; ++qt)
#pragma unroll
      for (int kt = 0; kt < KT; ++kt) s_acc[qt][kt] = f32x4{0.f, 0.f, 0.f, 0.f};
    if (active) {
#pragma unroll
      for (int kt = 0; kt < KT; ++kt) {
        // B fragments from the swizzled LDS K tile (staged once per
        // workgroup instead of 4x-redundant scattered L2 reads)
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

      // ---- causal + bounds mask, online softmax, P -> LDS ----
#pragma unroll
      for (int qt = 0; qt < 2; ++qt) {
        float corr[4];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row_g = t0 + qt * 16 + koct * 4 + r;
          float sv[KT];
          float mx = kNegInf;
#pragma unroll
          for (int kt = 0; kt < KT; ++kt) {
            float s = s_acc[qt][kt][r] * scale;
            const int c = kv0 + kt * 16 + jcol;
            if (c > row_g || c >= seq_end) s = kNegInf;
            sv[kt] = s;
            mx = fmaxf(mx, s);
          }
          // row max across the 16-lane group
#pragma unroll
          for (int off = 8; off > 0; off >>= 1)
            mx = fmaxf(mx, __shfl_xor(mx, off, kWave));
          const float m_new = fmaxf(m_st[qt][r], mx);
          corr[r] = __expf(m_st[qt][r] - m_new);
          m_st[qt][r] = m_new;
          float rs = 0.f;
          // store P row (bf16) to the per-wave LDS buffer
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
        // rescale O accumulators for this q-tile
#pragma unroll
        for (int ct = 0; ct < CT; ++ct)
#pragma unroll
          for (int r = 0; r < 4; ++r) o_acc[qt][ct][r] *= corr[r];
      }

      // ---- PV: A = P (from LDS, A-frag layout), B = V^T rows ----
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

  // ---- epilogue: normalize and store ----
  if (active) {
#pragma unroll
    for (int qt = 0; qt < 2; ++qt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row_g = t0 + qt * 16 + koct * 4 + r;
        if (row_g >= seq_end) continue;
        const float inv_l = l_st[qt][r] > 0.f ? 1.f / l_st[qt][r] : 0.f;
#pragma unroll
        for (int ct = 0; ct < CT; ++ct) {
          out[int64_t(row_g) * n_q_heads * D + head * D + ct * 16 + jcol] =
              f32_to_bf16(o_acc[qt][ct][r] * inv_l);
        }
      }
    }
  }
}

}  // namespace

void launch_prefill_mfma(const uint16_t* qkv, const int32_t* chunk_t0,
                         const int32_t* chunk_seq_start,
                         const int32_t* chunk_seq_end, uint16_t* out,
                         int n_chunks, int n_kv_heads, int group,
                         int head_dim, int qkv_stride, float scale,
                         hipStream_t stream) {
  const int n_hw = group > 4 ? group / 4 : 1;
  const int blocks = n_chunks * n_kv_heads * n_hw;
  const size_t smem = size_t(head_dim) * PPAD * 2 +
                      size_t(KVBLK) * head_dim * 2 +
                      size_t(4) * QBLK * PPAD * 2;
  if (head_dim == 128) {
    hipLaunchKernelGGL(prefill_mfma_kernel<128>, dim3(blocks), dim3(256),
                       smem, stream, qkv, chunk_t0, chunk_seq_start,
                       chunk_seq_end, out, n_kv_heads, group, n_hw,
                       qkv_stride, scale);
  } else if (head_dim == 64) {
    hipLaunchKernelGGL(prefill_mfma_kernel<64>, dim3(blocks), dim3(256),
                       smem, stream, qkv, chunk_t0, chunk_seq_start,
                       chunk_seq_end, out, n_kv_heads, group, n_hw,
                       qkv_stride, scale);
  }
}

}  // namespace rlli
