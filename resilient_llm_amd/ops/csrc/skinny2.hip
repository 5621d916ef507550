// Skinny-M GEMM v2 for decode projections: out[M,N] = x[M,K] @ W[N,K]^T,
// bf16 I/O, f32 accumulation, M <= 64.  Replaces skinny_gemm.hip's
// LDS-staged design, which measured 46-62% of the streaming ceiling
// (profiles/r01_gemm_streaming_analysis.md).
//
// Why v2 is shaped this way (the r01 postmortem):
//  - r01 staged x in LDS with a __syncthreads per K-chunk: four waves in
//    lockstep, W-stream stalls at every barrier.  v2 has ZERO LDS and
//    ZERO barriers — each workgroup is ONE wave owning 16 output columns.
//  - Both MFMA operands load straight from memory in fragment order
//    (mfma_f32_16x16x32_bf16: lane l holds 8 CONTIGUOUS k of row/col
//    l&15 at k-octet l>>4, guide §3 Fragment layout):
//      B-frag = 16 B of one W row (HBM, nontemporal: read-once stream),
//      A-frag = 16 B of one x row (L2-hot: all N/16 groups re-read the
//               same tiny x, so HBM sees it ~once per XCD).
//    No transpose, no staging, no cross-lane traffic: the only
//    instructions in the hot loop are loads and MFMAs, so the W stream
//    can stay as deep in flight as the probe's (streamprobe.hip).
//  - Split-K for grid fill on small N (one wave per 16 columns gives
//    N=4096 only 256 groups): slices write f32 slabs, a second kernel
//    reduces.  Slab traffic is M*N*4*splitk — tiny next to W.
//
// PROLOGUE variants fuse the producer elementwise op into the A-frag
// load path (the activation is tiny; the fusion deletes a whole kernel
// launch + an intermediate tensor round-trip):
//   PLAIN:    a = x[m][k]
//   SILU_MUL: a = silu(g[m][k]) * u[m][k],  x = gu[M, 2K] (down-proj)

#include "common.h"

#include <cstdlib>

namespace rlli {

namespace {

using bf16x8_vec = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;
using u32x4 = __attribute__((ext_vector_type(4))) unsigned int;

enum { PRO_PLAIN = 0, PRO_SILU = 1 };

DEV_INLINE bf16x8_vec as_bf16x8(u32x4 v) {
  return *reinterpret_cast<bf16x8_vec*>(&v);
}

// silu(g)*u on an 8-element fragment pair, f32 math, RNE back to bf16.
DEV_INLINE u32x4 silu_mul_frag(u32x4 graw, u32x4 uraw) {
  const uint16_t* g = reinterpret_cast<const uint16_t*>(&graw);
  const uint16_t* u = reinterpret_cast<const uint16_t*>(&uraw);
  u32x4 out;
  uint16_t* o = reinterpret_cast<uint16_t*>(&out);
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    const float gf = bf16_to_f32(g[i]);
    const float uf = bf16_to_f32(u[i]);
    const float s = gf / (1.f + __expf(-gf));
    o[i] = f32_to_bf16(s * uf);
  }
  return out;
}

// One wave = one 16-column output group x one K-slice.
//  blockIdx.x = group + n_groups * slice
// KSTEP canonical chunk: 8 mfma k-steps = 256 k per iteration, fully
// unrolled so LLVM hoists the whole chunk's loads ahead of its MFMAs
// (the deep-prefetch ring, without hand-rolled buffers).
template <int M_TILES, int PROLOGUE>
__global__ __launch_bounds__(64)
void skinny2_kernel(const uint16_t* __restrict__ x,
                    const uint16_t* __restrict__ w,
                    float* __restrict__ out_ws,       // [splitk][M][N] f32
                    uint16_t* __restrict__ out_bf16,  // [M][N] when splitk==1
                    int M, int N, int K,
                    int k_per_slice, int splitk) {
  const int n_groups = N / 16;
  const int group = blockIdx.x % n_groups;
  const int slice = blockIdx.x / n_groups;
  const int kbeg = slice * k_per_slice;
  const int kend = min(kbeg + k_per_slice, K);
  if (kbeg >= kend) return;

  const int lane = threadIdx.x;
  const int jcol = lane & 15;   // B column / A row (within tile)
  const int koct = lane >> 4;   // k-octet: 8 contiguous bf16

  // x row stride in bf16 elements (SILU reads gu[M, 2K])
  const int xstride = PROLOGUE == PRO_SILU ? 2 * K : K;
  // hardware-bounds-checked x loads: rows >= M return 0 (padded M tiles
  // contribute nothing) — guide T8/T20: descriptor from wave-uniform ptr
  const auto xrsrc = __builtin_amdgcn_make_buffer_rsrc(
      const_cast<uint16_t*>(x), /*stride*/ (short)0,
      /*bytes*/ M * xstride * 2, /*flags*/ 0x00020000);

  const uint16_t* wrow = w + int64_t(group * 16 + jcol) * K + koct * 8;

  f32x4 acc[M_TILES];
#pragma unroll
  for (int mt = 0; mt < M_TILES; ++mt) acc[mt] = f32x4{0.f, 0.f, 0.f, 0.f};

  // byte offset of this lane's A fragment for row tile mt at k
  auto a_off = [&](int mt, int k) {
    return ((mt * 16 + jcol) * xstride + k + koct * 8) * 2;
  };

  // Double-buffered chunk pipeline with STATICALLY indexed register
  // arrays (they must stay in VGPRs): load chunk c+1 while computing
  // chunk c, so every MFMA's operands were issued a full chunk earlier
  // and the s_waitcnt before it is a PARTIAL vmcnt (the naive loop
  // compiled to vmcnt(0) before every MFMA — zero overlap).
  constexpr int KSTEPS = PROLOGUE == PRO_SILU ? 2 : 4;  // x 32 k per chunk
  constexpr int CHUNK = 32 * KSTEPS;
  constexpr int NA = PROLOGUE == PRO_SILU ? 2 : 1;      // regs per A frag

  u32x4 bbuf[2][KSTEPS];
  u32x4 abuf[2][KSTEPS][M_TILES][NA];

  auto load_chunk = [&](int buf, int k) {
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
      const int kk = k + ks * 32;
      bbuf[buf][ks] = __builtin_nontemporal_load(
          reinterpret_cast<const u32x4*>(wrow + kk));
#pragma unroll
      for (int mt = 0; mt < M_TILES; ++mt) {
        abuf[buf][ks][mt][0] = __builtin_amdgcn_raw_buffer_load_b128(
            xrsrc, a_off(mt, kk), 0, 0);
        if (PROLOGUE == PRO_SILU)
          abuf[buf][ks][mt][1] = __builtin_amdgcn_raw_buffer_load_b128(
              xrsrc, a_off(mt, kk) + K * 2, 0, 0);
      }
    }
  };
  auto compute_chunk = [&](int buf) {
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
      const bf16x8_vec bfrag = as_bf16x8(bbuf[buf][ks]);
#pragma unroll
      for (int mt = 0; mt < M_TILES; ++mt) {
        u32x4 araw = abuf[buf][ks][mt][0];
        if (PROLOGUE == PRO_SILU)
          araw = silu_mul_frag(araw, abuf[buf][ks][mt][1]);
        acc[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            as_bf16x8(araw), bfrag, acc[mt], 0, 0, 0);
      }
    }
  };

  // k_per_slice and K are multiples of 256, so every slice has an even
  // number of 128-k chunks (and of 64-k chunks for SILU) — the 2x
  // unrolled steady-state below never needs a tail.
  load_chunk(0, kbeg);
  if (kbeg + CHUNK < kend) load_chunk(1, kbeg + CHUNK);
  int k = kbeg;
  for (; k + 2 * CHUNK < kend; k += 2 * CHUNK) {
    compute_chunk(0);
    load_chunk(0, k + 2 * CHUNK);
    compute_chunk(1);
    if (k + 3 * CHUNK < kend) load_chunk(1, k + 3 * CHUNK);
  }
  compute_chunk(0);
  if (k + CHUNK < kend) compute_chunk(1);

  // C layout (guide §3): col = lane&15, row = (lane>>4)*4 + reg
  const int out_col = group * 16 + jcol;
#pragma unroll
  for (int mt = 0; mt < M_TILES; ++mt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int out_row = mt * 16 + koct * 4 + r;
      if (out_row >= M) continue;
      if (splitk == 1) {
        out_bf16[int64_t(out_row) * N + out_col] = f32_to_bf16(acc[mt][r]);
      } else {
        out_ws[(int64_t(slice) * M + out_row) * N + out_col] = acc[mt][r];
      }
    }
  }
}

__global__ void skinny2_reduce_kernel(const float* __restrict__ ws,
                                      uint16_t* __restrict__ out,
                                      int64_t mn, int splitk) {
  const int64_t stride = int64_t(gridDim.x) * blockDim.x;
  for (int64_t i = blockIdx.x * int64_t(blockDim.x) + threadIdx.x; i < mn;
       i += stride) {
    float acc = 0.f;
    for (int s = 0; s < splitk; ++s) acc += ws[int64_t(s) * mn + i];
    out[i] = f32_to_bf16(acc);
  }
}

template <int PROLOGUE>
void launch_any(const uint16_t* x, const uint16_t* w, float* ws,
                uint16_t* out, int M, int N, int K, int splitk,
                hipStream_t stream) {
  const int m_tiles = (M + 15) / 16;
  const int k_per_slice = ((K / 256 + splitk - 1) / splitk) * 256;
  const int blocks = (N / 16) * splitk;
  auto launch = [&](auto mt_tag) {
    constexpr int MT = decltype(mt_tag)::value;
    hipLaunchKernelGGL((skinny2_kernel<MT, PROLOGUE>), dim3(blocks), dim3(64),
                       0, stream, x, w, ws, out, M, N, K, k_per_slice, splitk);
  };
  using T1 = std::integral_constant<int, 1>;
  using T2 = std::integral_constant<int, 2>;
  using T3 = std::integral_constant<int, 3>;
  using T4 = std::integral_constant<int, 4>;
  if (m_tiles == 1) launch(T1{});
  else if (m_tiles == 2) launch(T2{});
  else if (m_tiles == 3) launch(T3{});
  else launch(T4{});
  if (splitk > 1) {
    const int64_t mn = int64_t(M) * N;
    const int threads = 256;
    const int rblocks = int(std::min<int64_t>((mn / 4 + threads - 1) / threads,
                                              2048));
    hipLaunchKernelGGL(skinny2_reduce_kernel, dim3(rblocks), dim3(threads), 0,
                       stream, ws, out, mn, splitk);
  }
}

}  // namespace

void launch_skinny2(const uint16_t* x, const uint16_t* w, float* ws,
                    uint16_t* out, int M, int N, int K, int splitk,
                    hipStream_t stream) {
  launch_any<PRO_PLAIN>(x, w, ws, out, M, N, K, splitk, stream);
}

void launch_skinny2_silu(const uint16_t* gu, const uint16_t* w, float* ws,
                         uint16_t* out, int M, int N, int K, int splitk,
                         hipStream_t stream) {
  launch_any<PRO_SILU>(gu, w, ws, out, M, N, K, splitk, stream);
}

}  // namespace rlli
