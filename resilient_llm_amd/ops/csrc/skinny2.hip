// Skinny-M GEMM v2 for decode projections: out[M,N] = x[M,K] @ W[N,K]^T,
// bf16 I/O, f32 accumulation, M <= 64.  Replaces skinny_gemm.hip's
// LDS-staged design, which measured 46-62% of the streaming ceiling
// (profiles/r01_gemm_streaming_analysis.md).
//
// Shape of the design (r01 + first-v2 postmortems):
//  - ZERO LDS, ZERO barriers: one wave per workgroup, owning a 64-column
//    output panel and a K-slice.  Both MFMA operands load straight from
//    memory in fragment order (mfma_f32_16x16x32_bf16: lane l holds 8
//    CONTIGUOUS k of row/col l&15 at k-octet l>>4 — guide §3):
//      B-frag = 16 B of one W row (HBM, nontemporal read-once stream),
//      A-frag = 16 B of one x row (tiny, L2-resident).
//  - 64 columns per wave, not 16: each A-fragment feeds FOUR MFMAs, so
//    the redundant x re-reads across panels stay at 1x the W stream
//    (the 16-col variant measured L2-bound: A-traffic was 4x W).
//  - Double-buffered chunk pipeline in STATICALLY indexed register
//    arrays: chunk c+1's loads issue while chunk c computes, so the
//    s_waitcnt ahead of each MFMA is a partial vmcnt (the naive loop
//    compiled to vmcnt(0) before every MFMA — zero overlap).
//  - Split-K for grid fill (N=4096 is only 64 panels): slices write f32
//    slabs, a second kernel reduces.  Slab traffic is M*N*4*splitk —
//    tiny next to W.
//
// PROLOGUE variants fuse the producer elementwise op into the A-frag
// consume path (deletes a kernel launch + an intermediate round-trip):
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

constexpr int BN = 64;          // columns per wave (4 MFMA column tiles)
constexpr int CT = BN / 16;     // column tiles

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

// One wave = one 64-column panel x one K-slice.
//  blockIdx.x = panel + n_panels * slice
// G = pipeline depth in 32-k GRANULES (one granule = CT B-frags +
// M_TILES A-frags = the operand set of 16 MFMAs).  The r02 chunk
// pipeline (2 buffers of KSTEPS granules, compute-all-then-load-all)
// measured ~6 GB/s of stream per wave: the compiler placed every
// next-chunk load after the whole MFMA block, so only one chunk was
// ever in flight.  The granule ring consumes granule p and immediately
// refills its registers with granule p+G — (G-1) granules stay in
// flight at all times — and sched_group_barrier pins the
// [16 MFMA][8 VMEM] alternation so the scheduler cannot re-bunch them.
template <int M_TILES, int PROLOGUE, int G>
__global__ __launch_bounds__(64)
void skinny2_kernel(const uint16_t* __restrict__ x,
                    const uint16_t* __restrict__ w,
                    float* __restrict__ out_ws,       // [splitk][M][N] f32
                    uint16_t* __restrict__ out_bf16,  // [M][N] when splitk==1
                    int M, int N, int K,
                    int k_per_slice, int splitk) {
  const int n_panels = N / BN;
  const int panel = blockIdx.x % n_panels;
  const int slice = blockIdx.x / n_panels;
  const int kbeg = slice * k_per_slice;
  const int kend = min(kbeg + k_per_slice, K);
  if (kbeg >= kend) return;

  const int lane = threadIdx.x;
  const int jcol = lane & 15;   // B column / A row (within a 16-tile)
  const int koct = lane >> 4;   // k-octet: 8 contiguous bf16

  // x row stride in bf16 elements (SILU reads gu[M, 2K])
  const int xstride = PROLOGUE == PRO_SILU ? 2 * K : K;
  // Both operands load through buffer descriptors (T8/T20): the k part
  // of every address is WAVE-UNIFORM and rides the SGPR soffset, so a
  // load costs zero VALU and zero address VGPRs — the per-lane voffset
  // of each fragment stream is a loop-invariant register.  x rows >= M
  // are hardware-bounds-checked to 0 (padded M tiles add nothing).
  const auto xrsrc = __builtin_amdgcn_make_buffer_rsrc(
      const_cast<uint16_t*>(x), /*stride*/ (short)0,
      /*bytes*/ M * xstride * 2, /*flags*/ 0x00020000);
  const auto wrsrc = __builtin_amdgcn_make_buffer_rsrc(
      const_cast<uint16_t*>(w), /*stride*/ (short)0,
      /*bytes*/ int(int64_t(N) * K * 2), /*flags*/ 0x00020000);

  int voff_b[CT];
#pragma unroll
  for (int ct = 0; ct < CT; ++ct)
    voff_b[ct] = ((panel * BN + ct * 16 + jcol) * K + koct * 8) * 2;
  int voff_a[M_TILES];
#pragma unroll
  for (int mt = 0; mt < M_TILES; ++mt)
    voff_a[mt] = ((mt * 16 + jcol) * xstride + koct * 8) * 2;

  f32x4 acc[M_TILES][CT];
#pragma unroll
  for (int mt = 0; mt < M_TILES; ++mt)
#pragma unroll
    for (int ct = 0; ct < CT; ++ct) acc[mt][ct] = f32x4{0.f, 0.f, 0.f, 0.f};

  constexpr int NA = PROLOGUE == PRO_SILU ? 2 : 1;   // regs per A frag
  constexpr int AUX_NT = 2;                          // nontemporal policy

  u32x4 bbuf[G][CT];
  u32x4 abuf[G][M_TILES][NA];

  auto load_gran = [&](int p, int kk) {
    const int soff = kk * 2;                 // SGPR: kk is wave-uniform
#pragma unroll
    for (int ct = 0; ct < CT; ++ct)
      bbuf[p][ct] = __builtin_amdgcn_raw_buffer_load_b128(
          wrsrc, voff_b[ct], soff, AUX_NT);
#pragma unroll
    for (int mt = 0; mt < M_TILES; ++mt) {
      abuf[p][mt][0] = __builtin_amdgcn_raw_buffer_load_b128(
          xrsrc, voff_a[mt], soff, 0);
      if (PROLOGUE == PRO_SILU)
        abuf[p][mt][1] = __builtin_amdgcn_raw_buffer_load_b128(
            xrsrc, voff_a[mt], soff + K * 2, 0);
    }
  };
  auto mfma_gran = [&](int p) {
#pragma unroll
    for (int mt = 0; mt < M_TILES; ++mt) {
      u32x4 araw = abuf[p][mt][0];
      if (PROLOGUE == PRO_SILU)
        araw = silu_mul_frag(araw, abuf[p][mt][1]);
      const bf16x8_vec afrag = as_bf16x8(araw);
#pragma unroll
      for (int ct = 0; ct < CT; ++ct)
        acc[mt][ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag, as_bf16x8(bbuf[p][ct]), acc[mt][ct], 0, 0, 0);
    }
  };

  const int n_g = (kend - kbeg) / 32;        // granules in this K-slice
  // prologue: fill the ring
#pragma unroll
  for (int p = 0; p < G; ++p)
    if (p < n_g) load_gran(p, kbeg + p * 32);
  // steady: consume granule g (ring slot p), refill slot p with g+G.
  // The refill MUST follow the MFMAs (same registers), but only ONE
  // granule's MFMAs — the other G-1 granules' loads stay in flight, so
  // the wait ahead of each MFMA group is a counted vmcnt, not a drain.
  int g = 0;
  for (; g + G < n_g; g += G) {
#pragma unroll
    for (int p = 0; p < G; ++p) {
      mfma_gran(p);
      if (g + p + G < n_g) load_gran(p, kbeg + (g + p + G) * 32);
      // pin the alternation: one granule of MFMAs, then its refill
      // loads — the scheduler may not re-bunch loads behind compute
      __builtin_amdgcn_sched_group_barrier(0x008, M_TILES * CT, 0);
      __builtin_amdgcn_sched_group_barrier(0x020, CT + M_TILES * NA, 0);
    }
  }
  // epilogue: drain the ring
#pragma unroll
  for (int p = 0; p < G; ++p)
    if (g + p < n_g) mfma_gran(p);

  // C layout (guide §3): col = lane&15, row = (lane>>4)*4 + reg
#pragma unroll
  for (int mt = 0; mt < M_TILES; ++mt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int out_row = mt * 16 + koct * 4 + r;
      if (out_row >= M) continue;
#pragma unroll
      for (int ct = 0; ct < CT; ++ct) {
        const int out_col = panel * BN + ct * 16 + jcol;
        if (splitk == 1) {
          out_bf16[int64_t(out_row) * N + out_col] =
              f32_to_bf16(acc[mt][ct][r]);
        } else {
          out_ws[(int64_t(slice) * M + out_row) * N + out_col] =
              acc[mt][ct][r];
        }
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
  // rounding can leave trailing slices with no K range: drop them, or
  // the reduce would sum their UNINITIALIZED slabs
  splitk = (K + k_per_slice - 1) / k_per_slice;
  const int blocks = (N / BN) * splitk;
  // G: granule-ring depth = granules kept in flight.  Deep rings buy
  // per-wave streaming rate (in-flight bytes/latency) until the VGPR
  // budget forces spills or 1-wave occupancy; big fragment sets
  // (M_TILES > 2, SILU's 2x A regs) take a shallower ring.
  // RLLI_SK2_G=2..8 overrides for tuning.
  static const char* g_env = std::getenv("RLLI_SK2_G");
  const int g_req = g_env ? atoi(g_env) : 0;
  auto launch = [&](auto mt_tag) {
    constexpr int MT = decltype(mt_tag)::value;
    const int g = g_req ? g_req
                        : (PROLOGUE == PRO_SILU ? 3 : (MT > 2 ? 4 : 6));
    auto go = [&](auto g_tag) {
      hipLaunchKernelGGL((skinny2_kernel<MT, PROLOGUE,
                                         decltype(g_tag)::value>),
                         dim3(blocks), dim3(64), 0, stream, x, w, ws, out,
                         M, N, K, k_per_slice, splitk);
    };
    if (g <= 2) go(std::integral_constant<int, 2>{});
    else if (g == 3) go(std::integral_constant<int, 3>{});
    else if (g == 4) go(std::integral_constant<int, 4>{});
    else if (g <= 6) go(std::integral_constant<int, 6>{});
    else go(std::integral_constant<int, 8>{});
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
