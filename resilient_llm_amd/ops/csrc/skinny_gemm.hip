// Skinny-M GEMM for decode projections: out[M,N] = x[M,K] @ W[N,K]^T,
// bf16 I/O, f32 accumulation, M <= 64.
//
// Decode GEMMs are pure weight streaming (W is read once, x is tiny), so
// the kernel is built around that: each workgroup owns a BN=64 column
// panel and a K-slice, streams W with nontemporal 16-B loads (nt: one CU
// reads each weight byte exactly once — guide nt-weights), stages the x
// panel in LDS (XOR-swizzled rows, conflict-free ds_read_b128 — guide T2)
// and rides mfma_f32_16x16x32_bf16 with the B fragment shared by all
// M-tiles.  Split-K fills the 256 CUs for small N (N=4096 alone is only
// 64 panels): each slice writes a private f32 slab, a tiny second kernel
// reduces slabs and casts to bf16.
//
// B-fragment trick: W rows are K-contiguous, and the 16x16x32 B layout
// wants lane l = (j = l&15, k-octet l>>4) — i.e. each lane reads 16
// CONTIGUOUS bf16 of one W row; lanes l, l+16, l+32, l+48 cover one row's
// 64-byte k-chunk, so the stream is 64-B-granular without any transpose.

#include "common.h"

#include <cstdlib>

namespace rlli {

namespace {

using bf16x8_vec = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int BN = 64;          // column panel per workgroup

DEV_INLINE int swz(int row, int byte_off) {
  return byte_off ^ ((row & 15) << 4);
}

template <int M_TILES, int KC, int WDEPTH>
__global__ __launch_bounds__(256)
void skinny_gemm_kernel(const uint16_t* __restrict__ x,
                        const uint16_t* __restrict__ w,
                        float* __restrict__ out_ws,      // [splitk][M][N] f32
                        uint16_t* __restrict__ out_bf16, // [M][N] when splitk==1
                        int M, int N, int K,
                        int chunks_per_slice, int splitk) {
  constexpr int MROWS = M_TILES * 16;
  __shared__ __attribute__((aligned(16))) uint16_t xs[2 * MROWS * KC];

  const int n_panels = N / BN;
  const int panel = blockIdx.x % n_panels;
  const int slice = blockIdx.x / n_panels;
  const int chunk0 = slice * chunks_per_slice;
  const int total_chunks = K / KC;
  const int chunk1 = min(chunk0 + chunks_per_slice, total_chunks);
  if (chunk0 >= chunk1) return;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int n0 = panel * BN + wave * 16;       // this wave's 16 columns
  const int jcol = lane & 15;                  // B fragment column
  const int koct = lane >> 4;                  // B fragment k-octet (8 bf16)

  f32x4 acc[M_TILES];
#pragma unroll
  for (int mt = 0; mt < M_TILES; ++mt) acc[mt] = f32x4{0.f, 0.f, 0.f, 0.f};

  using u32x4 = __attribute__((ext_vector_type(4))) unsigned int;
  constexpr int KSTEPS = KC / 32;
  // W row base for this lane's B fragments
  const uint16_t* wrow = w + int64_t(n0 + jcol) * K + koct * 8;

  // ---- prologue: stage x chunk0 ----
  constexpr int PIECES = MROWS * KC / 8 / 256;
  auto stage_x = [&](int c) {
    const int kbase = c * KC;
#pragma unroll
    for (int p = 0; p < PIECES; ++p) {
      const int idx = p * 256 + threadIdx.x;   // which bf16x8 of the tile
      const int row = idx / (KC / 8);
      const int col8 = idx % (KC / 8);
      uint4 piece = {0, 0, 0, 0};
      if (row < M)
        piece = *reinterpret_cast<const uint4*>(
            x + int64_t(row) * K + kbase + col8 * 8);
      *reinterpret_cast<uint4*>(
          reinterpret_cast<char*>(xs) +
          ((c & 1) ? MROWS * KC * 2 : 0) +
          swz(row, (row * KC + col8 * 8) * 2)) = piece;
    }
  };
  u32x4 brawA[KSTEPS], brawB[KSTEPS];
  auto load_b = [&](int c, u32x4 (&dst)[KSTEPS]) {
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks)
      dst[ks] = __builtin_nontemporal_load(
          reinterpret_cast<const u32x4*>(wrow + c * KC + ks * 32));
  };
  auto compute = [&](int c, u32x4 (&bsrc)[KSTEPS]) {
    const char* xbase = reinterpret_cast<const char*>(xs) +
                        ((c & 1) ? MROWS * KC * 2 : 0);
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
      const int k0 = ks * 32;
      bf16x8_vec bfrag = *reinterpret_cast<bf16x8_vec*>(&bsrc[ks]);
#pragma unroll
      for (int mt = 0; mt < M_TILES; ++mt) {
        const int row = mt * 16 + jcol;
        uint4 araw = *reinterpret_cast<const uint4*>(
            xbase + swz(row, (row * KC + k0 + koct * 8) * 2));
        bf16x8_vec afrag = *reinterpret_cast<bf16x8_vec*>(&araw);
        acc[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag, bfrag, acc[mt], 0, 0, 0);
      }
    }
  };

  stage_x(chunk0);
  load_b(chunk0, brawA);
  if (WDEPTH == 2 && chunk0 + 1 < chunk1) load_b(chunk0 + 1, brawB);
  __syncthreads();

  // A/B-alternating pipeline (statically indexed buffers, rule 20);
  // WDEPTH=2 keeps TWO chunks of W-stream in flight per wave
  for (int c = chunk0; c < chunk1; ++c) {
    const bool useA = ((c - chunk0) & 1) == 0;
    if (c + 1 < chunk1) stage_x(c + 1);
    if (WDEPTH == 1) {
      if (useA) {
        if (c + 1 < chunk1) load_b(c + 1, brawB);
        compute(c, brawA);
      } else {
        if (c + 1 < chunk1) load_b(c + 1, brawA);
        compute(c, brawB);
      }
    } else {
      if (useA) {
        compute(c, brawA);
        if (c + 2 < chunk1) load_b(c + 2, brawA);
      } else {
        compute(c, brawB);
        if (c + 2 < chunk1) load_b(c + 2, brawB);
      }
    }
    __syncthreads();
  }

  // ---- epilogue: C layout col = lane&15, row = (lane>>4)*4 + reg ----
  const int out_col = n0 + jcol;
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

__global__ void splitk_reduce_kernel(const float* __restrict__ ws,
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

}  // namespace

void launch_skinny_gemm(const uint16_t* x, const uint16_t* w, float* ws,
                        uint16_t* out, int M, int N, int K, int splitk,
                        hipStream_t stream) {
  const int m_tiles = (M + 15) / 16;
  const int KC = (std::getenv("RLLI_SKINNY_KC") &&
                  atoi(std::getenv("RLLI_SKINNY_KC")) == 128) ? 128 : 256;
  const int wdepth = (std::getenv("RLLI_SKINNY_WDEPTH") &&
                      atoi(std::getenv("RLLI_SKINNY_WDEPTH")) == 1) ? 1 : 2;
  const int total_chunks = K / KC;
  const int chunks_per_slice = (total_chunks + splitk - 1) / splitk;
  const int blocks = (N / BN) * splitk;
  auto launch = [&](auto mt_tag) {
    constexpr int MT = decltype(mt_tag)::value;
    if (KC == 128 && wdepth == 2)
      hipLaunchKernelGGL((skinny_gemm_kernel<MT, 128, 2>), dim3(blocks),
                         dim3(256), 0, stream, x, w, ws, out, M, N, K,
                         chunks_per_slice, splitk);
    else if (KC == 128)
      hipLaunchKernelGGL((skinny_gemm_kernel<MT, 128, 1>), dim3(blocks),
                         dim3(256), 0, stream, x, w, ws, out, M, N, K,
                         chunks_per_slice, splitk);
    else if (wdepth == 2)
      hipLaunchKernelGGL((skinny_gemm_kernel<MT, 256, 2>), dim3(blocks),
                         dim3(256), 0, stream, x, w, ws, out, M, N, K,
                         chunks_per_slice, splitk);
    else
      hipLaunchKernelGGL((skinny_gemm_kernel<MT, 256, 1>), dim3(blocks),
                         dim3(256), 0, stream, x, w, ws, out, M, N, K,
                         chunks_per_slice, splitk);
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
    hipLaunchKernelGGL(splitk_reduce_kernel, dim3(rblocks), dim3(threads), 0,
                       stream, ws, out, mn, splitk);
  }
}

}  // namespace rlli
