// Fused SwiGLU activation: out[t, i] = silu(gu[t, i]) * gu[t, I+i].
//
// The gate|up halves come packed from one hipBLASLt GEMM (Llama MLP
// up-projection), so the activation is ONE elementwise pass instead of
// three (split, silu, mul) — memory-bound, bf16x8-vectorized (G13),
// grid-stride with the block cap of guide Guideline 11.

#include "common.h"

#include <type_traits>

namespace rlli {

namespace {

// 8 loads (4 gate + 4 up vectors) in flight per lane before any math:
// with a single outstanding load per lane this op measured 0.9 TB/s
// (latency-bound); batch-issuing restores streaming rate (guide G7/G15:
// waits belong at the first consumer).
constexpr int ILP = 4;

DEV_INLINE void silu_mul_one(uint4 graw, uint4 uraw, uint16_t* dst) {
  bf16x8 g, u, o;
  g.u = graw;
  u.u = uraw;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    const float gf = bf16_to_f32(g.s[i]);
    const float uf = bf16_to_f32(u.s[i]);
    const float s = gf / (1.f + __expf(-gf));
    o.s[i] = f32_to_bf16(s * uf);
  }
  *reinterpret_cast<uint4*>(dst) = o.u;
}

// 2D grid: blockIdx.y = row, blockIdx.x strides the row's vectors.
// The earlier flat-index version divided a 64-bit linear index by the
// runtime row width per vector — two software int64 divides per 16-B
// load (~8.7 us/call at decode batch 64, 0.6 TB/s, pure VALU stall).
// Row/col from the grid costs zero VALU.
__global__ __launch_bounds__(64)
void silu_mul_kernel(const uint16_t* __restrict__ gate_up,
                     uint16_t* __restrict__ out,
                     int64_t inter) {
  const int64_t nvec_row = inter / 8;
  const int64_t row = blockIdx.y;
  const uint16_t* g_row = gate_up + row * 2 * inter;
  uint16_t* o_row = out + row * inter;
  const int64_t stride = int64_t(gridDim.x) * blockDim.x;
  int64_t v = blockIdx.x * int64_t(blockDim.x) + threadIdx.x;
  for (; v + (ILP - 1) * stride < nvec_row; v += ILP * stride) {
    uint4 gr[ILP], ur[ILP];
#pragma unroll
    for (int j = 0; j < ILP; ++j) {
      const int64_t col = (v + j * stride) * 8;
      gr[j] = *reinterpret_cast<const uint4*>(g_row + col);
      ur[j] = *reinterpret_cast<const uint4*>(g_row + col + inter);
    }
#pragma unroll
    for (int j = 0; j < ILP; ++j)
      silu_mul_one(gr[j], ur[j], o_row + (v + j * stride) * 8);
  }
  for (; v < nvec_row; v += stride) {
    silu_mul_one(*reinterpret_cast<const uint4*>(g_row + v * 8),
                 *reinterpret_cast<const uint4*>(g_row + v * 8 + inter),
                 o_row + v * 8);
  }
}

// FP8-output variant (quantized-weight down-proj): one WG per row so
// the row amax is a single block reduction; values are kept in
// registers between the amax pass and the quantized write (VPL deep
// per-lane ILP covers the row: 14336/8 = 1792 vecs / 512 lanes -> 4).
template <int VPL>
__global__ __launch_bounds__(512)
void silu_mul_fp8_kernel(const uint16_t* __restrict__ gate_up,
                         uint8_t* __restrict__ out8,
                         float* __restrict__ scales, int64_t inter) {
  __shared__ float lds[8];
  const int64_t row = blockIdx.x;
  const int64_t nvec = inter / 8;
  const uint16_t* g_row = gate_up + row * 2 * inter;
  float vals[VPL][8];
  float amax = 0.f;
#pragma unroll
  for (int p = 0; p < VPL; ++p) {
    const int64_t v = threadIdx.x + int64_t(p) * blockDim.x;
    if (v >= nvec) {
#pragma unroll
      for (int i = 0; i < 8; ++i) vals[p][i] = 0.f;
      continue;
    }
    bf16x8 g, u;
    g.u = *reinterpret_cast<const uint4*>(g_row + v * 8);
    u.u = *reinterpret_cast<const uint4*>(g_row + v * 8 + inter);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const float gf = bf16_to_f32(g.s[i]);
      const float uf = bf16_to_f32(u.s[i]);
      vals[p][i] = gf / (1.f + __expf(-gf)) * uf;
      amax = fmaxf(amax, fabsf(vals[p][i]));
    }
  }
  const int lane = threadIdx.x & (kWave - 1);
  const int wave = threadIdx.x / kWave;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    amax = fmaxf(amax, __shfl_xor(amax, off, kWave));
  if (lane == 0) lds[wave] = amax;
  __syncthreads();
  const int n_waves = blockDim.x / kWave;
  amax = 0.f;
#pragma unroll
  for (int w = 0; w < 8; ++w)
    if (w < n_waves) amax = fmaxf(amax, lds[w]);
  const float scale = fmaxf(amax, 1e-12f) / 448.f;
  const float rs = 1.f / scale;
#pragma unroll
  for (int p = 0; p < VPL; ++p) {
    const int64_t v = threadIdx.x + int64_t(p) * blockDim.x;
    if (v >= nvec) continue;
    float q[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) q[i] = vals[p][i] * rs;
    *reinterpret_cast<CacheFP8::vec8*>(out8 + row * inter + v * 8) =
        CacheFP8::from_f32(q);
  }
  if (threadIdx.x == 0) scales[row] = scale;
}

}  // namespace

void launch_silu_mul_fp8(const uint16_t* gate_up, uint8_t* out8,
                         float* scales, int rows, int inter,
                         hipStream_t stream) {
  if (rows == 0) return;
  const int threads = 512;
  const int vpl = (inter / 8 + threads - 1) / threads;
  auto go = [&](auto tag) {
    hipLaunchKernelGGL((silu_mul_fp8_kernel<decltype(tag)::value>),
                       dim3(rows), dim3(threads), 0, stream, gate_up, out8,
                       scales, inter);
  };
  if (vpl <= 1) go(std::integral_constant<int, 1>{});
  else if (vpl <= 2) go(std::integral_constant<int, 2>{});
  else if (vpl <= 4) go(std::integral_constant<int, 4>{});
  else if (vpl <= 8) go(std::integral_constant<int, 8>{});
  else go(std::integral_constant<int, 16>{});
}

void launch_silu_mul(const uint16_t* gate_up, uint16_t* out, int rows,
                     int inter, hipStream_t stream) {
  if (rows == 0) return;
  const int64_t nvec_row = inter / 8;
  const int threads = 64;   // wave-granular: many WGs, deep per-lane ILP
  // enough row-chunks that rows*chunks >= ~2048 WGs fill the chip at
  // decode batch sizes; each WG covers ILP vectors per lane per pass
  int chunks = int(std::min<int64_t>(
      (nvec_row + threads * ILP - 1) / (threads * ILP),
      std::max<int64_t>(1, 2048 / rows)));
  hipLaunchKernelGGL(silu_mul_kernel, dim3(chunks, rows), dim3(threads), 0,
                     stream, gate_up, out, inter);
}

}  // namespace rlli
