// Fused SwiGLU activation: out[t, i] = silu(gu[t, i]) * gu[t, I+i].
//
// The gate|up halves come packed from one hipBLASLt GEMM (Llama MLP
// up-projection), so the activation is ONE elementwise pass instead of
// three (split, silu, mul) — memory-bound, bf16x8-vectorized (G13),
// grid-stride with the block cap of guide Guideline 11.

#include "common.h"

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

__global__ __launch_bounds__(64)
void silu_mul_kernel(const uint16_t* __restrict__ gate_up,
                     uint16_t* __restrict__ out,
                     int64_t rows, int64_t inter) {
  const int64_t nvec = rows * (inter / 8);
  const int64_t stride = int64_t(gridDim.x) * blockDim.x;
  int64_t v = blockIdx.x * int64_t(blockDim.x) + threadIdx.x;
  for (; v + (ILP - 1) * stride < nvec; v += ILP * stride) {
    uint4 gr[ILP], ur[ILP];
    int64_t base[ILP], col8[ILP], rw[ILP];
#pragma unroll
    for (int j = 0; j < ILP; ++j) {
      const int64_t vv = v + j * stride;
      rw[j] = vv / (inter / 8);
      col8[j] = (vv % (inter / 8)) * 8;
      base[j] = rw[j] * 2 * inter + col8[j];
      gr[j] = *reinterpret_cast<const uint4*>(gate_up + base[j]);
      ur[j] = *reinterpret_cast<const uint4*>(gate_up + base[j] + inter);
    }
#pragma unroll
    for (int j = 0; j < ILP; ++j)
      silu_mul_one(gr[j], ur[j], out + rw[j] * inter + col8[j]);
  }
  for (; v < nvec; v += stride) {
    const int64_t row = v / (inter / 8);
    const int64_t col8 = (v % (inter / 8)) * 8;
    const int64_t base = row * 2 * inter + col8;
    silu_mul_one(*reinterpret_cast<const uint4*>(gate_up + base),
                 *reinterpret_cast<const uint4*>(gate_up + base + inter),
                 out + row * inter + col8);
  }
}

}  // namespace

void launch_silu_mul(const uint16_t* gate_up, uint16_t* out, int rows,
                     int inter, hipStream_t stream) {
  if (rows == 0) return;
  const int64_t nvec = int64_t(rows) * (inter / 8);
  const int threads = 64;   // wave-granular: many WGs, deep per-lane ILP
  const int blocks = int(std::min<int64_t>(
      (nvec + threads * 4 - 1) / (threads * 4), 4096));
  hipLaunchKernelGGL(silu_mul_kernel, dim3(blocks), dim3(threads), 0, stream,
                     gate_up, out, rows, inter);
}

}  // namespace rlli
