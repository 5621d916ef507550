// Fused SwiGLU activation: out[t, i] = silu(gu[t, i]) * gu[t, I+i].
//
// The gate|up halves come packed from one hipBLASLt GEMM (Llama MLP
// up-projection), so the activation is ONE elementwise pass instead of
// three (split, silu, mul) — memory-bound, bf16x8-vectorized (G13),
// grid-stride with the block cap of guide Guideline 11.

#include "common.h"

namespace rlli {

namespace {

__global__ void silu_mul_kernel(const uint16_t* __restrict__ gate_up,
                                uint16_t* __restrict__ out,
                                int64_t rows, int64_t inter) {
  const int64_t nvec = rows * (inter / 8);
  const int64_t stride = int64_t(gridDim.x) * blockDim.x;
  for (int64_t v = blockIdx.x * int64_t(blockDim.x) + threadIdx.x; v < nvec;
       v += stride) {
    const int64_t row = v / (inter / 8);
    const int64_t col8 = (v % (inter / 8)) * 8;
    const int64_t base = row * 2 * inter + col8;
    bf16x8 g, u, o;
    g.u = *reinterpret_cast<const uint4*>(gate_up + base);
    u.u = *reinterpret_cast<const uint4*>(gate_up + base + inter);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const float gf = bf16_to_f32(g.s[i]);
      const float uf = bf16_to_f32(u.s[i]);
      const float s = gf / (1.f + __expf(-gf));
      o.s[i] = f32_to_bf16(s * uf);
    }
    *reinterpret_cast<uint4*>(out + row * inter + col8) = o.u;
  }
}

}  // namespace

void launch_silu_mul(const uint16_t* gate_up, uint16_t* out, int rows,
                     int inter, hipStream_t stream) {
  if (rows == 0) return;
  const int64_t nvec = int64_t(rows) * (inter / 8);
  const int threads = 256;
  const int blocks = int(std::min<int64_t>((nvec + threads - 1) / threads, 2048));
  hipLaunchKernelGGL(silu_mul_kernel, dim3(blocks), dim3(threads), 0, stream,
                     gate_up, out, rows, inter);
}

}  // namespace rlli
