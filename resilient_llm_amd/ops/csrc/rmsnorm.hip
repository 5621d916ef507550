// Fused (residual-add +) RMSNorm, bf16 I/O, f32 accumulation.
//
// Memory-bound: the lever on gfx950 is bf16x8 (16 B/lane) vectorized
// loads — scalar bf16 loads are ~2x slower (guide G13; measured 2.35 vs
// 4.89 TB/s on this op class).  One workgroup per row, one bf16x8 per
// lane when dim <= 8192 (Llama-8B dim 4096 -> 512 threads, 70B 8192 ->
// 1024), two-pass re-read (L2-hot) for larger dims.
//
// Replaces: no reference kernel exists (the reference runs no model code,
// SURVEY.md §2.1 native-code census) — greenfield per BASELINE.json.

#include "common.h"

#include <cstdlib>

namespace rlli {

namespace {

// Cross-wave reduction of one f32 per wave through LDS; every thread
// returns the block total.
template <int MAX_WAVES>
DEV_INLINE float block_sum(float v, float* lds) {
  const int lane = threadIdx.x & (kWave - 1);
  const int wave = threadIdx.x / kWave;
  v = wave_sum(v);
  if (lane == 0) lds[wave] = v;
  __syncthreads();
  const int n_waves = (blockDim.x + kWave - 1) / kWave;
  float total = 0.f;
#pragma unroll
  for (int w = 0; w < MAX_WAVES; ++w)
    if (w < n_waves) total += lds[w];
  return total;
}

// dim/8 <= blockDim: one bf16x8 per lane, data stays in registers.
template <bool FUSE_RESIDUAL>
__global__ void rmsnorm_one_vec(const uint16_t* __restrict__ x,
                                uint16_t* __restrict__ residual,
                                const uint16_t* __restrict__ w,
                                uint16_t* __restrict__ y,
                                int dim, float eps) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* lds = reinterpret_cast<float*>(smem_raw);
  const int row = blockIdx.x;
  const int64_t base = int64_t(row) * dim;
  const int v8 = threadIdx.x;            // which bf16x8 of the row
  const bool active = v8 * 8 < dim;

  float vals[8];
  if (active) {
    bf16x8 xv;
    xv.u = *reinterpret_cast<const uint4*>(x + base + v8 * 8);
#pragma unroll
    for (int i = 0; i < 8; ++i) vals[i] = bf16_to_f32(xv.s[i]);
    if constexpr (FUSE_RESIDUAL) {
      bf16x8 rv;
      rv.u = *reinterpret_cast<const uint4*>(residual + base + v8 * 8);
#pragma unroll
      for (int i = 0; i < 8; ++i) vals[i] += bf16_to_f32(rv.s[i]);
      bf16x8 out;
#pragma unroll
      for (int i = 0; i < 8; ++i) out.s[i] = f32_to_bf16(vals[i]);
      *reinterpret_cast<uint4*>(residual + base + v8 * 8) = out.u;
      // re-read the rounded residual so y == rmsnorm(stored residual)
#pragma unroll
      for (int i = 0; i < 8; ++i) vals[i] = bf16_to_f32(out.s[i]);
    }
  } else {
#pragma unroll
    for (int i = 0; i < 8; ++i) vals[i] = 0.f;
  }

  float ss = 0.f;
#pragma unroll
  for (int i = 0; i < 8; ++i) ss += vals[i] * vals[i];
  ss = block_sum<16>(ss, lds);
  const float inv = rsqrtf(ss / dim + eps);

  if (active) {
    bf16x8 wv, out;
    wv.u = *reinterpret_cast<const uint4*>(w + v8 * 8);
#pragma unroll
    for (int i = 0; i < 8; ++i)
      out.s[i] = f32_to_bf16(vals[i] * inv * bf16_to_f32(wv.s[i]));
    *reinterpret_cast<uint4*>(y + base + v8 * 8) = out.u;
  }
}

// Large dims: pass 1 accumulates sumsq (and stores fused residual),
// pass 2 re-reads (L2-hot) and writes.
template <bool FUSE_RESIDUAL>
__global__ void rmsnorm_two_pass(const uint16_t* __restrict__ x,
                                 uint16_t* __restrict__ residual,
                                 const uint16_t* __restrict__ w,
                                 uint16_t* __restrict__ y,
                                 int dim, float eps) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* lds = reinterpret_cast<float*>(smem_raw);
  const int row = blockIdx.x;
  const int64_t base = int64_t(row) * dim;
  const int nvec = dim / 8;

  float ss = 0.f;
  for (int v8 = threadIdx.x; v8 < nvec; v8 += blockDim.x) {
    bf16x8 xv;
    xv.u = *reinterpret_cast<const uint4*>(x + base + v8 * 8);
    float vals[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) vals[i] = bf16_to_f32(xv.s[i]);
    if constexpr (FUSE_RESIDUAL) {
      bf16x8 rv;
      rv.u = *reinterpret_cast<const uint4*>(residual + base + v8 * 8);
      bf16x8 out;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        vals[i] += bf16_to_f32(rv.s[i]);
        out.s[i] = f32_to_bf16(vals[i]);
        vals[i] = bf16_to_f32(out.s[i]);
      }
      *reinterpret_cast<uint4*>(residual + base + v8 * 8) = out.u;
    }
#pragma unroll
    for (int i = 0; i < 8; ++i) ss += vals[i] * vals[i];
  }
  ss = block_sum<16>(ss, lds);
  const float inv = rsqrtf(ss / dim + eps);

  const uint16_t* src = FUSE_RESIDUAL ? residual : x;
  for (int v8 = threadIdx.x; v8 < nvec; v8 += blockDim.x) {
    bf16x8 sv, wv, out;
    sv.u = *reinterpret_cast<const uint4*>(src + base + v8 * 8);
    wv.u = *reinterpret_cast<const uint4*>(w + v8 * 8);
#pragma unroll
    for (int i = 0; i < 8; ++i)
      out.s[i] = f32_to_bf16(bf16_to_f32(sv.s[i]) * inv * bf16_to_f32(wv.s[i]));
    *reinterpret_cast<uint4*>(y + base + v8 * 8) = out.u;
  }
}

// One WAVE per row, every load batch-issued.  The one-WG-per-row
// geometry above is kept for dim > 4096; at decode batch sizes it was
// latency-bound (batch 64 -> 64 WGs, one outstanding 16-B load per
// lane, 0.39 TB/s measured): a single wave with 2*VPL loads in flight
// streams its row at HBM rate and reduces wave-locally (no LDS, no
// barrier), and the small WGs pack many-per-CU.
template <int VPL, bool FUSE_RESIDUAL>
__global__ __launch_bounds__(64)
void rmsnorm_wave(const uint16_t* __restrict__ x,
                  uint16_t* __restrict__ residual,
                  const uint16_t* __restrict__ w,
                  uint16_t* __restrict__ y, int dim, float eps) {
  const int row = blockIdx.x;
  const int64_t base = int64_t(row) * dim;
  const int lane = threadIdx.x;
  const int nvec = dim / 8;

  uint4 xr[VPL], rr[VPL];
#pragma unroll
  for (int p = 0; p < VPL; ++p) {
    const int v8 = p * kWave + lane;
    xr[p] = uint4{0, 0, 0, 0};
    if (v8 < nvec) {
      xr[p] = *reinterpret_cast<const uint4*>(x + base + v8 * 8);
      if constexpr (FUSE_RESIDUAL)
        rr[p] = *reinterpret_cast<const uint4*>(residual + base + v8 * 8);
    }
  }

  float vals[VPL][8];
  float ss = 0.f;
#pragma unroll
  for (int p = 0; p < VPL; ++p) {
    const int v8 = p * kWave + lane;
    bf16x8 xv;
    xv.u = xr[p];
#pragma unroll
    for (int i = 0; i < 8; ++i) vals[p][i] = bf16_to_f32(xv.s[i]);
    if constexpr (FUSE_RESIDUAL) {
      if (v8 < nvec) {
        bf16x8 rv, outv;
        rv.u = rr[p];
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          vals[p][i] += bf16_to_f32(rv.s[i]);
          outv.s[i] = f32_to_bf16(vals[p][i]);
          // y must equal rmsnorm(STORED residual): round-trip via bf16
          vals[p][i] = bf16_to_f32(outv.s[i]);
        }
        *reinterpret_cast<uint4*>(residual + base + v8 * 8) = outv.u;
      }
    }
#pragma unroll
    for (int i = 0; i < 8; ++i) ss += vals[p][i] * vals[p][i];
  }
  ss = wave_sum(ss);
  const float inv = rsqrtf(ss / dim + eps);

#pragma unroll
  for (int p = 0; p < VPL; ++p) {
    const int v8 = p * kWave + lane;
    if (v8 >= nvec) continue;
    bf16x8 wv, outv;
    wv.u = *reinterpret_cast<const uint4*>(w + v8 * 8);
#pragma unroll
    for (int i = 0; i < 8; ++i)
      outv.s[i] = f32_to_bf16(vals[p][i] * inv * bf16_to_f32(wv.s[i]));
    *reinterpret_cast<uint4*>(y + base + v8 * 8) = outv.u;
  }
}

// FP8-output variant for the quantized-weight GEMM path: the normed row
// is emitted as fp8-e4m3 plus ONE per-row dequant scale (amax/448),
// computed with a second block reduction — the consumer is a rowwise
// torch._scaled_mm, so the 2-byte->1-byte write also halves the
// activation traffic.  Residual is still read-modify-written in bf16.
template <bool FUSE_RESIDUAL>
__global__ void rmsnorm_one_vec_fp8(const uint16_t* __restrict__ x,
                                    uint16_t* __restrict__ residual,
                                    const uint16_t* __restrict__ w,
                                    uint8_t* __restrict__ y8,
                                    float* __restrict__ scales,
                                    int dim, float eps) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* lds = reinterpret_cast<float*>(smem_raw);
  const int row = blockIdx.x;
  const int64_t base = int64_t(row) * dim;
  const int v8 = threadIdx.x;
  const bool active = v8 * 8 < dim;

  float vals[8];
  if (active) {
    bf16x8 xv;
    xv.u = *reinterpret_cast<const uint4*>(x + base + v8 * 8);
#pragma unroll
    for (int i = 0; i < 8; ++i) vals[i] = bf16_to_f32(xv.s[i]);
    if constexpr (FUSE_RESIDUAL) {
      bf16x8 rv;
      rv.u = *reinterpret_cast<const uint4*>(residual + base + v8 * 8);
#pragma unroll
      for (int i = 0; i < 8; ++i) vals[i] += bf16_to_f32(rv.s[i]);
      bf16x8 out;
#pragma unroll
      for (int i = 0; i < 8; ++i) out.s[i] = f32_to_bf16(vals[i]);
      *reinterpret_cast<uint4*>(residual + base + v8 * 8) = out.u;
#pragma unroll
      for (int i = 0; i < 8; ++i) vals[i] = bf16_to_f32(out.s[i]);
    }
  } else {
#pragma unroll
    for (int i = 0; i < 8; ++i) vals[i] = 0.f;
  }

  float ss = 0.f;
#pragma unroll
  for (int i = 0; i < 8; ++i) ss += vals[i] * vals[i];
  ss = block_sum<16>(ss, lds);
  const float inv = rsqrtf(ss / dim + eps);

  float out8[8];
  float amax = 0.f;
  if (active) {
    bf16x8 wv;
    wv.u = *reinterpret_cast<const uint4*>(w + v8 * 8);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      out8[i] = vals[i] * inv * bf16_to_f32(wv.s[i]);
      amax = fmaxf(amax, fabsf(out8[i]));
    }
  }
  // block max through the same LDS slots (barrier separates uses)
  {
    const int lane = threadIdx.x & (kWave - 1);
    const int wave = threadIdx.x / kWave;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      amax = fmaxf(amax, __shfl_xor(amax, off, kWave));
    __syncthreads();
    if (lane == 0) lds[wave] = amax;
    __syncthreads();
    const int n_waves = (blockDim.x + kWave - 1) / kWave;
    amax = 0.f;
#pragma unroll
    for (int wv = 0; wv < 16; ++wv)
      if (wv < n_waves) amax = fmaxf(amax, lds[wv]);
  }
  const float scale = fmaxf(amax, 1e-12f) / 448.f;
  if (active) {
    const float rs = 1.f / scale;
    float q[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) q[i] = out8[i] * rs;
    *reinterpret_cast<CacheFP8::vec8*>(y8 + base + v8 * 8) =
        CacheFP8::from_f32(q);
  }
  if (threadIdx.x == 0) scales[row] = scale;
}

template <bool FR>
void launch_wave(const uint16_t* x, uint16_t* residual, const uint16_t* w,
                 uint16_t* y, int rows, int dim, float eps,
                 hipStream_t stream) {
  const int nvec = dim / 8;
  const int vpl = (nvec + kWave - 1) / kWave;
  auto go = [&](auto tag) {
    hipLaunchKernelGGL((rmsnorm_wave<decltype(tag)::value, FR>), dim3(rows),
                       dim3(kWave), 0, stream, x, residual, w, y, dim, eps);
  };
  if (vpl <= 1) go(std::integral_constant<int, 1>{});
  else if (vpl <= 2) go(std::integral_constant<int, 2>{});
  else if (vpl <= 4) go(std::integral_constant<int, 4>{});
  else go(std::integral_constant<int, 8>{});
}

}  // namespace

void launch_rmsnorm_fp8(const uint16_t* x, uint16_t* residual,
                        const uint16_t* w, uint8_t* y8, float* scales,
                        int rows, int dim, float eps, hipStream_t stream) {
  if (rows == 0) return;
  const int smem = 16 * sizeof(float);
  const int threads = ((dim / 8 + kWave - 1) / kWave) * kWave;
  // fp8 activations are for the quantized-GEMM path; dims beyond the
  // one-vec envelope (8192) would need the two-pass form — none of the
  // supported models exceed it
  if (residual)
    hipLaunchKernelGGL((rmsnorm_one_vec_fp8<true>), dim3(rows), dim3(threads),
                       smem, stream, x, residual, w, y8, scales, dim, eps);
  else
    hipLaunchKernelGGL((rmsnorm_one_vec_fp8<false>), dim3(rows),
                       dim3(threads), smem, stream, x, nullptr, w, y8,
                       scales, dim, eps);
}

void launch_rmsnorm(const uint16_t* x, uint16_t* residual, const uint16_t* w,
                    uint16_t* y, int rows, int dim, float eps,
                    hipStream_t stream) {
  if (rows == 0) return;
  const int smem = 16 * sizeof(float);
  // A/B on MI355X (same box, full serving bench, batch 64, in-graph):
  // one-WG-per-row 138.3 reqs/s vs wave-per-row 130.2 — the wave
  // variant's deep per-lane ILP does not beat 8 waves' latency hiding
  // once launches are graph-captured, so it stays opt-in.
  static const bool wave = std::getenv("RLLI_WAVE_NORM") != nullptr;
  if (dim <= 4096 && wave) {
    if (residual)
      launch_wave<true>(x, residual, w, y, rows, dim, eps, stream);
    else
      launch_wave<false>(x, nullptr, w, y, rows, dim, eps, stream);
  } else if (dim <= 8192) {
    int threads = ((dim / 8 + kWave - 1) / kWave) * kWave;
    if (residual)
      hipLaunchKernelGGL((rmsnorm_one_vec<true>), dim3(rows), dim3(threads),
                         smem, stream, x, residual, w, y, dim, eps);
    else
      hipLaunchKernelGGL((rmsnorm_one_vec<false>), dim3(rows), dim3(threads),
                         smem, stream, x, nullptr, w, y, dim, eps);
  } else {
    if (residual)
      hipLaunchKernelGGL((rmsnorm_two_pass<true>), dim3(rows), dim3(1024),
                         smem, stream, x, residual, w, y, dim, eps);
    else
      hipLaunchKernelGGL((rmsnorm_two_pass<false>), dim3(rows), dim3(1024),
                         smem, stream, x, nullptr, w, y, dim, eps);
  }
}

}  // namespace rlli
