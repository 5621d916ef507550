// Torch bindings for the gfx950 kernels (TORCH_LIBRARY, not pybind — the
// ops dispatch through torch.ops.rlli.* and stay graph-capturable).
// This is the only translation unit that includes torch headers; the
// kernels live in pure-HIP .hip files behind kernels.h.

#include <torch/library.h>
#include <ATen/ATen.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <ATen/hip/impl/HIPGuardImplMasqueradingAsCUDA.h>
#include <c10/hip/HIPGraphsC10Utils.h>

#include "kernels.h"

#include <cstdlib>
#include <vector>

namespace {

using at::Tensor;

hipStream_t current_stream(const Tensor& t) {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA(
      t.device().index()).stream();
}

// after-launch error check (skipped inside hipGraph capture, where the
// runtime reports capture-sequencing pseudo-errors)
void check_launch(const char* op) {
  if (c10::hip::currentStreamCaptureStatusMayInitCtx() !=
      c10::hip::CaptureStatus::None)
    return;
  hipError_t e = hipGetLastError();
  TORCH_CHECK(e == hipSuccess, op, " launch failed: ", hipGetErrorString(e));
}

uint16_t* bf16_ptr(const Tensor& t) {
  return reinterpret_cast<uint16_t*>(t.data_ptr());
}

void check_bf16_contig(const Tensor& t, const char* name) {
  TORCH_CHECK(t.scalar_type() == at::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.is_cuda(), name, " must be on the GPU");
}


// cache tensors: bf16 or fp8-e4m3 (common.h codecs); returns the fp8
// flag after validating dtype/contiguity/device
bool check_cache(const Tensor& k_cache, const Tensor& v_cache) {
  TORCH_CHECK(k_cache.scalar_type() == v_cache.scalar_type(),
              "k/v cache dtype mismatch");
  const bool fp8 = k_cache.scalar_type() == at::kFloat8_e4m3fn;
  TORCH_CHECK(fp8 || k_cache.scalar_type() == at::kBFloat16,
              "kv cache must be bf16 or float8_e4m3fn");
  TORCH_CHECK(k_cache.is_contiguous() && v_cache.is_contiguous() &&
              k_cache.is_cuda() && v_cache.is_cuda());
  return fp8;
}

// ------------------------------------------------------------- rmsnorm
Tensor rmsnorm(const Tensor& x, const Tensor& w, double eps) {
  check_bf16_contig(x, "x");
  check_bf16_contig(w, "w");
  const int dim = int(x.size(-1));
  TORCH_CHECK(dim % 8 == 0, "dim must be a multiple of 8");
  TORCH_CHECK(w.numel() == dim, "weight/dim mismatch");
  const int rows = int(x.numel() / dim);
  Tensor y = at::empty_like(x);
  c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  rlli::launch_rmsnorm(bf16_ptr(x), nullptr, bf16_ptr(w), bf16_ptr(y), rows,
                       dim, float(eps), current_stream(x));
  check_launch("rmsnorm");
  return y;
}

Tensor rmsnorm_residual_(const Tensor& x, Tensor residual, const Tensor& w,
                         double eps) {
  check_bf16_contig(x, "x");
  check_bf16_contig(residual, "residual");
  check_bf16_contig(w, "w");
  TORCH_CHECK(x.sizes() == residual.sizes(), "x/residual shape mismatch");
  const int dim = int(x.size(-1));
  TORCH_CHECK(dim % 8 == 0 && w.numel() == dim, "bad dim/weight");
  const int rows = int(x.numel() / dim);
  Tensor y = at::empty_like(x);
  c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  rlli::launch_rmsnorm(bf16_ptr(x), bf16_ptr(residual), bf16_ptr(w),
                       bf16_ptr(y), rows, dim, float(eps), current_stream(x));
  check_launch("rmsnorm_residual_");
  return y;
}

// ---- fp8-activation emitters (quantized-weight GEMM path) ----
std::tuple<Tensor, Tensor> rmsnorm_fp8(const Tensor& x, const Tensor& w,
                                       double eps) {
  check_bf16_contig(x, "x");
  check_bf16_contig(w, "w");
  const int dim = int(x.size(-1));
  TORCH_CHECK(dim % 8 == 0 && dim <= 8192 && w.numel() == dim,
              "bad dim/weight for fp8 rmsnorm");
  const int rows = int(x.numel() / dim);
  Tensor y8 = at::empty(x.sizes(), x.options().dtype(at::kFloat8_e4m3fn));
  Tensor scales = at::empty({rows}, x.options().dtype(at::kFloat));
  c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  rlli::launch_rmsnorm_fp8(bf16_ptr(x), nullptr, bf16_ptr(w),
                           reinterpret_cast<uint8_t*>(y8.data_ptr()),
                           scales.data_ptr<float>(), rows, dim, float(eps),
                           current_stream(x));
  check_launch("rmsnorm_fp8");
  return {y8, scales};
}

std::tuple<Tensor, Tensor> rmsnorm_residual_fp8(const Tensor& x,
                                                Tensor residual,
                                                const Tensor& w, double eps) {
  check_bf16_contig(x, "x");
  check_bf16_contig(residual, "residual");
  check_bf16_contig(w, "w");
  TORCH_CHECK(x.sizes() == residual.sizes(), "x/residual shape mismatch");
  const int dim = int(x.size(-1));
  TORCH_CHECK(dim % 8 == 0 && dim <= 8192 && w.numel() == dim,
              "bad dim/weight for fp8 rmsnorm");
  const int rows = int(x.numel() / dim);
  Tensor y8 = at::empty(x.sizes(), x.options().dtype(at::kFloat8_e4m3fn));
  Tensor scales = at::empty({rows}, x.options().dtype(at::kFloat));
  c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  rlli::launch_rmsnorm_fp8(bf16_ptr(x), bf16_ptr(residual), bf16_ptr(w),
                           reinterpret_cast<uint8_t*>(y8.data_ptr()),
                           scales.data_ptr<float>(), rows, dim, float(eps),
                           current_stream(x));
  check_launch("rmsnorm_residual_fp8");
  return {y8, scales};
}

std::tuple<Tensor, Tensor> silu_mul_fp8(const Tensor& gate_up) {
  check_bf16_contig(gate_up, "gate_up");
  const int64_t inter = gate_up.size(-1) / 2;
  TORCH_CHECK(inter % 8 == 0, "inter must be a multiple of 8");
  const int rows = int(gate_up.numel() / (2 * inter));
  auto sizes = gate_up.sizes().vec();
  sizes.back() = inter;
  Tensor y8 = at::empty(sizes, gate_up.options().dtype(at::kFloat8_e4m3fn));
  Tensor scales = at::empty({rows}, gate_up.options().dtype(at::kFloat));
  c10::hip::HIPGuardMasqueradingAsCUDA guard(gate_up.device());
  rlli::launch_silu_mul_fp8(bf16_ptr(gate_up),
                            reinterpret_cast<uint8_t*>(y8.data_ptr()),
                            scales.data_ptr<float>(), rows, int(inter),
                            current_stream(gate_up));
  check_launch("silu_mul_fp8");
  return {y8, scales};
}

// ------------------------------------------------------------ silu_mul
Tensor silu_mul(const Tensor& gate_up) {
  check_bf16_contig(gate_up, "gate_up");
  const int twoi = int(gate_up.size(-1));
  TORCH_CHECK(twoi % 16 == 0, "2*intermediate must be a multiple of 16");
  const int inter = twoi / 2;
  const int rows = int(gate_up.numel() / twoi);
  auto sizes = gate_up.sizes().vec();
  sizes.back() = inter;
  Tensor y = at::empty(sizes, gate_up.options());
  c10::hip::HIPGuardMasqueradingAsCUDA guard(gate_up.device());
  rlli::launch_silu_mul(bf16_ptr(gate_up), bf16_ptr(y), rows, inter,
                        current_stream(gate_up));
  check_launch("silu_mul");
  return y;
}

// ------------------------------------------------------- rope_kv_append
void rope_kv_append_(Tensor q, Tensor k, const Tensor& v,
                     const Tensor& positions, const Tensor& cos_sin,
                     Tensor k_cache, Tensor v_cache,
                     const Tensor& slot_mapping) {
  check_bf16_contig(q, "q");
  check_bf16_contig(k, "k");
  check_bf16_contig(v, "v");
  const bool cache_fp8 = check_cache(k_cache, v_cache);
  TORCH_CHECK(positions.scalar_type() == at::kInt && positions.is_contiguous());
  TORCH_CHECK(slot_mapping.scalar_type() == at::kInt && slot_mapping.is_contiguous());
  TORCH_CHECK(cos_sin.scalar_type() == at::kFloat && cos_sin.is_contiguous());
  const int tokens = int(q.size(0));
  const int n_q = int(q.size(1));
  const int n_kv = int(k.size(1));
  const int D = int(q.size(2));
  TORCH_CHECK(k.size(2) == D && v.size(1) == n_kv && v.size(2) == D);
  TORCH_CHECK(D % 16 == 0, "head_dim must be a multiple of 16");
  TORCH_CHECK(cos_sin.size(1) == D, "cos_sin must be [max_pos, head_dim]");
  TORCH_CHECK(k_cache.size(1) == n_kv && k_cache.size(3) == D);
  const int block_size = int(k_cache.size(2));
  TORCH_CHECK(positions.numel() == tokens && slot_mapping.numel() == tokens);
  c10::hip::HIPGuardMasqueradingAsCUDA guard(q.device());
  rlli::launch_rope_kv_append(
      bf16_ptr(q), bf16_ptr(k), bf16_ptr(v),
      positions.data_ptr<int32_t>(), cos_sin.data_ptr<float>(),
      k_cache.data_ptr(), v_cache.data_ptr(),
      slot_mapping.data_ptr<int32_t>(),
      tokens, n_q, n_kv, D, block_size, n_q * D, n_kv * D, cache_fp8,
      current_stream(q));
  check_launch("rope_kv_append_");
}

// Fused-QKV form: qkv is the raw [T, (n_q + 2*n_kv) * D] GEMM output;
// rope + cache append run on strided views, zero copies.
void rope_kv_append_qkv_(Tensor qkv, const Tensor& positions,
                         const Tensor& cos_sin, Tensor k_cache,
                         Tensor v_cache, const Tensor& slot_mapping,
                         int64_t n_q) {
  check_bf16_contig(qkv, "qkv");
  const bool cache_fp8 = check_cache(k_cache, v_cache);
  const int n_kv = int(k_cache.size(1));
  const int D = int(k_cache.size(3));
  const int block_size = int(k_cache.size(2));
  const int stride = int(qkv.size(1));
  TORCH_CHECK(stride == (n_q + 2 * n_kv) * D, "qkv width mismatch");
  const int tokens = int(qkv.size(0));
  TORCH_CHECK(positions.numel() == tokens && slot_mapping.numel() == tokens);
  TORCH_CHECK(cos_sin.size(1) == D && D % 16 == 0);
  uint16_t* base = bf16_ptr(qkv);
  c10::hip::HIPGuardMasqueradingAsCUDA guard(qkv.device());
  rlli::launch_rope_kv_append(
      base, base + n_q * D, base + (n_q + n_kv) * D,
      positions.data_ptr<int32_t>(), cos_sin.data_ptr<float>(),
      k_cache.data_ptr(), v_cache.data_ptr(),
      slot_mapping.data_ptr<int32_t>(),
      tokens, int(n_q), n_kv, D, block_size, stride, stride, cache_fp8,
      current_stream(qkv));
  check_launch("rope_kv_append_qkv_");
}


// split-K scratch for decode attention (empty tensors when n_split==1);
// allocation participates in hipGraph capture pools, so graphed decode
// buckets capture their scratch once and reuse it on replay.
struct SplitScratch {
  int n_split = 1;
  Tensor part, ml;
  float* part_ptr() { return n_split > 1 ? part.data_ptr<float>() : nullptr; }
  float* ml_ptr() { return n_split > 1 ? ml.data_ptr<float>() : nullptr; }
};

static SplitScratch make_split_scratch(const Tensor& ref, int batch,
                                       int n_kv, int group, int D) {
  SplitScratch s;
  static const bool disabled = std::getenv("RLLI_NO_SPLITK") != nullptr;
  if (disabled) return s;
  s.n_split = rlli::decode_attn_n_split(batch, n_kv);
  if (s.n_split > 1) {
    auto opt = ref.options().dtype(at::kFloat);
    s.part = at::empty({int64_t(batch) * n_kv * s.n_split * group * D}, opt);
    s.ml = at::empty({int64_t(batch) * n_kv * s.n_split * group * 2}, opt);
  }
  return s;
}

// ---------------------------------------------------------- decode_attn
Tensor decode_attn(const Tensor& q, const Tensor& k_cache,
                   const Tensor& v_cache, const Tensor& block_table,
                   const Tensor& seq_lens, double scale) {
  check_bf16_contig(q, "q");
  const bool cache_fp8 = check_cache(k_cache, v_cache);
  TORCH_CHECK(block_table.scalar_type() == at::kInt && block_table.is_contiguous());
  TORCH_CHECK(seq_lens.scalar_type() == at::kInt && seq_lens.is_contiguous());
  const int batch = int(q.size(0));
  const int n_q = int(q.size(1));
  const int D = int(q.size(2));
  const int n_kv = int(k_cache.size(1));
  const int block_size = int(k_cache.size(2));
  const int max_blocks = int(block_table.size(1));
  const int group = n_q / n_kv;
  TORCH_CHECK(n_q % n_kv == 0 && (group == 1 || group == 2 || group == 4 || group == 8),
              "GQA group must be 1/2/4/8, got ", group);
  TORCH_CHECK(D == 64 || D == 128, "head_dim must be 64 or 128, got ", D);
  const int gpw = 64 / (D / 8);
  TORCH_CHECK(block_size % gpw == 0, "block_size must be a multiple of ", gpw);
  TORCH_CHECK(k_cache.size(3) == D && v_cache.sizes() == k_cache.sizes());
  TORCH_CHECK(block_table.size(0) == batch && seq_lens.numel() == batch);
  Tensor out = at::empty_like(q);
  c10::hip::HIPGuardMasqueradingAsCUDA guard(q.device());
  SplitScratch ss = make_split_scratch(q, batch, n_kv, group, D);
  rlli::launch_decode_attn(
      bf16_ptr(q), k_cache.data_ptr(), v_cache.data_ptr(),
      block_table.data_ptr<int32_t>(), seq_lens.data_ptr<int32_t>(),
      bf16_ptr(out), batch, n_q, n_kv, D, block_size, max_blocks,
      float(scale), n_q * D, ss.n_split, ss.part_ptr(), ss.ml_ptr(),
      cache_fp8, current_stream(q));
  check_launch("decode_attn");
  return out;
}

// Fused-QKV decode attention: q read strided from the qkv GEMM output.
Tensor decode_attn_qkv(const Tensor& qkv, const Tensor& k_cache,
                       const Tensor& v_cache, const Tensor& block_table,
                       const Tensor& seq_lens, double scale, int64_t n_q) {
  check_bf16_contig(qkv, "qkv");
  const bool cache_fp8 = check_cache(k_cache, v_cache);
  const int batch = int(qkv.size(0));
  const int n_kv = int(k_cache.size(1));
  const int D = int(k_cache.size(3));
  const int block_size = int(k_cache.size(2));
  const int max_blocks = int(block_table.size(1));
  const int stride = int(qkv.size(1));
  TORCH_CHECK(stride == (n_q + 2 * n_kv) * D, "qkv width mismatch");
  const int group = int(n_q) / n_kv;
  TORCH_CHECK(group == 1 || group == 2 || group == 4 || group == 8);
  TORCH_CHECK(D == 64 || D == 128);
  Tensor out = at::empty({batch, n_q * D}, qkv.options());
  c10::hip::HIPGuardMasqueradingAsCUDA guard(qkv.device());
  SplitScratch ss = make_split_scratch(qkv, batch, n_kv, group, D);
  rlli::launch_decode_attn(
      bf16_ptr(qkv), k_cache.data_ptr(), v_cache.data_ptr(),
      block_table.data_ptr<int32_t>(), seq_lens.data_ptr<int32_t>(),
      bf16_ptr(out), batch, int(n_q), n_kv, D, block_size, max_blocks,
      float(scale), stride, ss.n_split, ss.part_ptr(), ss.ml_ptr(),
      cache_fp8, current_stream(qkv));
  check_launch("decode_attn_qkv");
  return out;
}

// Fused decode: rope + append + attention in one kernel.
Tensor decode_attn_rope_qkv(const Tensor& qkv, const Tensor& positions,
                            const Tensor& cos_sin, Tensor k_cache,
                            Tensor v_cache, const Tensor& slot_mapping,
                            const Tensor& block_table, const Tensor& seq_lens,
                            double scale, int64_t n_q) {
  check_bf16_contig(qkv, "qkv");
  const bool cache_fp8 = check_cache(k_cache, v_cache);
  TORCH_CHECK(positions.scalar_type() == at::kInt && positions.is_contiguous());
  TORCH_CHECK(slot_mapping.scalar_type() == at::kInt && slot_mapping.is_contiguous());
  TORCH_CHECK(cos_sin.scalar_type() == at::kFloat && cos_sin.is_contiguous());
  TORCH_CHECK(block_table.scalar_type() == at::kInt && block_table.is_contiguous());
  TORCH_CHECK(seq_lens.scalar_type() == at::kInt && seq_lens.is_contiguous());
  const int batch = int(qkv.size(0));
  const int n_kv = int(k_cache.size(1));
  const int block_size = int(k_cache.size(2));
  const int D = int(k_cache.size(3));
  const int stride = int(qkv.size(1));
  TORCH_CHECK(stride == (n_q + 2 * n_kv) * D, "qkv width mismatch");
  const int group = int(n_q) / n_kv;
  TORCH_CHECK(group == 1 || group == 2 || group == 4 || group == 8);
  TORCH_CHECK(D == 64 || D == 128);
  TORCH_CHECK(cos_sin.size(1) == D);
  const int max_blocks = int(block_table.size(1));
  Tensor out = at::empty({batch, n_q * D}, qkv.options());
  c10::hip::HIPGuardMasqueradingAsCUDA guard(qkv.device());
  SplitScratch ss = make_split_scratch(qkv, batch, n_kv, group, D);
  rlli::launch_decode_attn_fused(
      bf16_ptr(qkv), k_cache.data_ptr(), v_cache.data_ptr(),
      block_table.data_ptr<int32_t>(), seq_lens.data_ptr<int32_t>(),
      positions.data_ptr<int32_t>(), cos_sin.data_ptr<float>(),
      slot_mapping.data_ptr<int32_t>(), bf16_ptr(out), batch, int(n_q),
      n_kv, D, block_size, max_blocks, float(scale), stride,
      ss.n_split, ss.part_ptr(), ss.ml_ptr(), cache_fp8,
      current_stream(qkv));
  check_launch("decode_attn_rope_qkv");
  return out;
}

// --------------------------------------------------------- prefill_attn
Tensor prefill_attn(const Tensor& q, const Tensor& k, const Tensor& v,
                    const Tensor& cu_seqlens, double scale) {
  check_bf16_contig(q, "q");
  check_bf16_contig(k, "k");
  check_bf16_contig(v, "v");
  TORCH_CHECK(cu_seqlens.scalar_type() == at::kInt && cu_seqlens.is_contiguous());
  const int T = int(q.size(0));
  const int n_q = int(q.size(1));
  const int D = int(q.size(2));
  const int n_kv = int(k.size(1));
  const int group = n_q / n_kv;
  TORCH_CHECK(n_q % n_kv == 0 && (group == 1 || group == 2 || group == 4 || group == 8),
              "GQA group must be 1/2/4/8, got ", group);
  TORCH_CHECK(D == 64 || D == 128, "head_dim must be 64 or 128, got ", D);
  TORCH_CHECK(k.size(0) == T && v.size(0) == T && k.size(2) == D);
  const int n_seqs = int(cu_seqlens.numel()) - 1;
  Tensor out = at::empty_like(q);
  c10::hip::HIPGuardMasqueradingAsCUDA guard(q.device());
  rlli::launch_prefill_attn(
      bf16_ptr(q), bf16_ptr(k), bf16_ptr(v), cu_seqlens.data_ptr<int32_t>(),
      bf16_ptr(out), n_seqs, T, n_q, n_kv, D, float(scale),
      n_q * D, n_kv * D, current_stream(q));
  check_launch("prefill_attn");
  return out;
}

// Fused-QKV prefill attention: q/k/v all read strided from qkv (k was
// rope-rotated in place by rope_kv_append_qkv_).
Tensor prefill_attn_qkv(const Tensor& qkv, const Tensor& cu_seqlens,
                        double scale, int64_t n_q, int64_t n_kv,
                        int64_t head_dim) {
  check_bf16_contig(qkv, "qkv");
  TORCH_CHECK(cu_seqlens.scalar_type() == at::kInt && cu_seqlens.is_contiguous());
  const int T = int(qkv.size(0));
  const int D = int(head_dim);
  const int stride = int(qkv.size(1));
  TORCH_CHECK(stride == (n_q + 2 * n_kv) * D, "qkv width mismatch");
  const int group = int(n_q) / int(n_kv);
  TORCH_CHECK(group == 1 || group == 2 || group == 4 || group == 8);
  TORCH_CHECK(D == 64 || D == 128);
  const int n_seqs = int(cu_seqlens.numel()) - 1;
  Tensor out = at::empty({T, n_q * D}, qkv.options());
  c10::hip::HIPGuardMasqueradingAsCUDA guard(qkv.device());
  // chunk tables for the MFMA kernel (host-side; prefill is not captured)
  Tensor cu_cpu = cu_seqlens.to(at::kCPU);
  const int32_t* cu = cu_cpu.data_ptr<int32_t>();
  std::vector<int32_t> t0s, starts, ends;
  for (int s = 0; s < n_seqs; ++s)
    for (int t = cu[s]; t < cu[s + 1]; t += 32) {
      t0s.push_back(t);
      starts.push_back(cu[s]);
      ends.push_back(cu[s + 1]);
    }
  const int n_chunks = int(t0s.size());
  auto iopt = at::TensorOptions().dtype(at::kInt);
  Tensor t0_d = at::tensor(t0s, iopt).to(qkv.device(), /*nb=*/true);
  Tensor st_d = at::tensor(starts, iopt).to(qkv.device(), true);
  Tensor en_d = at::tensor(ends, iopt).to(qkv.device(), true);
  rlli::launch_prefill_mfma(
      bf16_ptr(qkv), t0_d.data_ptr<int32_t>(), st_d.data_ptr<int32_t>(),
      en_d.data_ptr<int32_t>(), bf16_ptr(out), n_chunks, int(n_kv),
      group, D, stride, float(scale), current_stream(qkv));
  check_launch("prefill_attn_qkv");
  return out;
}

// ------------------------------------------------------ prefill (paged)
Tensor prefill_paged_attn(const Tensor& qkv, const Tensor& k_cache,
                          const Tensor& v_cache, const Tensor& chunk_row0,
                          const Tensor& chunk_pos0, const Tensor& chunk_nrows,
                          const Tensor& chunk_btrow,
                          const Tensor& block_tables, double scale,
                          int64_t n_q) {
  check_bf16_contig(qkv, "qkv");
  const bool cache_fp8 = check_cache(k_cache, v_cache);
  for (const Tensor* t : {&chunk_row0, &chunk_pos0, &chunk_nrows,
                          &chunk_btrow, &block_tables}) {
    TORCH_CHECK(t->scalar_type() == at::kInt && t->is_contiguous() &&
                t->is_cuda(), "chunk/table tensors must be contiguous "
                "int32 on the GPU");
  }
  const int T = int(qkv.size(0));
  const int n_kv = int(k_cache.size(1));
  const int block_size = int(k_cache.size(2));
  const int D = int(k_cache.size(3));
  const int stride = int(qkv.size(1));
  TORCH_CHECK(stride >= n_q * D, "qkv too narrow");
  const int group = int(n_q) / n_kv;
  TORCH_CHECK(group == 1 || group == 2 || group == 4 || group == 8);
  TORCH_CHECK(D == 64 || D == 128);
  const int n_chunks = int(chunk_row0.numel());
  const int max_blocks = int(block_tables.size(1));
  Tensor out = at::empty({T, n_q * D}, qkv.options());
  c10::hip::HIPGuardMasqueradingAsCUDA guard(qkv.device());
  rlli::launch_prefill_paged(
      bf16_ptr(qkv), k_cache.data_ptr(), v_cache.data_ptr(),
      chunk_row0.data_ptr<int32_t>(), chunk_pos0.data_ptr<int32_t>(),
      chunk_nrows.data_ptr<int32_t>(), chunk_btrow.data_ptr<int32_t>(),
      block_tables.data_ptr<int32_t>(), bf16_ptr(out), n_chunks, n_kv,
      group, D, stride, max_blocks, block_size, float(scale), cache_fp8,
      current_stream(qkv));
  check_launch("prefill_paged_attn");
  return out;
}

// --------------------------------------------------------- skinny_linear
Tensor skinny_linear(const Tensor& x, const Tensor& w) {
  check_bf16_contig(x, "x");
  check_bf16_contig(w, "w");
  const int M = int(x.size(0));
  const int K = int(x.size(1));
  const int N = int(w.size(0));
  TORCH_CHECK(w.size(1) == K, "x/W K mismatch");
  TORCH_CHECK(M >= 1 && M <= 64, "skinny_linear needs 1 <= M <= 64, got ", M);
  TORCH_CHECK(K % 256 == 0 && N % 64 == 0,
              "skinny_linear needs K%256==0 and N%64==0, got K=", K, " N=", N);
  const int n_panels = N / 64;
  int splitk = 1;
  if (n_panels < 256) {
    splitk = std::min<int>(8, std::max<int>(1, 512 / n_panels));
    splitk = std::min<int>(splitk, K / 256);
  }
  Tensor out = at::empty({M, N}, x.options());
  Tensor ws;
  float* ws_ptr = nullptr;
  if (splitk > 1) {
    ws = at::empty({splitk, M, N}, x.options().dtype(at::kFloat));
    ws_ptr = ws.data_ptr<float>();
  }
  c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  rlli::launch_skinny_gemm(bf16_ptr(x), bf16_ptr(w), ws_ptr, bf16_ptr(out),
                           M, N, K, splitk, current_stream(x));
  check_launch("skinny_linear");
  return out;
}

// ------------------------------------------------------------- skinny2
// v2 zero-LDS skinny GEMM; splitk <= 0 picks the grid-fill heuristic
// (>= ~512 workgroups).  Requirements: M <= 64, K % 256 == 0, N % 16 == 0.
static int skinny2_auto_splitk(int N, int K) {
  const int panels = N / 64;   // one 64-column panel per wave
  int sk = std::max(1, (512 + panels - 1) / panels);
  return std::min<int>(sk, std::max(1, K / 256));
}

Tensor skinny2_linear(const Tensor& x, const Tensor& w, int64_t splitk) {
  check_bf16_contig(x, "x");
  check_bf16_contig(w, "w");
  const int M = int(x.size(0));
  const int K = int(x.size(1));
  const int N = int(w.size(0));
  TORCH_CHECK(w.size(1) == K, "x/W K mismatch");
  TORCH_CHECK(M >= 1 && M <= 64, "skinny2_linear needs 1 <= M <= 64, got ", M);
  TORCH_CHECK(K % 256 == 0 && N % 64 == 0,
              "skinny2_linear needs K%256==0 and N%64==0, got K=", K, " N=", N);
  int sk = int(splitk) > 0 ? int(splitk) : skinny2_auto_splitk(N, K);
  sk = std::min(sk, K / 256);
  Tensor out = at::empty({M, N}, x.options());
  Tensor ws;
  float* ws_ptr = nullptr;
  if (sk > 1) {
    ws = at::empty({sk, M, N}, x.options().dtype(at::kFloat));
    ws_ptr = ws.data_ptr<float>();
  }
  c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  rlli::launch_skinny2(bf16_ptr(x), bf16_ptr(w), ws_ptr, bf16_ptr(out),
                       M, N, K, sk, current_stream(x));
  check_launch("skinny2_linear");
  return out;
}

// down-proj with the SwiGLU fused into the A-fragment path:
// out = (silu(gu[:, :K]) * gu[:, K:]) @ w^T,  gu = [M, 2K].
Tensor skinny2_silu_linear(const Tensor& gu, const Tensor& w, int64_t splitk) {
  check_bf16_contig(gu, "gu");
  check_bf16_contig(w, "w");
  const int M = int(gu.size(0));
  const int K2 = int(gu.size(1));
  TORCH_CHECK(K2 % 2 == 0, "gu must be [M, 2K]");
  const int K = K2 / 2;
  const int N = int(w.size(0));
  TORCH_CHECK(w.size(1) == K, "gu/W K mismatch");
  TORCH_CHECK(M >= 1 && M <= 64, "skinny2_silu needs 1 <= M <= 64, got ", M);
  TORCH_CHECK(K % 256 == 0 && N % 64 == 0,
              "skinny2_silu needs K%256==0 and N%64==0, got K=", K, " N=", N);
  int sk = int(splitk) > 0 ? int(splitk) : skinny2_auto_splitk(N, K);
  sk = std::min(sk, K / 256);
  Tensor out = at::empty({M, N}, gu.options());
  Tensor ws;
  float* ws_ptr = nullptr;
  if (sk > 1) {
    ws = at::empty({sk, M, N}, gu.options().dtype(at::kFloat));
    ws_ptr = ws.data_ptr<float>();
  }
  c10::hip::HIPGuardMasqueradingAsCUDA guard(gu.device());
  rlli::launch_skinny2_silu(bf16_ptr(gu), bf16_ptr(w), ws_ptr, bf16_ptr(out),
                            M, N, K, sk, current_stream(gu));
  check_launch("skinny2_silu_linear");
  return out;
}

Tensor stream_probe(const Tensor& w, int64_t splitk) {
  check_bf16_contig(w, "w");
  Tensor sink = at::zeros({256}, w.options().dtype(at::kFloat));
  c10::hip::HIPGuardMasqueradingAsCUDA guard(w.device());
  rlli::launch_stream_probe(bf16_ptr(w), sink.data_ptr<float>(),
                            int(w.size(0)), int(w.size(1)), int(splitk),
                            current_stream(w));
  check_launch("stream_probe");
  return sink;
}

// --------------------------------------------------------------- sample
Tensor sample(const Tensor& logits, const Tensor& temperatures,
              const Tensor& seeds, int64_t step) {
  check_bf16_contig(logits, "logits");
  TORCH_CHECK(temperatures.scalar_type() == at::kFloat &&
              temperatures.is_contiguous() && temperatures.is_cuda());
  TORCH_CHECK(seeds.scalar_type() == at::kLong && seeds.is_contiguous() &&
              seeds.is_cuda());
  const int batch = int(logits.size(0));
  const int vocab = int(logits.size(1));
  TORCH_CHECK(temperatures.numel() == batch && seeds.numel() == batch);
  Tensor out = at::empty({batch}, logits.options().dtype(at::kInt));
  c10::hip::HIPGuardMasqueradingAsCUDA guard(logits.device());
  const int n_split = rlli::sample_n_split(batch);
  uint64_t* pptr = nullptr;
  Tensor partials;
  if (n_split > 1) {
    // caching-allocator tensor: participates in hipGraph capture pools
    partials = at::empty({int64_t(batch) * n_split},
                         logits.options().dtype(at::kLong));
    pptr = reinterpret_cast<uint64_t*>(partials.data_ptr<int64_t>());
  }
  rlli::launch_sample(bf16_ptr(logits), temperatures.data_ptr<float>(),
                      reinterpret_cast<const uint64_t*>(seeds.data_ptr<int64_t>()),
                      uint64_t(step), out.data_ptr<int32_t>(), batch, vocab,
                      pptr, n_split, current_stream(logits));
  check_launch("sample");
  return out;
}

}  // namespace

TORCH_LIBRARY(rlli, m) {
  m.def("rmsnorm(Tensor x, Tensor w, float eps) -> Tensor");
  m.def("rmsnorm_residual_(Tensor x, Tensor(a!) residual, Tensor w, float eps) -> Tensor");
  m.def("silu_mul(Tensor gate_up) -> Tensor");
  m.def("rmsnorm_fp8(Tensor x, Tensor w, float eps) -> (Tensor, Tensor)");
  m.def("rmsnorm_residual_fp8(Tensor x, Tensor(a!) residual, Tensor w, float eps) -> (Tensor, Tensor)");
  m.def("silu_mul_fp8(Tensor gate_up) -> (Tensor, Tensor)");
  m.def("rope_kv_append_(Tensor(a!) q, Tensor(b!) k, Tensor v, Tensor positions, "
        "Tensor cos_sin, Tensor(c!) k_cache, Tensor(d!) v_cache, "
        "Tensor slot_mapping) -> ()");
  m.def("decode_attn(Tensor q, Tensor k_cache, Tensor v_cache, "
        "Tensor block_table, Tensor seq_lens, float scale) -> Tensor");
  m.def("rope_kv_append_qkv_(Tensor(a!) qkv, Tensor positions, Tensor cos_sin, "
        "Tensor(b!) k_cache, Tensor(c!) v_cache, Tensor slot_mapping, "
        "int n_q) -> ()");
  m.def("decode_attn_qkv(Tensor qkv, Tensor k_cache, Tensor v_cache, "
        "Tensor block_table, Tensor seq_lens, float scale, int n_q) -> Tensor");
  m.def("decode_attn_rope_qkv(Tensor qkv, Tensor positions, Tensor cos_sin, "
        "Tensor(a!) k_cache, Tensor(b!) v_cache, Tensor slot_mapping, "
        "Tensor block_table, Tensor seq_lens, float scale, int n_q) -> Tensor");
  m.def("prefill_attn_qkv(Tensor qkv, Tensor cu_seqlens, float scale, "
        "int n_q, int n_kv, int head_dim) -> Tensor");
  m.def("prefill_attn(Tensor q, Tensor k, Tensor v, Tensor cu_seqlens, "
        "float scale) -> Tensor");
  m.def("sample(Tensor logits, Tensor temperatures, Tensor seeds, int step) -> Tensor");
  m.def("skinny_linear(Tensor x, Tensor w) -> Tensor");
  m.def("skinny2_linear(Tensor x, Tensor w, int splitk) -> Tensor");
  m.def("skinny2_silu_linear(Tensor gu, Tensor w, int splitk) -> Tensor");
  m.def("stream_probe(Tensor w, int splitk) -> Tensor");
  m.def("prefill_paged_attn(Tensor qkv, Tensor k_cache, Tensor v_cache, "
        "Tensor chunk_row0, Tensor chunk_pos0, Tensor chunk_nrows, "
        "Tensor chunk_btrow, Tensor block_tables, float scale, int n_q) "
        "-> Tensor");
}

TORCH_LIBRARY_IMPL(rlli, CUDA, m) {
  m.impl("rmsnorm", &rmsnorm);
  m.impl("rmsnorm_residual_", &rmsnorm_residual_);
  m.impl("silu_mul", &silu_mul);
  m.impl("rmsnorm_fp8", &rmsnorm_fp8);
  m.impl("rmsnorm_residual_fp8", &rmsnorm_residual_fp8);
  m.impl("silu_mul_fp8", &silu_mul_fp8);
  m.impl("rope_kv_append_", &rope_kv_append_);
  m.impl("decode_attn", &decode_attn);
  m.impl("rope_kv_append_qkv_", &rope_kv_append_qkv_);
  m.impl("decode_attn_qkv", &decode_attn_qkv);
  m.impl("decode_attn_rope_qkv", &decode_attn_rope_qkv);
  m.impl("prefill_attn_qkv", &prefill_attn_qkv);
  m.impl("prefill_attn", &prefill_attn);
  m.impl("sample", &sample);
  m.impl("skinny_linear", &skinny_linear);
  m.impl("skinny2_linear", &skinny2_linear);
  m.impl("skinny2_silu_linear", &skinny2_silu_linear);
  m.impl("stream_probe", &stream_probe);
  m.impl("prefill_paged_attn", &prefill_paged_attn);
}
