// Launch-function declarations: pure-HIP translation units (one per
// kernel family) export these; ext.cpp (the only file that includes
// torch headers) validates tensors and calls them.
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

namespace rlli {

// y = rmsnorm(x) * w           (residual == nullptr)
// r += x; y = rmsnorm(r) * w   (fused residual-add form)
void launch_rmsnorm(const uint16_t* x, uint16_t* residual, const uint16_t* w,
                    uint16_t* y, int rows, int dim, float eps,
                    hipStream_t stream);

// out[t, i] = silu(gu[t, i]) * gu[t, I + i]   for packed gate|up rows
void launch_silu_mul(const uint16_t* gate_up, uint16_t* out, int rows,
                     int inter, hipStream_t stream);

// In-place RoPE (NeoX-interleaved-halves style) on q and k, then scatter
// k/v into the paged cache at slot_mapping[t].
// q/k/v may be strided slices of one fused [T, (n_q+2*n_kv)*D] qkv
// tensor: *_stride is the per-token element stride of each view.
// Cache pointers are void*: the paged cache stores bf16 (default) or
// OCP fp8-e4m3 (cache_fp8) — see common.h cache codecs.
// FP8-activation emitters for the quantized-weight GEMM path
// (rowwise torch._scaled_mm consumers): normed / activated rows out as
// fp8-e4m3 + one dequant scale per row.
void launch_rmsnorm_fp8(const uint16_t* x, uint16_t* residual,
                        const uint16_t* w, uint8_t* y8, float* scales,
                        int rows, int dim, float eps, hipStream_t stream);
void launch_silu_mul_fp8(const uint16_t* gate_up, uint8_t* out8,
                         float* scales, int rows, int inter,
                         hipStream_t stream);

void launch_rope_kv_append(
    uint16_t* q, uint16_t* k, uint16_t* v,
    const int32_t* positions, const float* cos_sin,   // [max_pos, head_dim]
    void* k_cache, void* v_cache,
    const int32_t* slot_mapping,
    int tokens, int n_q_heads, int n_kv_heads, int head_dim,
    int block_size, int q_stride, int kv_stride, bool cache_fp8,
    hipStream_t stream);

// Split-K factor for decode attention (flash-decode): >1 when
// batch*n_kv_heads workgroups cannot fill 256 CUs.  The caller sizes
// the f32 scratch as [batch*n_kv*n_split*group*head_dim] (part_out)
// and [batch*n_kv*n_split*group*2] (part_ml) and passes both; n_split=1
// (or null scratch) is the classic single-workgroup-per-(seq,head) path.
int decode_attn_n_split(int batch, int n_kv_heads);

// Paged GQA decode attention: one new q token per sequence.
void launch_decode_attn(
    const uint16_t* q,                 // [batch, n_q_heads, head_dim]
    const void* k_cache,               // [blocks, n_kv, block_size, head_dim]
    const void* v_cache,
    const int32_t* block_table,        // [batch, max_blocks]
    const int32_t* seq_lens,           // [batch]
    uint16_t* out,                     // [batch, n_q_heads, head_dim]
    int batch, int n_q_heads, int n_kv_heads, int head_dim,
    int block_size, int max_blocks, float scale, int q_stride,
    int n_split, float* part_out, float* part_ml, bool cache_fp8,
    hipStream_t stream);

// Fused decode: rope(q,k) + cache append + paged attention in ONE
// kernel (decode path; qkv is the raw fused GEMM output).
void launch_decode_attn_fused(
    const uint16_t* qkv, void* k_cache, void* v_cache,
    const int32_t* block_table, const int32_t* seq_lens,
    const int32_t* positions, const float* cos_sin,
    const int32_t* slot_mapping, uint16_t* out, int batch, int n_q_heads,
    int n_kv_heads, int head_dim, int block_size, int max_blocks,
    float scale, int qkv_stride, int n_split, float* part_out,
    float* part_ml, bool cache_fp8, hipStream_t stream);

// Varlen causal prefill attention over in-batch q/k/v.
void launch_prefill_attn(
    const uint16_t* q, const uint16_t* k, const uint16_t* v,
    const int32_t* cu_seqlens,         // [n_seqs + 1]
    uint16_t* out,
    int n_seqs, int total_tokens, int n_q_heads, int n_kv_heads,
    int head_dim, float scale, int q_stride, int kv_stride,
    hipStream_t stream);

// MFMA varlen causal prefill attention over a fused qkv tensor.
void launch_prefill_mfma(const uint16_t* qkv, const int32_t* chunk_t0,
                         const int32_t* chunk_seq_start,
                         const int32_t* chunk_seq_end, uint16_t* out,
                         int n_chunks, int n_kv_heads, int group,
                         int head_dim, int qkv_stride, float scale,
                         hipStream_t stream);

// MFMA chunked-prefill attention over the paged cache (mixed batches).
void launch_prefill_paged(const uint16_t* qkv, const void* k_cache,
                          const void* v_cache, const int32_t* chunk_row0,
                          const int32_t* chunk_pos0,
                          const int32_t* chunk_nrows,
                          const int32_t* chunk_btrow,
                          const int32_t* block_tables, uint16_t* out,
                          int n_chunks, int n_kv_heads, int group,
                          int head_dim, int qkv_stride, int max_blocks,
                          int block_size, float scale, bool cache_fp8,
                          hipStream_t stream);

// Skinny-M GEMM (decode projections): out[M,N] = x[M,K] @ W[N,K]^T.
// ws is a [splitk, M, N] f32 workspace (unused when splitk == 1).
// v2: zero-LDS one-wave-per-16-columns register-dataflow skinny GEMM
// (skinny2.hip); ws is the [splitk, M, N] f32 slab when splitk > 1.
void launch_skinny2(const uint16_t* x, const uint16_t* w, float* ws,
                    uint16_t* out, int M, int N, int K, int splitk,
                    hipStream_t stream);
// silu(g)*u fused into the A-fragment path: gu is [M, 2K] (gate|up).
void launch_skinny2_silu(const uint16_t* gu, const uint16_t* w, float* ws,
                         uint16_t* out, int M, int N, int K, int splitk,
                         hipStream_t stream);

void launch_skinny_gemm(const uint16_t* x, const uint16_t* w, float* ws,
                        uint16_t* out, int M, int N, int K, int splitk,
                        hipStream_t stream);

// Diagnostic: pure nt-stream of W at skinny_gemm geometry.
void launch_stream_probe(const uint16_t* w, float* sink, int N, int K,
                         int splitk, hipStream_t stream);

// Fused sampling: greedy argmax when temperature[i] == 0, else Gumbel-max
// sampling of softmax(logits / temperature[i]) with an in-kernel counter
// hash RNG keyed on (seed, row, column) — no 32 MB random tensor per step.
int sample_n_split(int batch);       // phase-A split factor (1 = single)
void launch_sample(
    const uint16_t* logits,            // [batch, vocab] bf16
    const float* temperatures,         // [batch]
    const uint64_t* seeds,             // [batch] per-request seeds
    uint64_t step,                     // per-sequence step counter (mixed in)
    int32_t* out_tokens,               // [batch]
    int batch, int vocab,
    uint64_t* partials, int n_split,   // [batch, n_split] workspace
    hipStream_t stream);

}  // namespace rlli
