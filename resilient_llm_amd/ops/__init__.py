"""Op dispatch: hand-written gfx950 HIP kernels on GPU, fp32 torch
references on CPU.

On a CUDA/ROCm device the HIP extension is REQUIRED — a missing .so
raises immediately rather than silently falling back to eager PyTorch
(the round-end check records which native libraries the GPU tests
actually loaded).  On CPU the ops/ref.py implementations run, which is
what the no-GPU plumbing tests and the gloo multi-process tests use.
"""

from __future__ import annotations

import os
import torch

from . import ref  # noqa: F401

_SO_PATH = os.path.join(os.path.dirname(os.path.abspath(__file__)), "_rlli_hip.so")
_loaded = False


class ExtensionMissing(RuntimeError):
    pass


def load_extension(required: bool = False) -> bool:
    """Load the in-tree .so (idempotent).  ``required=True`` (or any GPU
    call) raises if it is absent — no silent eager fallback on GPU."""
    global _loaded
    if _loaded:
        return True
    if os.path.exists(_SO_PATH):
        torch.ops.load_library(_SO_PATH)
        _loaded = True
        return True
    if required:
        raise ExtensionMissing(
            f"HIP extension not built: {_SO_PATH} missing. "
            f"Run `python -m resilient_llm_amd.ops.build`.")
    return False


def extension_loaded() -> bool:
    return _loaded


def _gpu() -> None:
    load_extension(required=True)


# ------------------------------------------------------------------ ops
def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    if x.is_cuda:
        _gpu()
        return torch.ops.rlli.rmsnorm(x, w, eps)
    return ref.rmsnorm(x, w, eps)


def rmsnorm_residual_(x: torch.Tensor, residual: torch.Tensor, w: torch.Tensor,
                      eps: float = 1e-5) -> torch.Tensor:
    if x.is_cuda:
        _gpu()
        return torch.ops.rlli.rmsnorm_residual_(x, residual, w, eps)
    return ref.rmsnorm_residual_(x, residual, w, eps)


def rmsnorm_fp8(x, w, eps: float = 1e-5):
    """Fused rmsnorm + per-row fp8-e4m3 quantization -> (q, scales)."""
    if x.is_cuda:
        _gpu()
        return torch.ops.rlli.rmsnorm_fp8(x, w, eps)
    return ref.rmsnorm_fp8(x, w, eps)


def rmsnorm_residual_fp8(x, residual, w, eps: float = 1e-5):
    if x.is_cuda:
        _gpu()
        return torch.ops.rlli.rmsnorm_residual_fp8(x, residual, w, eps)
    return ref.rmsnorm_residual_fp8(x, residual, w, eps)


def silu_mul_fp8(gate_up):
    if gate_up.is_cuda:
        _gpu()
        return torch.ops.rlli.silu_mul_fp8(gate_up)
    return ref.silu_mul_fp8(gate_up)


def silu_mul(gate_up: torch.Tensor) -> torch.Tensor:
    if gate_up.is_cuda:
        _gpu()
        return torch.ops.rlli.silu_mul(gate_up)
    return ref.silu_mul(gate_up)


def rope_kv_append_(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                    positions: torch.Tensor, cos_sin: torch.Tensor,
                    k_cache: torch.Tensor, v_cache: torch.Tensor,
                    slot_mapping: torch.Tensor) -> None:
    if q.is_cuda:
        _gpu()
        torch.ops.rlli.rope_kv_append_(q, k, v, positions, cos_sin,
                                       k_cache, v_cache, slot_mapping)
    else:
        ref.rope_kv_append_(q, k, v, positions, cos_sin, k_cache, v_cache,
                            slot_mapping)


def rope_kv_append_qkv_(qkv: torch.Tensor, positions: torch.Tensor,
                        cos_sin: torch.Tensor, k_cache: torch.Tensor,
                        v_cache: torch.Tensor, slot_mapping: torch.Tensor,
                        n_q: int) -> None:
    """Fused-QKV form: rope + cache append on strided views of the raw
    qkv GEMM output (no .contiguous() copies)."""
    if qkv.is_cuda:
        _gpu()
        torch.ops.rlli.rope_kv_append_qkv_(qkv, positions, cos_sin, k_cache,
                                           v_cache, slot_mapping, n_q)
        return
    T = qkv.shape[0]
    n_kv, _, D = k_cache.shape[1], k_cache.shape[2], k_cache.shape[3]
    q = qkv[:, :n_q * D].view(T, n_q, D)
    k = qkv[:, n_q * D:(n_q + n_kv) * D].view(T, n_kv, D)
    v = qkv[:, (n_q + n_kv) * D:].view(T, n_kv, D)
    ref.rope_kv_append_(q, k, v, positions, cos_sin, k_cache, v_cache,
                        slot_mapping)


def decode_attn_qkv(qkv: torch.Tensor, k_cache: torch.Tensor,
                    v_cache: torch.Tensor, block_table: torch.Tensor,
                    seq_lens: torch.Tensor, scale: float,
                    n_q: int) -> torch.Tensor:
    if qkv.is_cuda:
        _gpu()
        return torch.ops.rlli.decode_attn_qkv(qkv, k_cache, v_cache,
                                              block_table, seq_lens, scale, n_q)
    B = qkv.shape[0]
    D = k_cache.shape[3]
    q = qkv[:, :n_q * D].reshape(B, n_q, D)
    out = ref.decode_attn(q, k_cache, v_cache, block_table, seq_lens, scale)
    return out.reshape(B, n_q * D)


def decode_attn_rope_qkv(qkv, positions, cos_sin, k_cache, v_cache,
                         slot_mapping, block_table, seq_lens, scale: float,
                         n_q: int) -> torch.Tensor:
    """Fused decode step: rope(q,k) + cache append + paged attention in
    one kernel on GPU; composed from the unfused reference ops on CPU."""
    if qkv.is_cuda:
        _gpu()
        return torch.ops.rlli.decode_attn_rope_qkv(
            qkv, positions, cos_sin, k_cache, v_cache, slot_mapping,
            block_table, seq_lens, scale, n_q)
    rope_kv_append_qkv_(qkv, positions, cos_sin, k_cache, v_cache,
                        slot_mapping, n_q)
    return decode_attn_qkv(qkv, k_cache, v_cache, block_table, seq_lens,
                           scale, n_q)


def prefill_attn_qkv(qkv: torch.Tensor, cu_seqlens: torch.Tensor,
                     scale: float, n_q: int, n_kv: int,
                     head_dim: int) -> torch.Tensor:
    if qkv.is_cuda:
        _gpu()
        return torch.ops.rlli.prefill_attn_qkv(qkv, cu_seqlens, scale, n_q,
                                               n_kv, head_dim)
    T = qkv.shape[0]
    D = head_dim
    q = qkv[:, :n_q * D].reshape(T, n_q, D)
    k = qkv[:, n_q * D:(n_q + n_kv) * D].reshape(T, n_kv, D)
    v = qkv[:, (n_q + n_kv) * D:].reshape(T, n_kv, D)
    return ref.prefill_attn(q, k, v, cu_seqlens, scale).reshape(T, n_q * D)


def decode_attn(q: torch.Tensor, k_cache: torch.Tensor, v_cache: torch.Tensor,
                block_table: torch.Tensor, seq_lens: torch.Tensor,
                scale: float) -> torch.Tensor:
    if q.is_cuda:
        _gpu()
        return torch.ops.rlli.decode_attn(q, k_cache, v_cache, block_table,
                                          seq_lens, scale)
    return ref.decode_attn(q, k_cache, v_cache, block_table, seq_lens, scale)


def prefill_attn(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                 cu_seqlens: torch.Tensor, scale: float) -> torch.Tensor:
    if q.is_cuda:
        _gpu()
        return torch.ops.rlli.prefill_attn(q, k, v, cu_seqlens, scale)
    return ref.prefill_attn(q, k, v, cu_seqlens, scale)


def sample(logits: torch.Tensor, temperatures: torch.Tensor,
           seeds: torch.Tensor, step: int = 0) -> torch.Tensor:
    """Greedy argmax / Gumbel-max temperature sampling; ``seeds`` is a
    per-row int64 tensor (per-request reproducibility), ``step`` a
    uniform step offset mixed in-kernel."""
    if logits.is_cuda:
        _gpu()
        return torch.ops.rlli.sample(logits, temperatures, seeds, step)
    return ref.sample(logits, temperatures, seeds, step)


def prefill_paged_attn(qkv, k_cache, v_cache, chunk_row0, chunk_pos0,
                       chunk_nrows, chunk_btrow, block_tables,
                       scale: float, n_q: int) -> torch.Tensor:
    if qkv.is_cuda:
        _gpu()
        return torch.ops.rlli.prefill_paged_attn(
            qkv, k_cache, v_cache, chunk_row0, chunk_pos0, chunk_nrows,
            chunk_btrow, block_tables, scale, n_q)
    return ref.prefill_paged_attn(qkv, k_cache, v_cache, chunk_row0,
                                  chunk_pos0, chunk_nrows, chunk_btrow,
                                  block_tables, scale, n_q)


def skinny_linear(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """Decode-projection GEMM (M<=64): custom MFMA weight-streaming
    kernel on GPU, F.linear on CPU."""
    if x.is_cuda:
        _gpu()
        return torch.ops.rlli.skinny_linear(x, w)
    return torch.nn.functional.linear(x, w)


def linear_auto(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """Pick the skinny-M kernel when the shape qualifies, else
    hipBLASLt via F.linear."""
    if (x.is_cuda and x.dim() == 2 and 1 <= x.shape[0] <= 64
            and x.shape[1] % 256 == 0 and w.shape[0] % 64 == 0):
        _gpu()
        return torch.ops.rlli.skinny_linear(x, w)
    return torch.nn.functional.linear(x, w)


def skinny2_linear(x: torch.Tensor, w: torch.Tensor,
                   splitk: int = 0) -> torch.Tensor:
    """v2 zero-LDS skinny GEMM (M <= 64); splitk <= 0 = auto."""
    if x.is_cuda:
        _gpu()
        return torch.ops.rlli.skinny2_linear(x, w, splitk)
    return torch.nn.functional.linear(x, w)


def skinny2_silu_linear(gu: torch.Tensor, w: torch.Tensor,
                        splitk: int = 0) -> torch.Tensor:
    """Fused SwiGLU + down-proj: (silu(gu[:, :K]) * gu[:, K:]) @ w^T with
    the activation computed inside the GEMM's A-fragment loads — deletes
    the silu_mul launch and its intermediate tensor."""
    if gu.is_cuda:
        _gpu()
        return torch.ops.rlli.skinny2_silu_linear(gu, w, splitk)
    return torch.nn.functional.linear(ref.silu_mul(gu), w)


build_cos_sin = ref.build_cos_sin
