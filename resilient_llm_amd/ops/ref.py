"""Pure-PyTorch fp32 reference implementations of every HIP op.

These are (a) the ground truth the GPU numerics tests compare the gfx950
kernels against (SURVEY.md §4 implication (b)) and (b) the CPU execution
path for the plumbing tests — the model runs end-to-end on CPU through
these exact functions.  Semantics match the kernels bit-for-bit where the
kernels round (bf16 storage boundaries), computed in fp32.
"""

from __future__ import annotations

import torch


def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
    xf = x.float()
    inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * inv * w.float()).to(x.dtype)


def rmsnorm_residual_(x: torch.Tensor, residual: torch.Tensor, w: torch.Tensor,
                      eps: float) -> torch.Tensor:
    """residual += x (stored at residual dtype); y = rmsnorm(residual)."""
    summed = (x.float() + residual.float()).to(residual.dtype)
    residual.copy_(summed)
    return rmsnorm(residual, w, eps)


def silu_mul(gate_up: torch.Tensor) -> torch.Tensor:
    inter = gate_up.shape[-1] // 2
    g = gate_up[..., :inter].float()
    u = gate_up[..., inter:].float()
    return (torch.nn.functional.silu(g) * u).to(gate_up.dtype)


def build_cos_sin(max_pos: int, head_dim: int, theta: float = 500000.0,
                  device="cpu") -> torch.Tensor:
    """[max_pos, head_dim] f32: row = [cos(0..D/2) | sin(0..D/2)]."""
    half = head_dim // 2
    inv_freq = 1.0 / (theta ** (torch.arange(half, dtype=torch.float64) / half))
    pos = torch.arange(max_pos, dtype=torch.float64)
    ang = torch.outer(pos, inv_freq)
    return torch.cat([ang.cos(), ang.sin()], dim=-1).float().to(device)


def _rotate(x: torch.Tensor, positions: torch.Tensor,
            cos_sin: torch.Tensor) -> torch.Tensor:
    """GPT-NeoX half rotation; x [T, H, D]."""
    D = x.shape[-1]
    half = D // 2
    cs = cos_sin[positions.long()]                  # [T, D]
    cos = cs[:, :half].unsqueeze(1)                 # [T, 1, half]
    sin = cs[:, half:].unsqueeze(1)
    xf = x.float()
    x1, x2 = xf[..., :half], xf[..., half:]
    out = torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin], dim=-1)
    return out.to(x.dtype)


def rope_kv_append_(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                    positions: torch.Tensor, cos_sin: torch.Tensor,
                    k_cache: torch.Tensor, v_cache: torch.Tensor,
                    slot_mapping: torch.Tensor) -> None:
    q.copy_(_rotate(q, positions, cos_sin))
    k.copy_(_rotate(k, positions, cos_sin))
    block_size = k_cache.shape[2]
    slots = slot_mapping.long()
    valid = slots >= 0
    idx = slots[valid]
    blocks = idx // block_size
    rows = idx % block_size
    # cache: [num_blocks, n_kv, block_size, D]; k/v: [T, n_kv, D]
    # cast through the cache dtype (bf16 identity; fp8-e4m3 quantizes —
    # mirrors the GPU codec's cvt_pk_fp8 RNE conversion)
    k_cache[blocks, :, rows] = k[valid].to(k_cache.dtype)
    v_cache[blocks, :, rows] = v[valid].to(v_cache.dtype)


def decode_attn(q: torch.Tensor, k_cache: torch.Tensor, v_cache: torch.Tensor,
                block_table: torch.Tensor, seq_lens: torch.Tensor,
                scale: float) -> torch.Tensor:
    batch, n_q, D = q.shape
    n_kv = k_cache.shape[1]
    block_size = k_cache.shape[2]
    group = n_q // n_kv
    out = torch.empty_like(q)
    for b in range(batch):
        L = int(seq_lens[b])
        n_blocks = (L + block_size - 1) // block_size
        blocks = block_table[b, :n_blocks].long()
        k = k_cache[blocks].transpose(0, 1).reshape(n_kv, -1, D)[:, :L].float()
        v = v_cache[blocks].transpose(0, 1).reshape(n_kv, -1, D)[:, :L].float()
        qb = q[b].float().view(n_kv, group, D)
        s = torch.einsum("hgd,htd->hgt", qb, k) * scale
        p = torch.softmax(s, dim=-1)
        o = torch.einsum("hgt,htd->hgd", p, v)
        out[b] = o.reshape(n_q, D).to(q.dtype)
    return out


def prefill_attn(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                 cu_seqlens: torch.Tensor, scale: float) -> torch.Tensor:
    T, n_q, D = q.shape
    n_kv = k.shape[1]
    group = n_q // n_kv
    out = torch.empty_like(q)
    cu = cu_seqlens.tolist()
    for s in range(len(cu) - 1):
        s0, s1 = cu[s], cu[s + 1]
        L = s1 - s0
        qs = q[s0:s1].float().view(L, n_kv, group, D)
        ks = k[s0:s1].float()
        vs = v[s0:s1].float()
        scores = torch.einsum("qhgd,thd->hgqt", qs, ks) * scale
        mask = torch.triu(torch.ones(L, L, dtype=torch.bool, device=q.device), 1)
        scores.masked_fill_(mask, float("-inf"))
        p = torch.softmax(scores, dim=-1)
        o = torch.einsum("hgqt,thd->qhgd", p, vs)
        out[s0:s1] = o.reshape(L, n_q, D).to(q.dtype)
    return out


def prefill_paged_attn(qkv: torch.Tensor, k_cache: torch.Tensor,
                       v_cache: torch.Tensor, chunk_row0, chunk_pos0,
                       chunk_nrows, chunk_btrow, block_tables,
                       scale: float, n_q: int) -> torch.Tensor:
    """Chunked prefill over the paged cache: each chunk's q rows attend
    causally over their sequence's cached prefix (which includes the
    chunk itself — rope_kv_append ran first)."""
    T = qkv.shape[0]
    n_kv, block_size, D = k_cache.shape[1], k_cache.shape[2], k_cache.shape[3]
    group = n_q // n_kv
    out = torch.zeros(T, n_q * D, dtype=qkv.dtype)
    for c in range(len(chunk_row0)):
        row0 = int(chunk_row0[c]); pos0 = int(chunk_pos0[c])
        nrows = int(chunk_nrows[c]); btr = int(chunk_btrow[c])
        kv_hi = pos0 + nrows
        n_blocks = (kv_hi + block_size - 1) // block_size
        blocks = block_tables[btr, :n_blocks].long()
        k = k_cache[blocks].transpose(0, 1).reshape(n_kv, -1, D)[:, :kv_hi].float()
        v = v_cache[blocks].transpose(0, 1).reshape(n_kv, -1, D)[:, :kv_hi].float()
        q = qkv[row0:row0 + nrows, :n_q * D].reshape(nrows, n_kv, group, D).float()
        s = torch.einsum("qhgd,htd->hgqt", q, k) * scale
        qpos = torch.arange(pos0, pos0 + nrows).unsqueeze(1)
        kvpos = torch.arange(kv_hi).unsqueeze(0)
        s.masked_fill_((kvpos > qpos), float("-inf"))
        p = torch.softmax(s, dim=-1)
        o = torch.einsum("hgqt,htd->qhgd", p, v)
        out[row0:row0 + nrows] = o.reshape(nrows, n_q * D).to(qkv.dtype)
    return out


def sample(logits: torch.Tensor, temperatures: torch.Tensor,
           seeds: torch.Tensor, step: int = 0) -> torch.Tensor:
    """Greedy rows match the kernel exactly; stochastic rows use torch
    RNG seeded per row from (seeds[i], step) — same reproducibility
    contract as the kernel, not bitwise the same stream."""
    out = torch.empty(logits.shape[0], dtype=torch.int32, device=logits.device)
    lf = logits.float()
    greedy = temperatures <= 0
    if greedy.any():
        out[greedy] = lf[greedy].argmax(-1).to(torch.int32)
    C = 0x9e3779b97f4a7c15
    for i in (~greedy).nonzero(as_tuple=True)[0].tolist():
        gen = torch.Generator(device=logits.device)
        gen.manual_seed((int(seeds[i]) + C * step) & 0x7fffffffffffffff)
        probs = torch.softmax(lf[i] / float(temperatures[i]), dim=-1)
        out[i] = int(torch.multinomial(probs, 1, generator=gen))
    return out


# ---- fp8-activation emitters (quantized-weight GEMM path) ----
def quantize_fp8_rowwise(t: torch.Tensor):
    """Per-row fp8-e4m3 quantization: (q, scales) with q*scales ~= t."""
    f = t.float()
    amax = f.abs().amax(dim=-1).clamp_min(1e-12)
    scales = amax / 448.0
    q = (f / scales.unsqueeze(-1)).to(torch.float8_e4m3fn)
    return q, scales


def rmsnorm_fp8(x, w, eps=1e-5):
    return quantize_fp8_rowwise(rmsnorm(x, w, eps))


def rmsnorm_residual_fp8(x, residual, w, eps=1e-5):
    return quantize_fp8_rowwise(rmsnorm_residual_(x, residual, w, eps))


def silu_mul_fp8(gate_up):
    return quantize_fp8_rowwise(silu_mul(gate_up))


def scaled_mm_ref(x8, xs, w8, ws):
    """Dequantized reference of a rowwise torch._scaled_mm: x8 [M,K] fp8
    with per-row scales xs [M], w8 [N,K] fp8 with per-row (out-channel)
    scales ws [N]; bf16 result.  The CPU engine path uses this (no
    hipBLASLt fp8 GEMM off-GPU)."""
    xf = x8.float() * xs.reshape(-1, 1)
    wf = w8.float() * ws.reshape(-1, 1)
    return (xf @ wf.t()).to(torch.bfloat16)
