"""CPU sanity tests of the reference ops (these also ARE the CPU model
path, so they get their own correctness checks against torch built-ins)."""

import math

import torch

from resilient_llm_amd.ops import ref


def test_rmsnorm_matches_manual():
    x = torch.randn(4, 64)
    w = torch.randn(64)
    y = ref.rmsnorm(x, w, 1e-6)
    expect = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + 1e-6) * w
    torch.testing.assert_close(y, expect, atol=1e-5, rtol=1e-5)


def test_silu_mul_matches_manual():
    gu = torch.randn(3, 32)
    y = ref.silu_mul(gu)
    g, u = gu.chunk(2, -1)
    torch.testing.assert_close(y, torch.nn.functional.silu(g) * u,
                               atol=1e-6, rtol=1e-6)


def test_prefill_attn_matches_sdpa():
    T, n_q, n_kv, D = 24, 8, 2, 64
    lens = [10, 14]
    cu = torch.tensor([0, 10, 24], dtype=torch.int32)
    q = torch.randn(T, n_q, D)
    k = torch.randn(T, n_kv, D)
    v = torch.randn(T, n_kv, D)
    scale = 1.0 / math.sqrt(D)
    out = ref.prefill_attn(q, k, v, cu, scale)
    # per-seq dense sdpa with GQA expansion
    group = n_q // n_kv
    for s, (s0, L) in enumerate(zip([0, 10], lens)):
        qs = q[s0:s0 + L].transpose(0, 1)            # [n_q, L, D]
        ks = k[s0:s0 + L].repeat_interleave(group, dim=1).transpose(0, 1)
        vs = v[s0:s0 + L].repeat_interleave(group, dim=1).transpose(0, 1)
        expect = torch.nn.functional.scaled_dot_product_attention(
            qs, ks, vs, is_causal=True, scale=scale)
        torch.testing.assert_close(out[s0:s0 + L].transpose(0, 1), expect,
                                   atol=1e-4, rtol=1e-4)


def test_decode_consistent_with_prefill_last_token():
    """Decoding the last token of a prompt must equal the prefill output
    at that position (same math, paged vs contiguous layouts)."""
    torch.manual_seed(0)
    L, n_q, n_kv, D, bs = 37, 8, 2, 64, 16
    q_all = torch.randn(L, n_q, D)
    k_all = torch.randn(L, n_kv, D)
    v_all = torch.randn(L, n_kv, D)
    cu = torch.tensor([0, L], dtype=torch.int32)
    scale = 1.0 / math.sqrt(D)
    pre = ref.prefill_attn(q_all, k_all, v_all, cu, scale)

    n_blocks = (L + bs - 1) // bs
    kc = torch.zeros(n_blocks + 2, n_kv, bs, D)
    vc = torch.zeros_like(kc)
    bt = torch.tensor([[2, 0, 4][i % 3] if False else i for i in range(n_blocks)],
                      dtype=torch.int32).unsqueeze(0)
    for t in range(L):
        blk, row = t // bs, t % bs
        kc[bt[0, blk], :, row] = k_all[t]
        vc[bt[0, blk], :, row] = v_all[t]
    out = ref.decode_attn(q_all[-1:].clone(), kc, vc, bt,
                          torch.tensor([L], dtype=torch.int32), scale)
    torch.testing.assert_close(out[0], pre[-1], atol=1e-4, rtol=1e-4)


def test_rope_kv_roundtrip():
    T, n_q, n_kv, D, bs = 6, 4, 2, 64, 16
    q = torch.randn(T, n_q, D)
    k = torch.randn(T, n_kv, D)
    v = torch.randn(T, n_kv, D)
    pos = torch.arange(T, dtype=torch.int32)
    cs = ref.build_cos_sin(32, D)
    kc = torch.zeros(2, n_kv, bs, D)
    vc = torch.zeros_like(kc)
    slots = torch.arange(T, dtype=torch.int32)
    k_before = k.clone()
    ref.rope_kv_append_(q, k, v, pos, cs, kc, vc, slots)
    # position 0 rotation is identity
    torch.testing.assert_close(k[0], k_before[0], atol=1e-5, rtol=1e-5)
    # cache rows hold the rotated k and plain v
    torch.testing.assert_close(kc[0, :, :T].transpose(0, 1), k)
    torch.testing.assert_close(vc[0, :, :T].transpose(0, 1), v)
    # rotation preserves pair norms
    half = D // 2
    before = k_before[..., :half] ** 2 + k_before[..., half:] ** 2
    after = k[..., :half] ** 2 + k[..., half:] ** 2
    torch.testing.assert_close(before, after, atol=1e-4, rtol=1e-4)


def test_sample_greedy():
    logits = torch.randn(5, 100)
    out = ref.sample(logits.bfloat16(), torch.zeros(5),
                     torch.zeros(5, dtype=torch.int64))
    assert torch.equal(out.long(), logits.bfloat16().float().argmax(-1))
