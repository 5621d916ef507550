"""Numerics tests: each gfx950 HIP kernel vs its plain-PyTorch fp32
reference (tolerances sized for bf16 I/O).  All gpu-marked."""

import math

import pytest
import torch

import resilient_llm_amd.ops as ops
from resilient_llm_amd.ops import ref

pytestmark = pytest.mark.gpu

DEV = "cuda"


def assert_close_bf16(a, b, atol=2e-2, rtol=2e-2, msg=""):
    af, bf = a.float(), b.float()
    torch.testing.assert_close(af, bf, atol=atol, rtol=rtol, msg=msg)


@pytest.fixture(autouse=True)
def _seed():
    torch.manual_seed(1234)


@pytest.mark.parametrize("rows,dim", [(1, 256), (64, 4096), (17, 4096),
                                      (8, 8192), (3, 16384)])
def test_rmsnorm(rows, dim):
    x = torch.randn(rows, dim, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(dim, dtype=torch.bfloat16, device=DEV)
    y = ops.rmsnorm(x, w, 1e-5)
    y_ref = ref.rmsnorm(x.cpu(), w.cpu(), 1e-5)
    assert_close_bf16(y.cpu(), y_ref)


def test_rmsnorm_residual_fused():
    x = torch.randn(32, 4096, dtype=torch.bfloat16, device=DEV)
    r = torch.randn(32, 4096, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(4096, dtype=torch.bfloat16, device=DEV)
    r_cpu = r.cpu().clone()
    x_cpu = x.cpu().clone()
    y = ops.rmsnorm_residual_(x, r, w, 1e-5)
    y_ref = ref.rmsnorm_residual_(x_cpu, r_cpu, w.cpu(), 1e-5)
    assert_close_bf16(y.cpu(), y_ref)
    assert_close_bf16(r.cpu(), r_cpu)   # updated residual matches


@pytest.mark.parametrize("rows,inter", [(4, 128), (64, 14336), (7, 1024)])
def test_silu_mul(rows, inter):
    gu = torch.randn(rows, 2 * inter, dtype=torch.bfloat16, device=DEV)
    y = ops.silu_mul(gu)
    y_ref = ref.silu_mul(gu.cpu())
    assert_close_bf16(y.cpu(), y_ref)


@pytest.mark.parametrize("n_q,n_kv,D", [(32, 8, 128), (8, 8, 128), (4, 2, 64)])
def test_rope_kv_append(n_q, n_kv, D):
    T, blocks, bs = 10, 8, 16
    q = torch.randn(T, n_q, D, dtype=torch.bfloat16, device=DEV)
    k = torch.randn(T, n_kv, D, dtype=torch.bfloat16, device=DEV)
    v = torch.randn(T, n_kv, D, dtype=torch.bfloat16, device=DEV)
    pos = torch.randint(0, 500, (T,), dtype=torch.int32, device=DEV)
    cs = ops.build_cos_sin(512, D, device=DEV)
    kc = torch.zeros(blocks, n_kv, bs, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.zeros_like(kc)
    slots = torch.arange(3, 3 + T, dtype=torch.int32, device=DEV) * 7 % (blocks * bs)
    q2, k2, v2 = q.cpu().clone(), k.cpu().clone(), v.cpu().clone()
    kc2, vc2 = kc.cpu().clone(), vc.cpu().clone()

    ops.rope_kv_append_(q, k, v, pos, cs, kc, vc, slots)
    ref.rope_kv_append_(q2, k2, v2, pos.cpu(), cs.cpu(), kc2, vc2, slots.cpu())
    assert_close_bf16(q.cpu(), q2)
    assert_close_bf16(k.cpu(), k2)
    assert_close_bf16(kc.cpu(), kc2)
    assert_close_bf16(vc.cpu(), vc2)


@pytest.mark.parametrize("batch,n_q,n_kv,D,lens", [
    (4, 32, 8, 128, [1, 17, 100, 256]),
    (2, 8, 8, 128, [33, 64]),
    (3, 16, 2, 64, [5, 90, 41]),
    (1, 8, 1, 128, [300]),
])
def test_decode_attn(batch, n_q, n_kv, D, lens):
    bs = 16
    max_blocks = (max(lens) + bs - 1) // bs + 1
    n_blocks_total = batch * max_blocks + 2
    kc = torch.randn(n_blocks_total, n_kv, bs, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.randn_like(kc)
    q = torch.randn(batch, n_q, D, dtype=torch.bfloat16, device=DEV)
    perm = torch.randperm(n_blocks_total, device=DEV)[:batch * max_blocks]
    bt = perm.reshape(batch, max_blocks).int().contiguous()
    seq_lens = torch.tensor(lens, dtype=torch.int32, device=DEV)
    scale = 1.0 / math.sqrt(D)
    out = ops.decode_attn(q, kc, vc, bt, seq_lens, scale)
    out_ref = ref.decode_attn(q.cpu(), kc.cpu(), vc.cpu(), bt.cpu(),
                              seq_lens.cpu(), scale)
    assert_close_bf16(out.cpu(), out_ref)


@pytest.mark.parametrize("n_q,n_kv,D,lens", [
    (32, 8, 128, [5, 128, 63]),
    (8, 8, 128, [200]),
    (16, 2, 64, [1, 300, 37]),
])
def test_prefill_attn(n_q, n_kv, D, lens):
    T = sum(lens)
    cu = torch.tensor([0] + list(torch.tensor(lens).cumsum(0)),
                      dtype=torch.int32, device=DEV)
    q = torch.randn(T, n_q, D, dtype=torch.bfloat16, device=DEV)
    k = torch.randn(T, n_kv, D, dtype=torch.bfloat16, device=DEV)
    v = torch.randn(T, n_kv, D, dtype=torch.bfloat16, device=DEV)
    scale = 1.0 / math.sqrt(D)
    out = ops.prefill_attn(q, k, v, cu, scale)
    out_ref = ref.prefill_attn(q.cpu(), k.cpu(), v.cpu(), cu.cpu(), scale)
    assert_close_bf16(out.cpu(), out_ref)


def _seeds(vals):
    return torch.tensor(vals, dtype=torch.int64, device=DEV)


def test_sample_greedy_matches_argmax():
    torch.manual_seed(7)
    logits = torch.randn(16, 128256, dtype=torch.bfloat16, device=DEV)
    temps = torch.zeros(16, device=DEV)
    toks = ops.sample(logits, temps, _seeds([1] * 16), 0)
    ref_toks = logits.float().argmax(-1).int()
    assert torch.equal(toks.cpu(), ref_toks.cpu())


def test_sample_temperature_distribution():
    """Gumbel-max sampling follows softmax(logits/T): chi-square-ish
    sanity on a 4-way categorical."""
    vocab = 8
    logits = torch.full((1, vocab), -1e4, dtype=torch.bfloat16, device=DEV)
    logits[0, :4] = torch.tensor([2.0, 1.0, 0.0, -1.0], dtype=torch.bfloat16)
    temps = torch.ones(1, device=DEV)
    counts = torch.zeros(vocab)
    n = 2000
    for s in range(n):
        t = ops.sample(logits, temps, _seeds([s]), 0)
        counts[t.item()] += 1
    probs = torch.softmax(logits[0, :4].float(), -1)
    emp = counts[:4] / n
    assert counts[4:].sum() == 0
    assert torch.allclose(emp, probs.cpu(), atol=0.05), (emp, probs)


def test_sample_deterministic_given_seed_and_step():
    logits = torch.randn(4, 1000, dtype=torch.bfloat16, device=DEV)
    temps = torch.full((4,), 0.8, device=DEV)
    a = ops.sample(logits, temps, _seeds([42, 42, 7, 7]), 3)
    b = ops.sample(logits, temps, _seeds([42, 42, 7, 7]), 3)
    c = ops.sample(logits, temps, _seeds([43, 42, 7, 7]), 3)
    d = ops.sample(logits, temps, _seeds([42, 42, 7, 7]), 4)
    assert torch.equal(a, b)
    assert a[0] == a[1] and a[2] == a[3]   # same seed+step -> same draw
    assert not torch.equal(a, c) or not torch.equal(a, d)


def test_fused_qkv_ops_match_unfused():
    """rope_kv_append_qkv_ / decode_attn_qkv / prefill_attn_qkv on strided
    slices of one qkv tensor match the unfused contiguous ops."""
    import resilient_llm_amd.ops as ops
    n_q, n_kv, D, T, bs = 8, 2, 128, 12, 16
    width = (n_q + 2 * n_kv) * D
    qkv = torch.randn(T, width, dtype=torch.bfloat16, device=DEV)
    qkv2 = qkv.clone()
    pos = torch.arange(T, dtype=torch.int32, device=DEV)
    cs = ops.build_cos_sin(64, D, device=DEV)
    kc = torch.zeros(4, n_kv, bs, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.zeros_like(kc)
    kc2, vc2 = kc.clone(), vc.clone()
    slots = torch.arange(T, dtype=torch.int32, device=DEV)

    # unfused path on contiguous copies
    q = qkv2[:, :n_q * D].reshape(T, n_q, D).contiguous()
    k = qkv2[:, n_q * D:(n_q + n_kv) * D].reshape(T, n_kv, D).contiguous()
    v = qkv2[:, (n_q + n_kv) * D:].reshape(T, n_kv, D).contiguous()
    ops.rope_kv_append_(q, k, v, pos, cs, kc2, vc2, slots)

    ops.rope_kv_append_qkv_(qkv, pos, cs, kc, vc, slots, n_q)
    torch.testing.assert_close(
        qkv[:, :n_q * D].reshape(T, n_q, D).float(), q.float())
    torch.testing.assert_close(kc.float(), kc2.float())
    torch.testing.assert_close(vc.float(), vc2.float())

    cu = torch.tensor([0, 5, T], dtype=torch.int32, device=DEV)
    scale = 0.088
    out_f = ops.prefill_attn_qkv(qkv, cu, scale, n_q, n_kv, D)
    out_u = ops.prefill_attn(q, k, v, cu, scale)
    # MFMA vs VALU kernels: same math, different f32 summation orders
    assert_close_bf16(out_f.reshape(T, n_q, D).cpu(), out_u.cpu())

    bt = torch.tensor([[0, 1], [2, 3]], dtype=torch.int32, device=DEV)
    lens = torch.tensor([5, 7], dtype=torch.int32, device=DEV)
    qkv_d = qkv[:2].contiguous()
    q_d = qkv_d[:, :n_q * D].reshape(2, n_q, D).contiguous()
    out_fd = ops.decode_attn_qkv(qkv_d, kc, vc, bt, lens, scale, n_q)
    out_ud = ops.decode_attn(q_d, kc, vc, bt, lens, scale)
    torch.testing.assert_close(out_fd.reshape(2, n_q, D).float(),
                               out_ud.float())


@pytest.mark.parametrize("M,N,K", [
    (64, 5120, 4096),    # 8B qkv
    (64, 4096, 4096),    # 8B o-proj
    (64, 28672, 4096),   # 8B gate_up
    (64, 4096, 14336),   # 8B down
    (64, 128256, 4096),  # 8B lm_head
    (1, 4096, 4096), (17, 4096, 4096), (33, 1024, 512), (48, 64, 256),
])
def test_skinny_linear_matches_blaslt(M, N, K):
    x = torch.randn(M, K, dtype=torch.bfloat16, device=DEV) * 0.1
    w = torch.randn(N, K, dtype=torch.bfloat16, device=DEV) * 0.1
    y = ops.skinny_linear(x, w)
    y_ref = torch.nn.functional.linear(x.float(), w.float())
    torch.testing.assert_close(y.float(), y_ref, atol=0.05, rtol=0.05)


@pytest.mark.parametrize("M,N,K", [
    (64, 6144, 4096),    # 8B qkv
    (64, 4096, 4096),    # 8B o-proj
    (64, 4096, 14336),   # 8B down
    (1, 4096, 4096), (17, 4096, 4096), (33, 1024, 512), (48, 64, 256),
])
@pytest.mark.parametrize("splitk", [1, 2, 4])
def test_skinny2_linear_matches_fp32(M, N, K, splitk):
    if splitk > K // 256:
        pytest.skip("splitk > K/256")
    x = torch.randn(M, K, dtype=torch.bfloat16, device=DEV) * 0.1
    w = torch.randn(N, K, dtype=torch.bfloat16, device=DEV) * 0.1
    y = ops.skinny2_linear(x, w, splitk)
    y_ref = torch.nn.functional.linear(x.float(), w.float())
    torch.testing.assert_close(y.float(), y_ref, atol=0.05, rtol=0.05)


@pytest.mark.parametrize("M,I,H", [
    (64, 14336, 4096),   # 8B down-proj geometry
    (13, 1024, 512),
    (64, 3584, 1024),    # 70B TP=8-ish shard geometry
])
@pytest.mark.parametrize("splitk", [1, 4])
def test_skinny2_silu_linear_matches_fp32(M, I, H, splitk):
    gu = torch.randn(M, 2 * I, dtype=torch.bfloat16, device=DEV) * 0.3
    w = torch.randn(H, I, dtype=torch.bfloat16, device=DEV) * 0.05
    y = ops.skinny2_silu_linear(gu, w, splitk)
    gf, uf = gu.float().chunk(2, -1)
    y_ref = (torch.nn.functional.silu(gf) * uf) @ w.float().t()
    torch.testing.assert_close(y.float(), y_ref, atol=0.08, rtol=0.08)


@pytest.mark.parametrize("n_q,n_kv,D,lens", [
    (32, 8, 128, [5, 128, 63, 200]),     # 8B GQA shape, ragged
    (8, 8, 128, [33]),                   # group 1
    (16, 8, 128, [100, 17]),             # group 2
    (16, 2, 128, [70, 31]),              # group 8 (two head-split WGs)
    (8, 2, 64, [129, 64, 1]),            # tiny D=64
])
def test_prefill_mfma_vs_valu_and_ref(n_q, n_kv, D, lens):
    """The MFMA fused prefill matches both the VALU v1 kernel and the
    fp32 CPU reference on ragged causal batches."""
    T = sum(lens)
    width = (n_q + 2 * n_kv) * D
    qkv = torch.randn(T, width, dtype=torch.bfloat16, device=DEV)
    cu = torch.tensor([0] + list(torch.tensor(lens).cumsum(0)),
                      dtype=torch.int32, device=DEV)
    scale = 1.0 / math.sqrt(D)
    q = qkv[:, :n_q * D].reshape(T, n_q, D).contiguous()
    k = qkv[:, n_q * D:(n_q + n_kv) * D].reshape(T, n_kv, D).contiguous()
    v = qkv[:, (n_q + n_kv) * D:].reshape(T, n_kv, D).contiguous()

    out_mfma = ops.prefill_attn_qkv(qkv, cu, scale, n_q, n_kv, D)
    out_valu = ops.prefill_attn(q, k, v, cu, scale)
    assert_close_bf16(out_mfma.reshape(T, n_q, D).cpu(), out_valu.cpu())
    out_ref = ref.prefill_attn(q.cpu(), k.cpu(), v.cpu(), cu.cpu(), scale)
    assert_close_bf16(out_mfma.reshape(T, n_q, D).cpu(), out_ref)


@pytest.mark.parametrize("n_q,n_kv,D", [(32, 8, 128), (16, 2, 128), (8, 2, 64)])
def test_prefill_paged_vs_ref(n_q, n_kv, D):
    """Paged chunked-prefill attention vs the fp32 reference: chunks of a
    sequence attending over cached prefix + chunk."""
    torch.manual_seed(3)
    bs = 16
    width = (n_q + 2 * n_kv) * D
    # sequence of 75 tokens split as chunks [0:40) and [40:75); cache
    # pre-filled for the whole 75 (rope_kv_append ran first in real flow)
    L = 75
    nb = (L + bs - 1) // bs
    kc = torch.randn(nb + 3, n_kv, bs, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.randn_like(kc)
    bt = torch.tensor([[4, 1, 3, 0, 2]], dtype=torch.int32, device=DEV)
    qkv = torch.randn(L, width, dtype=torch.bfloat16, device=DEV)

    def run(row0s, pos0s, nrowss, btrows):
        return ops.prefill_paged_attn(
            qkv, kc, vc,
            torch.tensor(row0s, dtype=torch.int32, device=DEV),
            torch.tensor(pos0s, dtype=torch.int32, device=DEV),
            torch.tensor(nrowss, dtype=torch.int32, device=DEV),
            torch.tensor(btrows, dtype=torch.int32, device=DEV),
            bt, 0.088, n_q)

    # engine splits chunks into <=32-row kernel chunks
    out = run([0, 32, 40, 72], [0, 32, 40, 72], [32, 8, 32, 3], [0, 0, 0, 0])
    out_ref = ref.prefill_paged_attn(
        qkv.cpu(), kc.cpu(), vc.cpu(),
        [0, 32, 40, 72], [0, 32, 40, 72], [32, 8, 32, 3], [0, 0, 0, 0],
        bt.cpu(), 0.088, n_q)
    assert_close_bf16(out.cpu(), out_ref)


@pytest.mark.parametrize("n_q,n_kv,D", [(32, 8, 128), (8, 8, 128), (16, 2, 128),
                                        (8, 2, 64)])
def test_decode_attn_rope_fused_vs_unfused(n_q, n_kv, D):
    """The fused rope+append+attention decode kernel matches the unfused
    (rope_kv_append_qkv_ then decode_attn_qkv) sequence: identical cache
    writes and attention output."""
    torch.manual_seed(11)
    B, bs = 5, 16
    lens = [1, 17, 33, 64, 160]           # ATTENTION lengths incl. self
    width = (n_q + 2 * n_kv) * D
    nbt = max((l + bs - 1) // bs for l in lens)
    nblocks = B * nbt + 2
    kc = torch.randn(nblocks, n_kv, bs, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.randn_like(kc)
    bt = torch.arange(B * nbt, dtype=torch.int32, device=DEV).reshape(B, nbt).contiguous()
    qkv = torch.randn(B, width, dtype=torch.bfloat16, device=DEV)
    pos = torch.tensor([l - 1 for l in lens], dtype=torch.int32, device=DEV)
    slots = torch.tensor([int(bt[i, (l - 1) // bs]) * bs + (l - 1) % bs
                          for i, l in enumerate(lens)],
                         dtype=torch.int32, device=DEV)
    seq_lens = torch.tensor(lens, dtype=torch.int32, device=DEV)
    cs = ops.build_cos_sin(512, D, device=DEV)
    scale = 1.0 / math.sqrt(D)

    qkv_u, kc_u, vc_u = qkv.clone(), kc.clone(), vc.clone()
    ops.rope_kv_append_qkv_(qkv_u, pos, cs, kc_u, vc_u, slots, n_q)
    out_u = ops.decode_attn_qkv(qkv_u, kc_u, vc_u, bt, seq_lens, scale, n_q)

    out_f = torch.ops.rlli.decode_attn_rope_qkv(
        qkv, pos, cs, kc, vc, slots, bt, seq_lens, scale, n_q)
    assert_close_bf16(out_f.cpu(), out_u.cpu())
    torch.testing.assert_close(kc.float(), kc_u.float())
    torch.testing.assert_close(vc.float(), vc_u.float())


@pytest.mark.gpu
def test_lt_linear_matches_torch():
    """Direct hipBLASLt linear with explicit algo index == F.linear
    (bf16 in, f32 accumulate — summation-order tolerance only)."""
    import resilient_llm_amd.ops as ops  # noqa: F401  (loads the library)
    for M, N, K in [(8, 4096, 4096), (64, 6144, 4096), (64, 4096, 14336)]:
        x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
        w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.02
        want = torch.nn.functional.linear(x, w).float()
        idxs = torch.ops.rlli.lt_heuristics(x, w, 4).tolist()
        assert idxs, "no hipblaslt heuristics returned"
        for idx in idxs[:3]:
            got = torch.ops.rlli.lt_linear(x, w, idx).float()
            torch.testing.assert_close(got, want, rtol=3e-2, atol=3e-1)


@pytest.mark.gpu
def test_tuned_linear_caches_and_matches(monkeypatch):
    from resilient_llm_amd.ops import autotune
    monkeypatch.setattr(autotune, "_DISABLED", False)   # opt-in lever
    x = torch.randn(16, 2048, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(1024, 2048, dtype=torch.bfloat16, device="cuda") * 0.02
    want = torch.nn.functional.linear(x, w).float()
    got = autotune.tuned_linear(x, w).float()
    torch.testing.assert_close(got, want, rtol=3e-2, atol=3e-1)
    key = (16, 1024, 2048)
    assert key in autotune.tuned_shapes()
    # second call hits the cache (either pinned algo or torch fallback)
    got2 = autotune.tuned_linear(x, w).float()
    torch.testing.assert_close(got2, want, rtol=3e-2, atol=3e-1)
