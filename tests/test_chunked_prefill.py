"""Chunked prefill + mixed prefill/decode batches: outputs must be
IDENTICAL to the unchunked engine (greedy, fp32 CPU refs)."""

import torch

from resilient_llm_amd.engine import LLMEngine, PagedKVCache, SamplingParams
from resilient_llm_amd.models import LlamaForCausalLM, get_config


def make_engine(chunk_size=2048, num_blocks=128, **kw):
    cfg = get_config("tiny")
    model = LlamaForCausalLM(cfg, device="cpu", dtype=torch.float32, seed=7)
    kv = PagedKVCache.for_model(cfg, num_blocks, device="cpu")
    kv.k = kv.k.float()
    kv.v = kv.v.float()
    return LLMEngine(model, kv, chunk_size=chunk_size, **kw)


def drain(engine, max_steps=600):
    outs = {}
    for _ in range(max_steps):
        if not engine.has_work():
            break
        for o in engine.step():
            outs.setdefault(o.req_id, []).append(o.token_id)
    assert not engine.has_work()
    return outs


def test_chunked_equals_unchunked():
    prompt = list(range(10, 93))   # 83 tokens, ragged vs chunk 8 and 32
    big = make_engine(chunk_size=2048)
    big.add_request("a", prompt, SamplingParams(max_tokens=7))
    want = drain(big)["a"]
    for cs in (8, 32, 50):
        e = make_engine(chunk_size=cs)
        e.add_request("a", prompt, SamplingParams(max_tokens=7))
        got = drain(e)["a"]
        assert got == want, (cs, got, want)


def test_mixed_decode_unperturbed_by_prefill():
    """A decoding sequence produces the same greedy tokens whether or not
    a long prompt prefills (in chunks) beside it."""
    p_a = list(range(5, 45))
    solo = make_engine()
    solo.add_request("a", p_a, SamplingParams(max_tokens=12))
    want = drain(solo)["a"]

    mixed = make_engine(chunk_size=16)
    mixed.add_request("a", p_a, SamplingParams(max_tokens=12))
    early = mixed.step() + mixed.step() + mixed.step()
    mixed.add_request("b", list(range(100, 170)), SamplingParams(max_tokens=4))
    outs = drain(mixed)
    got = [o.token_id for o in early if o.req_id == "a"] + outs.get("a", [])
    assert got == want
    assert len(outs["b"]) == 4


def test_budget_spreads_prefill_across_steps():
    e = make_engine(chunk_size=16, max_prefill_tokens=16)
    e.add_request("a", list(range(60)), SamplingParams(max_tokens=3))
    steps_until_first_token = 0
    outs = []
    while not outs and steps_until_first_token < 20:
        outs = e.step()
        steps_until_first_token += 1
    assert steps_until_first_token == 4   # ceil(60/16) chunks
    drain(e)


def test_many_seqs_chunked_concurrently():
    e = make_engine(chunk_size=32, max_prefill_tokens=64, num_blocks=256,
                    max_batch_size=8)
    for i in range(6):
        e.add_request(f"r{i}", list(range(3 + i, 60 + 2 * i)),
                      SamplingParams(max_tokens=5))
    outs = drain(e)
    assert len(outs) == 6 and all(len(v) == 5 for v in outs.values())


def test_latency_targeted_budget_shrinks_and_regrows():
    """AIMD scheduler (target_step_ms): an impossible SLO collapses the
    per-step prefill budget to the floor; a generous SLO grows it back
    to max_prefill_tokens.  Output tokens must be unaffected (the budget
    only re-slices chunks across steps)."""
    prompt = list(range(10, 93))
    ref = make_engine(chunk_size=2048)
    ref.add_request("a", prompt, SamplingParams(max_tokens=7))
    want = drain(ref)["a"]

    # CPU steps take milliseconds >> 1e-6 ms -> every mixed step shrinks
    e = make_engine(chunk_size=8, max_prefill_tokens=1024,
                    target_step_ms=1e-6)
    e.add_request("a", prompt, SamplingParams(max_tokens=7))
    got = drain(e)["a"]
    assert got == want
    assert e._prefill_budget == e._budget_floor

    # generous SLO: budget climbs back (additive increase per step)
    e.target_step_ms = 1e9
    for _ in range(600):
        if e._prefill_budget >= e.max_prefill_tokens:
            break
        e.add_request(f"r{_}", prompt[:16], SamplingParams(max_tokens=1))
        drain(e)
    assert e._prefill_budget == e.max_prefill_tokens


def test_budget_never_starves_prefill():
    """Even at the floor, at least one chunk advances per step — a
    too-tight SLO degrades throughput, never liveness."""
    e = make_engine(chunk_size=8, max_prefill_tokens=1024,
                    target_step_ms=1e-6)
    for i in range(6):
        e.add_request(f"s{i}", list(range(3, 70)),
                      SamplingParams(max_tokens=3))
    outs = drain(e)
    assert len(outs) == 6
    assert all(len(v) == 3 for v in outs.values())


def test_abort_mid_chunked_prefill_frees_blocks():
    """Aborting while a long prompt is mid-chunk returns ALL its blocks
    (reserved at admission) and other requests are unaffected."""
    e = make_engine(chunk_size=8, num_blocks=128)
    free0 = e.kv.free_blocks
    e.add_request("long", list(range(3, 120)), SamplingParams(max_tokens=4))
    e.add_request("other", list(range(7, 30)), SamplingParams(max_tokens=3))
    e.step()                               # admit + first chunks
    assert e.kv.free_blocks < free0
    e.abort("long")
    outs = drain(e)
    assert "long" not in outs
    assert len(outs["other"]) == 3
    assert e.kv.free_blocks == free0       # everything returned


def test_abort_waiting_request_never_allocates():
    e = make_engine(chunk_size=8, num_blocks=128)
    free0 = e.kv.free_blocks
    e.add_request("w", list(range(3, 40)), SamplingParams(max_tokens=2))
    e.abort("w")                           # aborted while still queued
    assert drain(e) == {}
    assert e.kv.free_blocks == free0
