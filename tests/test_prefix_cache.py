"""Prefix caching: shared prompt prefixes reuse cached KV blocks with
identical outputs, refcounts protect shared blocks, eviction reclaims."""

import torch

from resilient_llm_amd.engine import LLMEngine, PagedKVCache, SamplingParams
from resilient_llm_amd.engine.kv_cache import block_hash_chain
from resilient_llm_amd.models import LlamaForCausalLM, get_config


def make_engine(num_blocks=96, prefix=True, **kw):
    cfg = get_config("tiny")
    model = LlamaForCausalLM(cfg, device="cpu", dtype=torch.float32, seed=3)
    kv = PagedKVCache.for_model(cfg, num_blocks, device="cpu")
    kv.k = kv.k.float()
    kv.v = kv.v.float()
    return LLMEngine(model, kv, enable_prefix_caching=prefix, **kw)


def drain(e, max_steps=500):
    outs = {}
    while e.has_work() and max_steps:
        max_steps -= 1
        for o in e.step():
            outs.setdefault(o.req_id, []).append(o.token_id)
    assert not e.has_work()
    return outs


def test_hash_chain_properties():
    a = block_hash_chain(list(range(40)), 16)
    b = block_hash_chain(list(range(40)), 16)
    c = block_hash_chain(list(range(1, 41)), 16)
    assert a == b and len(a) == 2
    assert a[0] != c[0]
    # chained: same 2nd block tokens but different 1st -> different key
    d = block_hash_chain(list(range(16, 48)), 16)
    assert d[0] != a[1]


def test_repeat_prompt_hits_cache_same_output():
    prompt = list(range(7, 60))            # 53 tokens -> 3 full blocks
    base = make_engine(prefix=False)
    base.add_request("a", prompt, SamplingParams(max_tokens=6))
    want = drain(base)["a"]

    e = make_engine(prefix=True)
    e.add_request("a", prompt, SamplingParams(max_tokens=6))
    first = drain(e)["a"]
    assert first == want
    hits0 = e.kv.prefix_hits
    e.add_request("b", prompt, SamplingParams(max_tokens=6))
    second = drain(e)["b"]
    assert second == want
    assert e.kv.prefix_hits - hits0 == 3   # reused all 3 full blocks


def test_partial_prefix_share():
    p1 = list(range(100, 164))             # 64 tokens, 4 full blocks
    p2 = p1[:48] + list(range(400, 430))   # shares 3 blocks then diverges
    base = make_engine(prefix=False)
    base.add_request("x", p2, SamplingParams(max_tokens=5))
    want = drain(base)["x"]

    e = make_engine(prefix=True)
    e.add_request("x1", p1, SamplingParams(max_tokens=5))
    drain(e)
    e.add_request("x2", p2, SamplingParams(max_tokens=5))
    got = drain(e)["x2"]
    assert got == want
    assert e.kv.prefix_hits >= 3


def test_refcount_shared_blocks_survive_finisher():
    e = make_engine(prefix=True, max_batch_size=4)
    prompt = list(range(5, 40))
    e.add_request("a", prompt, SamplingParams(max_tokens=30))
    outs = {}
    for o in e.step():                     # prefill a (first token emits)
        outs.setdefault(o.req_id, []).append(o.token_id)
    e.add_request("b", prompt, SamplingParams(max_tokens=2))
    while e.has_work():
        for o in e.step():
            outs.setdefault(o.req_id, []).append(o.token_id)
    assert len(outs["a"]) == 30 and len(outs["b"]) == 2
    # a and b shared the prompt blocks; both finished -> all refs back
    assert all(r == 0 for r in e.kv._ref)


def test_eviction_reclaims_cached_blocks():
    e = make_engine(num_blocks=24, prefix=True, max_batch_size=2)
    for i in range(6):
        e.add_request(f"r{i}", list(range(i * 50, i * 50 + 40)),
                      SamplingParams(max_tokens=3))
    outs = drain(e)
    assert len(outs) == 6
    assert e.kv.free_blocks == 24          # everything reclaimed or cached-free


def test_full_block_prompt_keeps_one_suffix_token():
    """plen % 16 == 0 with a fully-cached prompt must still recompute the
    last block so first-token logits exist."""
    e = make_engine(prefix=True)
    prompt = list(range(32))               # exactly 2 blocks
    e.add_request("a", prompt, SamplingParams(max_tokens=4))
    t1 = drain(e)["a"]
    e.add_request("b", prompt, SamplingParams(max_tokens=4))
    t2 = drain(e)["b"]
    assert t1 == t2


def test_prefix_stats_in_worker_health():
    import asyncio

    from resilient_llm_amd.workers.engine_worker import EngineWorker
    from resilient_llm_amd.workers.base import GenerationRequest

    async def run():
        w = EngineWorker(device="cpu", model_name="tiny", device_label="pc",
                         num_blocks=64, seed=0)
        try:
            req = GenerationRequest(
                request_id="a", model="tiny",
                messages=[{"role": "user", "content": "same prompt " * 8}],
                max_tokens=2)
            await w.generate(req)
            await w.generate(GenerationRequest(
                request_id="b", model="tiny", messages=req.messages,
                max_tokens=2))
            h = await w.health()
            pc = h["prefix_cache"]
            assert pc["lookups"] >= 2 and pc["hits"] >= 1
        finally:
            await w.close()
    asyncio.run(run())
