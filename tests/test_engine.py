"""Engine + model tests on CPU (tiny model, fp32 reference ops)."""

import dataclasses

import pytest
import torch

from resilient_llm_amd.engine import LLMEngine, PagedKVCache, SamplingParams
from resilient_llm_amd.engine.engine import CapacityExceeded
from resilient_llm_amd.models import LlamaForCausalLM, get_config


def make_engine(num_blocks=64, model_seed=7, **kw):
    cfg = get_config("tiny")
    model = LlamaForCausalLM(cfg, device="cpu", dtype=torch.float32,
                             seed=model_seed)
    kv = PagedKVCache.for_model(cfg, num_blocks, device="cpu")
    kv.k = kv.k.float()
    kv.v = kv.v.float()
    return LLMEngine(model, kv, **kw)


def drain(engine, max_steps=500):
    outs = {}
    for _ in range(max_steps):
        if not engine.has_work():
            break
        for o in engine.step():
            outs.setdefault(o.req_id, []).append(o)
    assert not engine.has_work(), "engine did not drain"
    return outs


def tokens_of(outs, rid):
    return [o.token_id for o in outs[rid]]


def test_generate_shapes_and_determinism():
    prompt = list(range(5, 25))
    e1 = make_engine()
    e1.add_request("a", prompt, SamplingParams(max_tokens=8))
    outs1 = drain(e1)
    assert len(outs1["a"]) == 8
    assert outs1["a"][-1].finished and outs1["a"][-1].finish_reason == "length"

    e2 = make_engine()
    e2.add_request("a", prompt, SamplingParams(max_tokens=8))
    outs2 = drain(e2)
    assert tokens_of(outs1, "a") == tokens_of(outs2, "a")


def test_continuous_batching_matches_solo_greedy():
    """A sequence decoded while another joins mid-flight produces the
    same greedy tokens as decoded alone (paged attention correctness
    under batching)."""
    p_a = list(range(10, 40))
    p_b = list(range(50, 67))

    solo = make_engine()
    solo.add_request("a", p_a, SamplingParams(max_tokens=10))
    toks_solo = tokens_of(drain(solo), "a")

    mixed = make_engine()
    mixed.add_request("a", p_a, SamplingParams(max_tokens=10))
    early = mixed.step() + mixed.step()      # prefill a + decode 1
    mixed.add_request("b", p_b, SamplingParams(max_tokens=6))
    outs = drain(mixed)
    for o in reversed(early):
        outs.setdefault(o.req_id, []).insert(0, o)
    assert tokens_of(outs, "a") == toks_solo
    assert len(outs["b"]) == 6


def test_batched_prefill_multiple_requests():
    e = make_engine()
    for i in range(4):
        e.add_request(f"r{i}", list(range(3 + i, 20 + i)),
                      SamplingParams(max_tokens=4))
    outs = drain(e)
    assert set(outs) == {"r0", "r1", "r2", "r3"}
    assert all(len(v) == 4 for v in outs.values())


def test_capacity_exceeded_on_impossible_request():
    e = make_engine(num_blocks=4)   # 64 tokens of KV
    with pytest.raises(CapacityExceeded):
        e.add_request("big", list(range(50)), SamplingParams(max_tokens=100))


def test_waiting_queue_drains_as_blocks_free():
    # 2 requests that cannot fit together but fit sequentially
    e = make_engine(num_blocks=6, max_batch_size=8)
    e.add_request("a", list(range(30)), SamplingParams(max_tokens=30))  # 4 blocks
    e.add_request("b", list(range(30)), SamplingParams(max_tokens=30))  # 4 blocks
    outs = drain(e)
    assert len(outs["a"]) == 30 and len(outs["b"]) == 30


def test_blocks_freed_after_completion():
    e = make_engine(num_blocks=16)
    before = e.kv.free_blocks
    e.add_request("a", list(range(20)), SamplingParams(max_tokens=4))
    drain(e)
    assert e.kv.free_blocks == before


def test_abort_releases_resources():
    e = make_engine(num_blocks=16)
    before = e.kv.free_blocks
    e.add_request("a", list(range(20)), SamplingParams(max_tokens=50))
    e.step()   # prefill
    e.abort("a")
    e.step()   # abort processed
    assert e.kv.free_blocks == before
    assert not e.has_work()


def test_temperature_sampling_varies_with_seed():
    e1 = make_engine(model_seed=7)
    e1.seed = 1
    e1.add_request("a", list(range(20)), SamplingParams(max_tokens=12,
                                                        temperature=1.0))
    t1 = tokens_of(drain(e1), "a")
    e2 = make_engine(model_seed=7)
    e2.seed = 2
    e2.add_request("a", list(range(20)), SamplingParams(max_tokens=12,
                                                        temperature=1.0))
    t2 = tokens_of(drain(e2), "a")
    assert t1 != t2


def test_model_forward_no_nans():
    cfg = get_config("tiny")
    m = LlamaForCausalLM(cfg, device="cpu", dtype=torch.float32)
    kv = PagedKVCache.for_model(cfg, 8, device="cpu")
    kv.k = kv.k.float()
    kv.v = kv.v.float()
    T = 12
    blocks = kv.allocate(1)
    logits = m.forward_prefill(
        torch.arange(T, dtype=torch.int32),
        torch.arange(T, dtype=torch.int32),
        kv,
        torch.tensor([blocks[0] * 16 + i for i in range(T)], dtype=torch.int32),
        torch.tensor([0, T], dtype=torch.int32))
    assert logits.shape == (1, cfg.vocab_size)
    assert torch.isfinite(logits.float()).all()


def test_per_request_seed_reproducible():
    """The same prompt with the same seed samples the same tokens, even
    from different batch positions / engine instances."""
    def run(engine_seed, reqs):
        e = make_engine(model_seed=7)
        e.seed = engine_seed
        for rid, prompt, seed in reqs:
            e.add_request(rid, prompt,
                          SamplingParams(max_tokens=8, temperature=0.9,
                                         seed=seed))
        return {k: [o.token_id for o in v] for k, v in drain(e).items()}

    a = run(1, [("x", list(range(20)), 1234)])
    # different engine seed, different batch company: same request seed
    b = run(99, [("pad", list(range(40, 60)), None),
                 ("x", list(range(20)), 1234)])
    assert a["x"] == b["x"]


def test_top_p_sampling_path():
    e = make_engine(model_seed=7)
    e.add_request("p", list(range(30)),
                  SamplingParams(max_tokens=6, temperature=0.8, top_p=0.7,
                                 seed=5))
    outs = drain(e)
    assert len(outs["p"]) == 6
    # reproducible
    e2 = make_engine(model_seed=7)
    e2.add_request("p", list(range(30)),
                   SamplingParams(max_tokens=6, temperature=0.8, top_p=0.7,
                                  seed=5))
    assert [o.token_id for o in drain(e2)["p"]] == \
        [o.token_id for o in outs["p"]]


def test_additional_model_presets_generate():
    """Every non-flagship preset steps through the engine on CPU (the
    same LlamaForCausalLM covers the whole family — dims only)."""
    from resilient_llm_amd.models import get_config

    for name in ("llama-3-1b", "mistral-7b"):
        cfg = get_config(name)
        assert cfg.n_heads * cfg.head_dim in (cfg.hidden_size, cfg.hidden_size * 2)
        # shrink to CPU-testable size but keep the preset's SHAPE rules
        small = dataclasses.replace(cfg, n_layers=2, vocab_size=256,
                                    hidden_size=cfg.hidden_size // 16,
                                    intermediate_size=cfg.intermediate_size // 16,
                                    n_heads=max(2, cfg.n_heads // 16),
                                    n_kv_heads=max(1, cfg.n_kv_heads // 8),
                                    head_dim=64, max_position=512)
        model = LlamaForCausalLM(small, device="cpu", dtype=torch.float32,
                                 seed=1)
        kv = PagedKVCache.for_model(small, 64, device="cpu")
        kv.k = kv.k.float(); kv.v = kv.v.float()
        eng = LLMEngine(model, kv, max_batch_size=2)
        eng.add_request("a", list(range(5, 25)), SamplingParams(max_tokens=3))
        toks = []
        while eng.has_work():
            toks += [o.token_id for o in eng.step()]
        assert len(toks) == 3


def _drain_tokens(eng, rid="a"):
    toks = []
    while eng.has_work():
        toks += [o.token_id for o in eng.step() if o.req_id == rid]
    return toks


def test_frequency_penalty_reduces_repetition():
    """A strong frequency penalty strictly lowers repeat counts vs the
    unpenalized greedy run on the same prompt (and never crashes the
    mixed torch/greedy path)."""
    def run(pen):
        cfg = get_config("tiny")
        model = LlamaForCausalLM(cfg, device="cpu", dtype=torch.float32,
                                 seed=3)
        kv = PagedKVCache.for_model(cfg, 64, device="cpu")
        kv.k = kv.k.float(); kv.v = kv.v.float()
        eng = LLMEngine(model, kv, max_batch_size=2)
        eng.add_request("a", [7, 8, 9, 10], SamplingParams(
            max_tokens=24, frequency_penalty=pen))
        return _drain_tokens(eng)

    base = run(0.0)
    pen = run(50.0)
    def max_repeat(ts):
        from collections import Counter
        return max(Counter(ts).values())
    assert len(pen) == len(base) == 24
    assert max_repeat(pen) < max_repeat(base) or max_repeat(base) == 1
    # a huge penalty forces all-distinct tokens
    assert max_repeat(pen) <= 2


def test_presence_penalty_deterministic_and_plumbed():
    cfg = get_config("tiny")
    model = LlamaForCausalLM(cfg, device="cpu", dtype=torch.float32, seed=3)
    kv = PagedKVCache.for_model(cfg, 64, device="cpu")
    kv.k = kv.k.float(); kv.v = kv.v.float()
    outs = []
    for _ in range(2):
        eng = LLMEngine(model, kv, max_batch_size=2)
        eng.add_request("a", [3, 4, 5], SamplingParams(
            max_tokens=6, temperature=0.7, top_p=0.9,
            presence_penalty=0.5, seed=11))
        outs.append(_drain_tokens(eng))
    assert outs[0] == outs[1]           # fully seeded, reproducible
