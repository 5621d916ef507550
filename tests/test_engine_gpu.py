"""GPU end-to-end engine tests (tiny models exercise all kernel shape
variants cheaply; one 8B test runs the flagship config)."""

import pytest
import torch

from resilient_llm_amd.engine import LLMEngine, PagedKVCache, SamplingParams
from resilient_llm_amd.models import LlamaForCausalLM, get_config

pytestmark = pytest.mark.gpu


def make_engine(model_name, num_blocks=128, seed=3, **kw):
    cfg = get_config(model_name)
    model = LlamaForCausalLM(cfg, device="cuda:0", dtype=torch.bfloat16,
                             seed=seed)
    kv = PagedKVCache.for_model(cfg, num_blocks, device="cuda:0")
    return LLMEngine(model, kv, **kw)


def drain(engine, max_steps=1000):
    outs = {}
    for _ in range(max_steps):
        if not engine.has_work():
            break
        for o in engine.step():
            outs.setdefault(o.req_id, []).append(o.token_id)
    assert not engine.has_work()
    return outs


@pytest.mark.parametrize("model_name", ["tiny", "tiny-128"])
def test_tiny_generation_deterministic(model_name):
    e1 = make_engine(model_name)
    e1.add_request("a", list(range(10, 50)), SamplingParams(max_tokens=8))
    t1 = drain(e1)["a"]
    e2 = make_engine(model_name)
    e2.add_request("a", list(range(10, 50)), SamplingParams(max_tokens=8))
    t2 = drain(e2)["a"]
    assert t1 == t2
    assert len(t1) == 8


def test_gpu_matches_cpu_reference_tiny():
    """The full GPU forward (HIP kernels) agrees with the CPU fp32
    reference path on greedy tokens for a short generation."""
    prompt = list(range(7, 40))
    gpu = make_engine("tiny", seed=11)
    gpu.add_request("a", prompt, SamplingParams(max_tokens=5))
    gpu_tokens = drain(gpu)["a"]

    cfg = get_config("tiny")
    model = LlamaForCausalLM(cfg, device="cpu", dtype=torch.float32, seed=11)
    # move the GPU model's exact bf16 weights to CPU fp32 so the compare
    # isolates kernels, not init RNG differences across devices
    for k, v in gpu.model.params.items():
        model.params[k] = v.detach().cpu().float()
    model.cos_sin = gpu.model.cos_sin.cpu()
    kv = PagedKVCache.for_model(cfg, 128, device="cpu")
    kv.k = kv.k.float()
    kv.v = kv.v.float()
    cpu = LLMEngine(model, kv)
    cpu.add_request("a", prompt, SamplingParams(max_tokens=5))
    cpu_tokens = drain(cpu)["a"]
    # bf16 vs fp32 forward: identical argmax for a few steps on random
    # weights is the expected outcome; allow 1 divergence step tail
    agree = 0
    for a, b in zip(gpu_tokens, cpu_tokens):
        if a != b:
            break
        agree += 1
    assert agree >= 1, (gpu_tokens, cpu_tokens)


def test_continuous_batching_gpu():
    e = make_engine("tiny", num_blocks=256, max_batch_size=16)
    for i in range(8):
        e.add_request(f"r{i}", list(range(5 + i, 40 + i)),
                      SamplingParams(max_tokens=10))
    outs = drain(e)
    assert len(outs) == 8
    assert all(len(v) == 10 for v in outs.values())


def test_llama8b_short_generation():
    e = make_engine("llama-3-8b", num_blocks=64, max_batch_size=2)
    e.add_request("a", list(range(5, 133)), SamplingParams(max_tokens=4))
    outs = drain(e)
    assert len(outs["a"]) == 4


def test_graph_runner_matches_eager():
    """hipGraph-captured decode produces the same greedy tokens as the
    eager path."""
    from resilient_llm_amd.engine.graph import install_graph_runner
    prompts = {f"g{i}": list(range(4 + i, 45 + 2 * i)) for i in range(5)}

    eager = make_engine("tiny-128", num_blocks=256, max_batch_size=8, seed=5)
    for rid, p in prompts.items():
        eager.add_request(rid, p, SamplingParams(max_tokens=12))
    eager_out = drain(eager)

    graphed = make_engine("tiny-128", num_blocks=256, max_batch_size=8, seed=5)
    install_graph_runner(graphed)
    for rid, p in prompts.items():
        graphed.add_request(rid, p, SamplingParams(max_tokens=12))
    graph_out = drain(graphed)
    assert eager_out == graph_out


def test_graph_runner_matches_eager_sampled():
    """Temperature sampling through the steady graph (seeds advanced
    in-graph by GOLDEN per replay) matches the eager path's step-offset
    premix token for token."""
    from resilient_llm_amd.engine.graph import install_graph_runner
    prompts = {f"s{i}": list(range(7 + i, 52 + 3 * i)) for i in range(4)}
    params = lambda: SamplingParams(max_tokens=10, temperature=0.8, seed=123)  # noqa: E731

    eager = make_engine("tiny-128", num_blocks=256, max_batch_size=8, seed=9)
    for rid, p in prompts.items():
        eager.add_request(rid, p, params())
    eager_out = drain(eager)

    graphed = make_engine("tiny-128", num_blocks=256, max_batch_size=8, seed=9)
    install_graph_runner(graphed)
    for rid, p in prompts.items():
        graphed.add_request(rid, p, params())
    graph_out = drain(graphed)
    assert eager_out == graph_out


def test_graph_runner_top_p_override_matches_eager():
    """A batch mixing top-p rows (torch-side sampling) with greedy rows
    still replays the steady graph; the override path must match the
    eager engine token for token."""
    from resilient_llm_amd.engine.graph import install_graph_runner

    def load(e):
        e.add_request("p0", list(range(5, 40)),
                      SamplingParams(max_tokens=8, temperature=0.9,
                                     top_p=0.8, seed=7))
        e.add_request("p1", list(range(9, 52)), SamplingParams(max_tokens=8))
        e.add_request("p2", list(range(3, 61)),
                      SamplingParams(max_tokens=8, temperature=0.7,
                                     presence_penalty=0.5, seed=11))

    eager = make_engine("tiny-128", num_blocks=256, max_batch_size=8, seed=3)
    load(eager)
    eager_out = drain(eager)

    graphed = make_engine("tiny-128", num_blocks=256, max_batch_size=8, seed=3)
    install_graph_runner(graphed)
    load(graphed)
    assert drain(graphed) == eager_out


def test_long_context_beyond_graph_envelope():
    """A sequence longer than the graph runner's block-table width falls
    back to the eager decode path and stays correct."""
    from resilient_llm_amd.engine.graph import install_graph_runner
    cfg = get_config("llama-3-8b")
    model = LlamaForCausalLM(cfg, device="cuda:0", dtype=torch.bfloat16, seed=1)
    kv = PagedKVCache.for_model(cfg, 256, device="cuda:0")
    e = LLMEngine(model, kv, max_batch_size=2, max_prefill_tokens=4096)
    install_graph_runner(e, max_blocks_per_seq=8)   # envelope: 128 tokens
    e.add_request("long", list(range(5, 1505)), SamplingParams(max_tokens=6))
    outs = drain(e)
    assert len(outs["long"]) == 6


def test_chunked_prefill_gpu_matches_unchunked():
    e1 = make_engine("tiny-128", num_blocks=256, max_batch_size=4, seed=5)
    e1.add_request("a", list(range(10, 150)), SamplingParams(max_tokens=6))
    want = drain(e1)["a"]
    cfg = get_config("tiny-128")
    model = LlamaForCausalLM(cfg, device="cuda:0", dtype=torch.bfloat16, seed=5)
    kv = PagedKVCache.for_model(cfg, 256, device="cuda:0")
    e2 = LLMEngine(model, kv, max_batch_size=4, chunk_size=48)
    e2.add_request("a", list(range(10, 150)), SamplingParams(max_tokens=6))
    got = drain(e2)["a"]
    assert got == want


def test_prefix_cache_gpu_identical_outputs():
    """GPU: repeated prompt reuses cached KV blocks with identical greedy
    output (the reused blocks were written by the HIP kernels)."""
    cfg = get_config("llama-3-8b")
    model = LlamaForCausalLM(cfg, device="cuda:0", dtype=torch.bfloat16, seed=2)
    kv = PagedKVCache.for_model(cfg, 128, device="cuda:0")
    e = LLMEngine(model, kv, max_batch_size=2, enable_prefix_caching=True)
    prompt = list(range(11, 140))
    e.add_request("a", prompt, SamplingParams(max_tokens=5))
    t1 = drain(e)["a"]
    hits0 = kv.prefix_hits
    e.add_request("b", prompt, SamplingParams(max_tokens=5))
    t2 = drain(e)["b"]
    assert t2 == t1
    assert kv.prefix_hits > hits0
