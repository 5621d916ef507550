"""Prompt-lookup speculative decoding (opt-in, greedy): verification by
the chunked-prefill machinery must preserve greedy semantics EXACTLY —
any accept pattern, any workload — while cutting decode steps on
repetitive continuations."""

import pytest
import torch

from resilient_llm_amd.engine import LLMEngine, PagedKVCache, SamplingParams
from resilient_llm_amd.models import LlamaForCausalLM, get_config

_CFG = get_config("tiny")
_MODEL = LlamaForCausalLM(_CFG, device="cpu", dtype=torch.float32, seed=7)


def eng(spec=0, **kw):
    kv = PagedKVCache.for_model(_CFG, 128, device="cpu")
    kv.k = kv.k.float()
    kv.v = kv.v.float()
    return LLMEngine(_MODEL, kv, max_batch_size=4, spec_lookup=spec, **kw)


def drain(e, max_steps=800):
    outs = {}
    for _ in range(max_steps):
        if not e.has_work():
            break
        for o in e.step():
            outs.setdefault(o.req_id, []).append(o.token_id)
    assert not e.has_work()
    return outs


PROMPTS = {
    "repetitive": [7, 8, 9, 10] * 12,            # lookup hits constantly
    "random": [((i * 37) + 11) % 500 for i in range(48)],
    "short": [5, 6, 7],
}


@pytest.mark.parametrize("name", list(PROMPTS))
def test_spec_greedy_token_exact(name):
    prompt = PROMPTS[name]
    base = eng(spec=0)
    base.add_request("a", prompt, SamplingParams(max_tokens=24,
                                                 stop_on_eos=False))
    want = drain(base)["a"]

    spec = eng(spec=4)
    spec.add_request("a", prompt, SamplingParams(max_tokens=24,
                                                 stop_on_eos=False))
    got = drain(spec)["a"]
    assert got == want, f"spec changed greedy output on {name}"
    assert spec.kv.free_blocks == spec.kv.num_blocks


def test_spec_reduces_steps_on_repetitive_output():
    """The tiny random-init model collapses into repeating loops —
    exactly what lookup speculation accelerates."""
    prompt = PROMPTS["repetitive"]
    base = eng(spec=0)
    base.add_request("a", prompt, SamplingParams(max_tokens=32,
                                                 stop_on_eos=False))
    drain(base)
    spec = eng(spec=4)
    spec.add_request("a", prompt, SamplingParams(max_tokens=32,
                                                 stop_on_eos=False))
    drain(spec)
    base_steps = base.stats["decode_steps"]
    spec_steps = spec.stats["decode_steps"] + spec.stats.get("spec_steps", 0)
    assert spec.stats.get("spec_accepted", 0) > 0
    assert spec_steps < base_steps, (base_steps, spec_steps)


def test_spec_mixed_batch_and_sampled_fallback():
    """A sampled request in the batch disables speculation (greedy-only
    eligibility) without changing anyone's tokens."""
    base = eng(spec=0)
    spec = eng(spec=4)
    for e in (base, spec):
        e.add_request("g", PROMPTS["repetitive"],
                      SamplingParams(max_tokens=16, stop_on_eos=False))
        e.add_request("s", PROMPTS["random"],
                      SamplingParams(max_tokens=16, temperature=0.8,
                                     seed=3, stop_on_eos=False))
    want = drain(base)
    got = drain(spec)
    assert got == want


def test_spec_respects_max_tokens_and_eos():
    e = eng(spec=4)
    e.add_request("a", PROMPTS["repetitive"],
                  SamplingParams(max_tokens=5, stop_on_eos=False))
    outs = drain(e)["a"]
    assert len(outs) == 5
    assert e.kv.free_blocks == e.kv.num_blocks


def test_out_of_vocab_prompt_rejected():
    """OOV ids must raise at the API boundary, not fault the GPU inside
    the embedding gather (found by a GPU HSA exception)."""
    e = eng()
    with pytest.raises(ValueError):
        e.add_request("bad", [5, 6, _CFG.vocab_size + 3],
                      SamplingParams(max_tokens=4))
    with pytest.raises(ValueError):
        e.add_request("neg", [5, -1], SamplingParams(max_tokens=4))


@pytest.mark.gpu
def test_gpu_spec_token_exact():
    """GPU spec path (chunked-prefill verification on HIP kernels) must
    equal the plain greedy run token for token."""
    cfg = get_config("tiny-128")
    model = LlamaForCausalLM(cfg, device="cuda:0", dtype=torch.bfloat16,
                             seed=7)

    def gpu_eng(spec):
        kv = PagedKVCache.for_model(cfg, 128, device="cuda:0")
        return LLMEngine(model, kv, max_batch_size=4, spec_lookup=spec)

    prompt = ([7, 8, 9, 10] * 12) + [3, 4]
    base = gpu_eng(0)
    base.add_request("a", prompt, SamplingParams(max_tokens=32,
                                                 stop_on_eos=False))
    want = drain(base)["a"]
    spec = gpu_eng(4)
    spec.add_request("a", prompt, SamplingParams(max_tokens=32,
                                                 stop_on_eos=False))
    got = drain(spec)["a"]
    assert got == want
