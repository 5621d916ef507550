"""Round-2 regression tests: seed premix consistency,
per-model EOS plumbing, abort-set hygiene.
"""

import asyncio

import pytest

from resilient_llm_amd.engine.engine import LLMEngine, SamplingParams, SeqState
from resilient_llm_amd.workers.base import GenerationRequest
from resilient_llm_amd.workers.engine_worker import EngineWorker

pytestmark = pytest.mark.timeout(120)

GOLDEN = 0x9e3779b97f4a7c15
M64 = (1 << 64) - 1


def u64(x: int) -> int:
    return x & M64


def greq(rid, text, n, **kw):
    return GenerationRequest(request_id=rid, model="tiny",
                             messages=[{"role": "user", "content": text}],
                             max_tokens=n, **kw)


def test_seq_seed_rebuild_invariant():
    """The host premix at output index n must equal the premix at index
    n0 plus the kernel's in-flight `+ GOLDEN*step` offset (mod 2^64) for
    any rebuild point n0 + step = n — i.e. sampled tokens cannot depend
    on when _rebuild_batch last ran."""
    s = SeqState(req_id="r", prompt_ids=[1, 2], params=SamplingParams(seed=12345))
    for n in (0, 1, 5, 63, 64, 200):
        s.output_ids = [0] * n
        premix_n = u64(LLMEngine._seq_seed(s))
        for step in (0, 1, 3, n):
            n0 = n - step
            if n0 < 0:
                continue
            s.output_ids = [0] * n0
            premix_n0 = u64(LLMEngine._seq_seed(s))
            assert u64(premix_n0 + GOLDEN * step) == premix_n, (n, step)
        s.output_ids = [0] * n
    # the value fits signed int64 (torch tensor dtype)
    s.params = SamplingParams(seed=(1 << 62) + 12345)
    s.output_ids = [0] * 100
    v = LLMEngine._seq_seed(s)
    assert -(1 << 63) <= v < (1 << 63)


def test_eos_override_stops_generation():
    """eos_id deployment override flows through to the engine and a
    sampled EOS finishes with reason 'stop' (non-2 EOS)."""
    async def run():
        w = EngineWorker(device="cpu", model_name="tiny", device_label="e",
                         num_blocks=64, eos_id=7)
        try:
            assert w.engine.eos_ids == frozenset({7})
            # _finish: a member id stops, the old default (2) does not
            seq = SeqState(req_id="x", prompt_ids=[1],
                           params=SamplingParams(max_tokens=10))
            seq.output_ids = [7]
            outs = []
            assert w.engine._finish(seq, outs, 7)
            assert outs[0].finish_reason == "stop"
            seq2 = SeqState(req_id="y", prompt_ids=[1],
                            params=SamplingParams(max_tokens=10))
            seq2.output_ids = [2]
            assert not w.engine._finish(seq2, [], 2)
        finally:
            await w.close()
    asyncio.run(run())


def test_model_preset_eos_default():
    from resilient_llm_amd.models import get_config
    assert set(get_config("llama-3-8b").eos_ids) == {128001, 128009}
    assert set(get_config("tiny").eos_ids) == {2}


def test_abort_set_does_not_grow_on_normal_completion():
    """Normal completions must not leave rids in engine._aborted or
    engine._live (one leaked entry per request forever)."""
    async def run():
        w = EngineWorker(device="cpu", model_name="tiny", device_label="a",
                         num_blocks=64)
        try:
            for i in range(5):
                res = await w.generate(greq(f"r{i}", f"ping {i}", 4))
                assert res.completion_tokens >= 1
            assert w.engine._aborted == set()
            assert w.engine._live == set()
            # a post-completion abort for a finished rid is a no-op
            w.engine.abort("r0-1")
            assert w.engine._aborted == set()
        finally:
            await w.close()
    asyncio.run(run())


def test_abort_still_works_for_live_requests():
    # bare engine (no worker thread) so the step sequence is deterministic
    from tests.test_engine import make_engine
    e = make_engine()
    e.add_request("live-1", [1, 2, 3], SamplingParams(max_tokens=50))
    assert "live-1" in e._live
    e.abort("live-1")
    assert "live-1" in e._aborted
    e.step()   # admits + drops the aborted request
    e.step()
    assert "live-1" not in e._live
    assert e._aborted == set()
    assert e.kv.free_blocks == e.kv.num_blocks
