"""EngineWorker (tiny model, CPU) behind the full gateway, and the RPC
worker-process path (subprocess with --device cpu) — the end-to-end slice
without a GPU."""

import asyncio
import os
import subprocess
import sys
import tempfile
import time

import pytest

from resilient_llm_amd.workers.base import (
    GenerationRequest, WorkerDead, WorkerThrottled,
)
from resilient_llm_amd.workers.engine_worker import EngineWorker
from resilient_llm_amd.workers.rpc import RpcWorkerClient

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def greq(**kw):
    defaults = dict(request_id="t1", model="tiny",
                    messages=[{"role": "user", "content": "hello world"}],
                    max_tokens=6)
    defaults.update(kw)
    return GenerationRequest(**defaults)


def test_engine_worker_generate_and_stream():
    async def run():
        w = EngineWorker(device="cpu", model_name="tiny", device_label="gpu:t",
                         num_blocks=32)
        try:
            res = await w.generate(greq())
            assert res.completion_tokens == 6
            assert res.prompt_tokens > 0
            assert res.ttft_ms is not None and res.ttft_ms >= 0

            chunks = []
            async for c in w.generate_stream(greq(request_id="t2")):
                chunks.append(c)
            assert len(chunks) == 6
            assert chunks[-1].finish_reason == "length"
            h = await w.health()
            assert h["total_served"] == 2 and h["in_flight"] == 0
        finally:
            await w.close()
    asyncio.run(run())


def test_engine_worker_concurrent_batching():
    async def run():
        w = EngineWorker(device="cpu", model_name="tiny", device_label="gpu:t",
                         num_blocks=64, max_batch_size=8)
        try:
            results = await asyncio.gather(
                *[w.generate(greq(request_id=f"c{i}", max_tokens=5))
                  for i in range(6)])
            assert all(r.completion_tokens == 5 for r in results)
        finally:
            await w.close()
    asyncio.run(run())


def test_engine_worker_throttles_on_queue_full():
    async def run():
        w = EngineWorker(device="cpu", model_name="tiny", device_label="gpu:t",
                         num_blocks=8, max_queue=1, max_batch_size=1)
        try:
            with pytest.raises(WorkerThrottled):
                await w.generate(greq(max_tokens=10_000))
        finally:
            await w.close()
    asyncio.run(run())


@pytest.fixture()
def rpc_worker():
    sock = os.path.join(tempfile.mkdtemp(prefix="rlli-test-"), "w.sock")
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    proc = subprocess.Popen(
        [sys.executable, "-m", "resilient_llm_amd.workers.gpu_main",
         "--device-label", "gpu:9", "--model", "tiny", "--socket", sock,
         "--device", "cpu", "--num-blocks", "64"], env=env)
    yield sock, proc
    if proc.poll() is None:
        proc.terminate()
        proc.wait(timeout=10)


def test_rpc_worker_process_end_to_end(rpc_worker):
    sock, proc = rpc_worker

    async def run():
        client = RpcWorkerClient("gpu:9", {"tiny"}, sock)
        client.proc = proc
        await client.connect(timeout=60)
        res = await client.generate(greq())
        assert res.completion_tokens == 6

        chunks = []
        async for c in client.generate_stream(greq(request_id="s1", max_tokens=4)):
            chunks.append(c)
        assert len(chunks) == 4

        h = await client.health()
        assert h["status"] == "ok"

        # real process kill -> WorkerDead on subsequent calls
        await client.inject_fault("kill")
        t0 = time.time()
        while proc.poll() is None and time.time() - t0 < 10:
            await asyncio.sleep(0.1)
        assert proc.poll() == 7
        with pytest.raises(WorkerDead):
            await client.health()
        await client.close()

    asyncio.run(run())


def test_worker_target_step_ms_plumbed():
    """The decode-cadence SLO reaches the engine and the AIMD controller
    converges to the floor under an impossible target (CPU steps are
    milliseconds); generation output is unaffected."""
    async def run():
        w = EngineWorker(device="cpu", model_name="tiny", device_label="slo",
                         num_blocks=64, seed=0, target_step_ms=1e-6,
                         chunk_size=8)
        try:
            assert w.engine.target_step_ms == 1e-6
            res = await w.generate(GenerationRequest(
                request_id="s1", model="tiny",
                messages=[{"role": "user", "content": "hello " * 30}],
                max_tokens=4))
            assert res.completion_tokens == 4
            assert w.engine._prefill_budget == w.engine._budget_floor
        finally:
            await w.close()
    asyncio.run(run())


def test_stop_sequences_final_and_stream():
    """OpenAI `stop`: generation truncates at the earliest stop string
    (never emitted), finish_reason becomes "stop" — both modes."""
    async def run():
        w = EngineWorker(device="cpu", model_name="tiny", device_label="st",
                         num_blocks=64, seed=0)
        try:
            base = await w.generate(GenerationRequest(
                request_id="b", model="tiny",
                messages=[{"role": "user", "content": "stop test"}],
                max_tokens=8))
            assert len(base.text) > 2
            stop_s = base.text[2:4]     # guaranteed to occur
            res = await w.generate(GenerationRequest(
                request_id="s", model="tiny",
                messages=[{"role": "user", "content": "stop test"}],
                max_tokens=8, stop=[stop_s]))
            assert stop_s not in res.text
            assert res.finish_reason == "stop"
            assert base.text.startswith(res.text)

            chunks = []
            async for c in w.generate_stream(GenerationRequest(
                    request_id="t", model="tiny",
                    messages=[{"role": "user", "content": "stop test"}],
                    max_tokens=8, stop=[stop_s], stream=True)):
                chunks.append(c)
            text = "".join(c.text for c in chunks)
            assert text == res.text
            assert chunks[-1].finish_reason == "stop"
        finally:
            await w.close()
    asyncio.run(run())


def test_stream_decode_holds_back_partial_utf8():
    from resilient_llm_amd.utils.tokenizer import BYTE_OFFSET, ByteTokenizer

    tok = ByteTokenizer(512)
    ids = tok.encode("é", add_bos=False)        # 2 bytes
    assert len(ids) == 2
    assert tok.decode_stream(ids[:1], final=False) == ""     # held back
    assert tok.decode_stream(ids, final=False) == "é"
    assert tok.decode_stream(ids[:1], final=True) == "�"
    # 4-byte emoji split at every point
    e = tok.encode("\N{ROCKET}", add_bos=False)
    for cut in range(1, 4):
        assert tok.decode_stream(e[:cut], final=False) == ""
    assert tok.decode_stream(e, final=False) == "\N{ROCKET}"
    # ascii passes through untouched
    a = tok.encode("ok", add_bos=False)
    assert tok.decode_stream(a, final=False) == "ok"
