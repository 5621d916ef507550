"""TP pool tests on CPU (gloo backend, world_size=2, tiny model).

The sharded-weights init draws the FULL weight per rank and slices, so a
TP=2 pool must produce EXACTLY the tokens a TP=1 engine produces (greedy,
same seed) — this validates column/row parallel layout, the per-layer
all-reduces, and the lockstep request-stream broadcast."""

import asyncio
import os
import tempfile

import pytest

from resilient_llm_amd.config import PoolDef
from resilient_llm_amd.workers.base import GenerationRequest
from resilient_llm_amd.workers.engine_worker import EngineWorker
from resilient_llm_amd.workers.pool import spawn_pool_worker
from resilient_llm_amd.workers.rpc import RpcWorkerClient

pytestmark = pytest.mark.timeout(240)


def greq(rid, text, n, **kw):
    return GenerationRequest(request_id=rid, model="tiny",
                             messages=[{"role": "user", "content": text}],
                             max_tokens=n, **kw)


@pytest.fixture()
def tp2_pool():
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    sock = os.path.join(tempfile.mkdtemp(prefix="rlli-tp-"), "pool.sock")
    pool = PoolDef(name="t", gpus=[0, 1], tensor_parallel=2)
    procs = spawn_pool_worker(pool, "tiny", sock, device_override="cpu",
                              tp_backend="gloo", max_batch=8)
    yield sock, procs
    for p in procs:
        if p.poll() is None:
            p.terminate()
    for p in procs:
        try:
            p.wait(timeout=10)
        except Exception:
            p.kill()


def tp1_reference_tokens(text, n, **kw):
    async def run():
        w = EngineWorker(device="cpu", model_name="tiny", device_label="ref",
                         num_blocks=64, seed=0)
        try:
            res = await w.generate(greq("ref", text, n, **kw))
            return res.text
        finally:
            await w.close()
    return asyncio.run(run())


def test_tp2_matches_tp1_and_serves_concurrently(tp2_pool):
    sock, procs = tp2_pool
    expected = tp1_reference_tokens("hello tensor parallel", 8)
    expected_pen = tp1_reference_tokens(
        "penalty parity", 6, presence_penalty=0.7, frequency_penalty=0.4)
    expected_sampled = tp1_reference_tokens(
        "sampled parity", 6, temperature=0.9, seed=424242)

    async def run():
        client = RpcWorkerClient("pool:t", {"tiny"}, sock)
        client.proc = procs[0]
        await client.connect(timeout=180)
        res = await client.generate(greq("a", "hello tensor parallel", 8))
        assert res.completion_tokens == 8
        assert res.text == expected, (res.text, expected)

        # concurrent batch through the lockstep pool
        results = await asyncio.gather(
            *[client.generate(greq(f"c{i}", f"prompt number {i}", 5))
              for i in range(4)])
        assert all(r.completion_tokens == 5 for r in results)

        # the lockstep 'add' broadcast must carry the FULL
        # sampling params — penalties and seeded sampling must match TP=1
        pen = await client.generate(greq("p", "penalty parity", 6,
                                         presence_penalty=0.7,
                                         frequency_penalty=0.4))
        assert pen.text == expected_pen
        sampled = await client.generate(greq("s", "sampled parity", 6,
                                             temperature=0.9, seed=424242))
        assert sampled.text == expected_sampled

        h = await client.health()
        assert h["status"] == "ok"
        await client.close()

    asyncio.run(run())


def test_idle_pool_survives_short_collective_timeout():
    """Idle-pool liveness: followers block in the control-group
    broadcast, so with a 3 s collective timeout an idle pool would die
    without the leader's heartbeat (RLLI_TP_HEARTBEAT_S) — run idle for
    3x the timeout, then serve."""
    env = {"RLLI_TP_TIMEOUT_S": "3", "RLLI_TP_HEARTBEAT_S": "0.5"}
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    old = {k: os.environ.get(k) for k in env}
    os.environ.update(env)
    sock = os.path.join(tempfile.mkdtemp(prefix="rlli-hb-"), "pool.sock")
    pool = PoolDef(name="hb", gpus=[0, 1], tensor_parallel=2)
    procs = spawn_pool_worker(pool, "tiny", sock, device_override="cpu",
                              tp_backend="gloo", max_batch=4)
    try:
        async def run():
            client = RpcWorkerClient("pool:hb", {"tiny"}, sock)
            client.proc = procs[0]
            await client.connect(timeout=180)
            res = await client.generate(greq("w", "warm", 2))
            assert res.completion_tokens == 2
            await asyncio.sleep(9.0)          # idle >> collective timeout
            assert all(p.poll() is None for p in procs), \
                "a pool rank died while idle"
            res = await client.generate(greq("a", "after idle", 3))
            assert res.completion_tokens == 3
            await client.close()
        asyncio.run(run())
    finally:
        for k, v in old.items():
            os.environ.pop(k, None) if v is None else os.environ.__setitem__(k, v)
        for p in procs:
            if p.poll() is None:
                p.kill()
