"""TP pool worker path on a real GPU (world=1 pool validates RCCL group
init + pool serving; multi-GPU TP is exercised by the CPU gloo test and
the 8-GPU driver runs)."""

import asyncio
import os
import tempfile

import pytest

from resilient_llm_amd.config import PoolDef
from resilient_llm_amd.workers.base import GenerationRequest
from resilient_llm_amd.workers.pool import spawn_pool_worker
from resilient_llm_amd.workers.rpc import RpcWorkerClient

pytestmark = [pytest.mark.gpu, pytest.mark.timeout(300)]


def test_pool_worker_gpu_world1():
    sock = os.path.join(tempfile.mkdtemp(prefix="rlli-poolgpu-"), "p.sock")
    pool = PoolDef(name="g", gpus=[0], tensor_parallel=1)
    procs = spawn_pool_worker(pool, "tiny-128", sock, max_batch=4)
    try:
        async def run():
            client = RpcWorkerClient("pool:g", {"tiny-128"}, sock)
            client.proc = procs[0]
            await client.connect(timeout=240)
            res = await client.generate(GenerationRequest(
                request_id="p1", model="tiny-128",
                messages=[{"role": "user", "content": "pool on gpu"}],
                max_tokens=5))
            assert res.completion_tokens == 5
            h = await client.health()
            assert h["status"] == "ok"
            await client.close()
        asyncio.run(run())
    finally:
        for p in procs:
            if p.poll() is None:
                p.terminate()
        for p in procs:
            try:
                p.wait(timeout=10)
            except Exception:
                p.kill()
