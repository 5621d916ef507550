"""TP-sharded safetensors loading.

Saves a tiny checkpoint from an unsharded model, then verifies that
(a) each TP rank's sharded load takes exactly the slices _init_weights
would have taken, and (b) a TP=2 gloo pool serving from that checkpoint
produces token-exact output vs a TP=1 worker loading the same files.
"""

import asyncio
import os
import tempfile

import pytest
import torch

from resilient_llm_amd.config import PoolDef
from resilient_llm_amd.models import LlamaForCausalLM, get_config
from resilient_llm_amd.workers.base import GenerationRequest
from resilient_llm_amd.workers.engine_worker import EngineWorker
from resilient_llm_amd.workers.pool import spawn_pool_worker
from resilient_llm_amd.workers.rpc import RpcWorkerClient

pytestmark = pytest.mark.timeout(240)


@pytest.fixture(scope="module")
def tiny_ckpt():
    d = tempfile.mkdtemp(prefix="rlli-ckpt-")
    full = LlamaForCausalLM(get_config("tiny"), device="cpu",
                            dtype=torch.float32, seed=99)
    full.save_safetensors(d)
    return d, full


def test_sharded_load_slices_match(tiny_ckpt):
    d, full = tiny_ckpt
    c = get_config("tiny")
    for r in range(2):
        m = LlamaForCausalLM(c, device="cpu", dtype=torch.float32,
                             tp_rank=r, tp_world=2, seed=1)
        n = m.load_safetensors(d)
        assert n > 0
        # column-parallel qkv: rank slice of q, k, v stacked
        q_full, k_full, v_full = torch.split(
            full.params["l0.qkv"], [c.q_size, c.kv_size, c.kv_size], dim=0)
        qs, ks, vs = torch.split(
            m.params["l0.qkv"], [m.q_size, m.kv_size, m.kv_size], dim=0)
        assert torch.equal(qs, q_full[r * m.q_size:(r + 1) * m.q_size])
        assert torch.equal(ks, k_full[r * m.kv_size:(r + 1) * m.kv_size])
        assert torch.equal(vs, v_full[r * m.kv_size:(r + 1) * m.kv_size])
        # row-parallel o/down: rank slice of input columns
        assert torch.equal(m.params["l0.o"],
                           full.params["l0.o"][:, r * m.q_size:(r + 1) * m.q_size])
        assert torch.equal(m.params["l1.down"],
                           full.params["l1.down"][:, r * m.inter:(r + 1) * m.inter])
        # replicated tensors load whole
        assert torch.equal(m.params["embed"], full.params["embed"])
        assert torch.equal(m.params["final_ln"], full.params["final_ln"])


def greq(rid, text, n, **kw):
    return GenerationRequest(request_id=rid, model="tiny",
                             messages=[{"role": "user", "content": text}],
                             max_tokens=n, **kw)


def test_tp2_pool_with_checkpoint_matches_tp1(tiny_ckpt):
    d, _ = tiny_ckpt
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")

    async def tp1():
        w = EngineWorker(device="cpu", model_name="tiny", device_label="ref",
                         num_blocks=64, weights=d, seed=0)
        try:
            return (await w.generate(greq("ref", "checkpoint parity", 8))).text
        finally:
            await w.close()
    expected = asyncio.run(tp1())

    sock = os.path.join(tempfile.mkdtemp(prefix="rlli-tpw-"), "pool.sock")
    pool = PoolDef(name="w", gpus=[0, 1], tensor_parallel=2)
    procs = spawn_pool_worker(pool, "tiny", sock, device_override="cpu",
                              tp_backend="gloo", max_batch=8, weights=d)
    try:
        async def run():
            client = RpcWorkerClient("pool:w", {"tiny"}, sock)
            client.proc = procs[0]
            await client.connect(timeout=180)
            res = await client.generate(greq("a", "checkpoint parity", 8))
            assert res.text == expected, (res.text, expected)
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
