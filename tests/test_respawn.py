"""Elastic recovery: a dead worker process is respawned by the gateway
health loop and rejoins rotation."""

import asyncio
import os
import subprocess
import sys
import tempfile
import threading
import time

import pytest

from resilient_llm_amd.client import OpenAIClient, APIError
from resilient_llm_amd.config import load_config
from resilient_llm_amd.gateway.app import GatewayApp
from resilient_llm_amd.gateway.http import HttpServer
from resilient_llm_amd.workers.base import WorkerRegistry
from resilient_llm_amd.workers.rpc import RpcWorkerClient
from tests.gateway_harness import free_port

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

pytestmark = pytest.mark.timeout(180)


def spawn_cpu_worker(sock):
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    return subprocess.Popen(
        [sys.executable, "-m", "resilient_llm_amd.workers.gpu_main",
         "--device-label", "gpu:0", "--model", "tiny", "--socket", sock,
         "--device", "cpu", "--num-blocks", "64"], env=env)


def test_worker_respawn_after_kill():
    sock = os.path.join(tempfile.mkdtemp(prefix="rlli-respawn-"), "w.sock")
    port = free_port()
    cfg = load_config(data={
        "cluster": {"port": port},
        "model_list": [
            {"model_name": "tiny-serve",
             "litellm_params": {"model": "gpu/0/tiny"},
             "model_info": {"id": "gpu0/tiny"}}],
        "router_settings": {"routing_strategy": "simple-shuffle"},
    })
    loop = asyncio.new_event_loop()
    ready = threading.Event()
    holder = {}

    async def main():
        holder["stop"] = asyncio.Event()
        registry = WorkerRegistry()
        client = RpcWorkerClient("gpu:0", {"tiny"}, sock)
        client.proc = spawn_cpu_worker(sock)
        client.respawn = lambda: spawn_cpu_worker(sock)
        await client.connect(timeout=120)
        registry.register("gpu", "0", client)
        holder["client"] = client
        app = GatewayApp(cfg, registry, health_interval_s=0.3)
        app.respawn_cooldown_s = 0.0
        server = HttpServer(app.handle, port=port)
        await server.start()
        await app.start_background()
        ready.set()
        await holder["stop"].wait()
        await app.stop_background()
        await server.stop()
        await registry.close()

    th = threading.Thread(target=lambda: loop.run_until_complete(main()),
                          daemon=True)
    th.start()
    assert ready.wait(120)
    try:
        http = OpenAIClient(f"http://127.0.0.1:{port}")
        msgs = [{"role": "user", "content": "hello"}]
        r = http.chat.completions.create(model="tiny-serve", messages=msgs,
                                         max_tokens=3)
        assert r.usage.completion_tokens == 3

        # real process kill
        http.inject_fault("gpu:0", "kill")
        first_proc = holder["client"].proc
        t0 = time.time()
        while first_proc.poll() is None and time.time() - t0 < 15:
            time.sleep(0.1)
        assert first_proc.poll() is not None

        # the health loop respawns; serving resumes
        deadline = time.time() + 90
        ok = False
        while time.time() < deadline:
            try:
                r = http.chat.completions.create(model="tiny-serve",
                                                 messages=msgs, max_tokens=3)
                ok = True
                break
            except APIError:
                time.sleep(1.0)
        assert ok, "worker never came back after respawn"
        assert holder["client"].proc is not first_proc
    finally:
        loop.call_soon_threadsafe(holder["stop"].set)
        th.join(timeout=15)


def test_pool_group_respawn():
    """A dead TP pool restarts as a WHOLE group: surviving followers are
    reaped, a fresh rank group is spawned (new RCCL/gloo rendezvous),
    and the pool serves again (ROADMAP item: pool elastic recovery)."""
    import os
    import tempfile

    from resilient_llm_amd.config import PoolDef
    from resilient_llm_amd.workers.base import GenerationRequest
    from resilient_llm_amd.workers.pool import spawn_pool_worker
    from resilient_llm_amd.workers.rpc import RpcWorkerClient

    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    sock = os.path.join(tempfile.mkdtemp(prefix="rlli-pr-"), "pool.sock")
    pool = PoolDef(name="pr", gpus=[0, 1], tensor_parallel=2)

    def spawn():
        return spawn_pool_worker(pool, "tiny", sock, device_override="cpu",
                                 tp_backend="gloo", max_batch=4)

    procs = spawn()

    def greq(rid):
        return GenerationRequest(request_id=rid, model="tiny",
                                 messages=[{"role": "user", "content": "hi"}],
                                 max_tokens=4)

    async def run():
        client = RpcWorkerClient("pool:pr", {"tiny"}, sock)
        client.proc = procs[0]
        client.proc_group = procs
        client.respawn = spawn
        await client.connect(timeout=180)
        res = await client.generate(greq("a"))
        assert res.completion_tokens == 4

        # kill the LEADER hard; follower becomes an orphan
        procs[0].kill()
        procs[0].wait(timeout=10)

        assert await client.respawn_now()
        # group was replaced wholesale
        assert client.proc_group is not procs
        assert all(p.poll() is not None for p in procs)
        res = await client.generate(greq("b"))
        assert res.completion_tokens == 4
        await client.close()

    try:
        asyncio.run(run())
    finally:
        for p in procs:
            if p.poll() is None:
                p.kill()


def test_health_loop_detects_dead_pool_follower():
    """A dead pool FOLLOWER strands the RCCL collective while the leader
    still answers health RPCs.  The health loop must notice (proc_group
    poll), take the leader down, and whole-group respawn must rebuild —
    end to end on a real tiny TP=2 gloo pool."""
    import tempfile as _tf

    from resilient_llm_amd.config import PoolDef, load_config
    from resilient_llm_amd.workers.base import GenerationRequest
    from resilient_llm_amd.workers.pool import spawn_pool_worker

    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    sock = os.path.join(_tf.mkdtemp(prefix="rlli-fd-"), "pool.sock")
    pool = PoolDef(name="fd", gpus=[0, 1], tensor_parallel=2)

    def spawn():
        return spawn_pool_worker(pool, "tiny", sock, device_override="cpu",
                                 tp_backend="gloo", max_batch=4)

    procs = spawn()
    client = RpcWorkerClient("pool:fd", {"tiny"}, sock)
    client.proc = procs[0]
    client.proc_group = procs
    client.respawn = spawn
    client.last_respawn = 0.0

    config = load_config(data={
        "cluster": {"port": free_port()},
        "model_list": [{"model_name": "fd-model",
                        "litellm_params": {"model": "stub/0/llama-3-8b"}}],
    })
    registry = WorkerRegistry()
    registry.register("pool", "fd", client)

    async def run():
        await client.connect(timeout=180)
        app = GatewayApp(config, registry, health_interval_s=0.3)
        app.respawn_cooldown_s = 0.0
        await app.start_background()
        try:
            # kill the FOLLOWER only; leader keeps serving health RPCs
            procs[1].kill()
            procs[1].wait(timeout=10)
            deadline = asyncio.get_event_loop().time() + 120
            while asyncio.get_event_loop().time() < deadline:
                await asyncio.sleep(1.0)
                if (client.proc_group is not procs
                        and client.proc.poll() is None):
                    break
            assert client.proc_group is not procs, \
                "health loop never rebuilt the pool after follower death"
            res = await client.generate(GenerationRequest(
                request_id="fd1", model="tiny",
                messages=[{"role": "user", "content": "after respawn"}],
                max_tokens=3))
            assert res.completion_tokens == 3
        finally:
            await app.stop_background()
            await client.close()

    try:
        asyncio.run(run())
    finally:
        for p in procs:
            if p.poll() is None:
                p.kill()


def test_rpc_server_survives_malformed_frames():
    """Garbage on the worker socket (bad lengths, non-msgpack, wrong
    types) drops that connection but never the worker: a well-formed
    client keeps working."""
    import struct
    import tempfile as _tf

    import msgpack

    from resilient_llm_amd.workers.engine_worker import EngineWorker
    from resilient_llm_amd.workers.rpc import WorkerRpcServer

    sock = os.path.join(_tf.mkdtemp(prefix="rlli-fz-"), "w.sock")

    async def run():
        w = EngineWorker(device="cpu", model_name="tiny", device_label="fz",
                         num_blocks=64, seed=0)
        srv = WorkerRpcServer(w, sock)
        await srv.start()
        try:
            for payload in (
                    b"\xff\xff\xff\xff",                     # 4 GiB length
                    b"\x05\x00\x00\x00hello",                # not msgpack
                    struct.pack("<I", 1) + msgpack.packb(7), # non-dict msg
                    b"\x02\x00",                             # truncated
            ):
                r, wtr = await asyncio.open_unix_connection(sock)
                wtr.write(payload)
                await wtr.drain()
                wtr.close()
            await asyncio.sleep(0.2)

            client = RpcWorkerClient("fz", {"tiny"}, sock)
            await client.connect(timeout=30)
            h = await client.health()
            assert h["status"] == "ok"
            await client.close()
        finally:
            await srv.stop()
            await w.close()

    asyncio.run(run())


def test_wedged_worker_terminated_and_respawned():
    """A worker whose PROCESS is alive but which stops answering health
    probes (hung loop / wedged engine) is terminated by the health loop
    after a streak of failed probes and respawned — r02 chaos soaks left
    such workers unhealthy forever because respawn only fired for dead
    processes."""
    sock = os.path.join(tempfile.mkdtemp(prefix="rlli-wedge-"), "w.sock")
    port = free_port()
    cfg = load_config(data={
        "cluster": {"port": port},
        "model_list": [
            {"model_name": "tiny-serve",
             "litellm_params": {"model": "gpu/0/tiny"},
             "model_info": {"id": "gpu0/tiny"}}],
        "router_settings": {"routing_strategy": "simple-shuffle"},
    })
    loop = asyncio.new_event_loop()
    ready = threading.Event()
    holder = {}

    async def main():
        holder["stop"] = asyncio.Event()
        registry = WorkerRegistry()
        client = RpcWorkerClient("gpu:0", {"tiny"}, sock)
        client.proc = spawn_cpu_worker(sock)
        client.respawn = lambda: spawn_cpu_worker(sock)
        await client.connect(timeout=120)
        registry.register("gpu", "0", client)
        holder["client"] = client
        app = GatewayApp(cfg, registry, health_interval_s=0.2)
        app.respawn_cooldown_s = 0.0
        server = HttpServer(app.handle, port=port)
        await server.start()
        await app.start_background()
        ready.set()
        await holder["stop"].wait()
        await app.stop_background()
        await server.stop()
        await registry.close()

    th = threading.Thread(target=lambda: loop.run_until_complete(main()),
                          daemon=True)
    th.start()
    assert ready.wait(120)
    try:
        http = OpenAIClient(f"http://127.0.0.1:{port}")
        msgs = [{"role": "user", "content": "hello"}]
        r = http.chat.completions.create(model="tiny-serve", messages=msgs,
                                         max_tokens=3)
        assert r.usage.completion_tokens == 3

        # 'hang' fault: process stays ALIVE but health probes fail
        first_proc = holder["client"].proc
        http.inject_fault("gpu:0", "hang")
        assert first_proc.poll() is None

        # streak >= 8 -> terminate -> dead-proc respawn path -> serving
        deadline = time.time() + 90
        ok = False
        while time.time() < deadline:
            try:
                r = http.chat.completions.create(model="tiny-serve",
                                                 messages=msgs, max_tokens=3,
                                                 timeout=10)
                if holder["client"].proc is not first_proc:
                    ok = True
                    break
            except APIError:
                pass
            time.sleep(1.0)
        assert ok, "wedged worker was never terminated+respawned"
        assert first_proc.poll() is not None, "old process still alive"
    finally:
        loop.call_soon_threadsafe(holder["stop"].set)
        th.join(timeout=15)


def test_full_outage_both_workers_respawn():
    """BOTH worker processes die at once (total outage): the health
    loop respawns both, and serving resumes — the cluster self-heals
    from zero capacity without operator action."""
    run_dir = tempfile.mkdtemp(prefix="rlli-outage-")
    socks = [os.path.join(run_dir, f"w{i}.sock") for i in range(2)]
    port = free_port()
    cfg = load_config(data={
        "cluster": {"port": port},
        "model_list": [
            {"model_name": "tiny-serve",
             "litellm_params": {"model": "gpu/0/tiny"},
             "model_info": {"id": "gpu0/tiny"}},
            {"model_name": "tiny-serve",
             "litellm_params": {"model": "gpu/1/tiny"},
             "model_info": {"id": "gpu1/tiny"}}],
        "router_settings": {"routing_strategy": "simple-shuffle"},
    })
    loop = asyncio.new_event_loop()
    ready = threading.Event()
    holder: dict = {}

    def spawn_one(sock, label):
        env = dict(os.environ)
        env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
        return subprocess.Popen(
            [sys.executable, "-m", "resilient_llm_amd.workers.gpu_main",
             "--device-label", label, "--model", "tiny", "--socket", sock,
             "--device", "cpu", "--num-blocks", "64"], env=env)

    async def main():
        holder["stop"] = asyncio.Event()
        registry = WorkerRegistry()
        for i in range(2):
            c = RpcWorkerClient(f"gpu:{i}", {"tiny"}, socks[i])
            c.proc = spawn_one(socks[i], f"gpu:{i}")
            c.respawn = (lambda s=socks[i], l=f"gpu:{i}": spawn_one(s, l))
            await c.connect(timeout=120)
            registry.register("gpu", str(i), c)
            holder[f"w{i}"] = c
        app = GatewayApp(cfg, registry, health_interval_s=0.3)
        app.respawn_cooldown_s = 0.0
        server = HttpServer(app.handle, port=port)
        await server.start()
        await app.start_background()
        ready.set()
        await holder["stop"].wait()
        await app.stop_background()
        await server.stop()
        await registry.close()

    th = threading.Thread(target=lambda: loop.run_until_complete(main()),
                          daemon=True)
    th.start()
    assert ready.wait(120)
    try:
        http = OpenAIClient(f"http://127.0.0.1:{port}")
        msgs = [{"role": "user", "content": "outage"}]
        r = http.chat.completions.create(model="tiny-serve", messages=msgs,
                                         max_tokens=3)
        assert r.usage.completion_tokens == 3
        old = [holder["w0"].proc, holder["w1"].proc]
        for p in old:
            p.kill()
        for p in old:
            p.wait(timeout=10)
        # total outage -> self-heal -> serving resumes
        deadline = time.time() + 90
        ok = False
        while time.time() < deadline:
            try:
                r = http.chat.completions.create(
                    model="tiny-serve", messages=msgs, max_tokens=3,
                    timeout=10)
                ok = True
                break
            except APIError:
                time.sleep(0.5)
        assert ok, "cluster never recovered from total outage"
        assert holder["w0"].proc is not old[0]
        assert holder["w1"].proc is not old[1]
    finally:
        loop.call_soon_threadsafe(holder["stop"].set)
        th.join(timeout=15)
        for i in range(2):
            w = holder.get(f"w{i}")
            if w is not None and w.proc is not None and w.proc.poll() is None:
                w.proc.kill()
