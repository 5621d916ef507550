"""Live request migration: extract a running sequence (tokens + sampling
identity + KV blocks) from one engine and adopt it into another, with
TOKEN-EXACT continuation — the state-migration primitive SURVEY.md §5.8
earmarked for failover without recompute."""

import pytest
import torch

from resilient_llm_amd.engine import LLMEngine, PagedKVCache, SamplingParams
from resilient_llm_amd.models import LlamaForCausalLM, get_config

_CFG = get_config("tiny")
_MODEL = LlamaForCausalLM(_CFG, device="cpu", dtype=torch.float32, seed=7)


def fresh_engine(**kw):
    kv = PagedKVCache.for_model(_CFG, 96, device="cpu")
    kv.k = kv.k.float()
    kv.v = kv.v.float()
    return LLMEngine(_MODEL, kv, max_batch_size=4, **kw)


def run_steps(e, n):
    outs = []
    for _ in range(n):
        outs.extend(e.step())
    return outs


@pytest.mark.parametrize("temperature", [0.0, 0.8])
def test_extract_adopt_token_exact(temperature):
    params = lambda: SamplingParams(max_tokens=14, temperature=temperature,  # noqa: E731
                                    seed=99, stop_on_eos=False)
    prompt = list(range(9, 51))

    ref = fresh_engine()
    ref.add_request("m", prompt, params())
    want = []
    while ref.has_work():
        for o in ref.step():
            want.append(o.token_id)

    src = fresh_engine()
    dst = fresh_engine()
    src.add_request("m", prompt, params())
    got = [o.token_id for o in run_steps(src, 6)]     # prefill + ~5 decodes
    assert 0 < len(got) < 14
    src.request_extract("m")
    src.step()
    state = src.take_extracted("m")
    assert state is not None and state["kv"] is not None
    assert state["output_ids"] == got
    assert not src.has_work()
    # all source KV blocks recovered
    assert src.kv.free_blocks == src.kv.num_blocks

    dst.queue_adopt(state)
    for _ in range(100):
        if not dst.has_work():
            break
        for o in dst.step():
            got.append(o.token_id)
    assert dst.take_adopt_result("m") == "ok"
    assert got == want, f"migrated continuation diverged: {got} != {want}"
    assert dst.kv.free_blocks == dst.kv.num_blocks


def test_extract_waiting_request_re_prefills():
    src = fresh_engine()
    dst = fresh_engine()
    src.add_request("w", list(range(5, 40)), SamplingParams(max_tokens=5,
                                                            stop_on_eos=False))
    src.request_extract("w")          # extracted before any step
    src.step()
    state = src.take_extracted("w")
    assert state is not None and state["kv"] is None
    dst.queue_adopt(state)
    toks = []
    for _ in range(60):
        if not dst.has_work() and toks:
            break
        for o in dst.step():
            toks.append(o.token_id)
    assert len(toks) == 5


def test_extract_unknown_rid_reports_missing():
    e = fresh_engine()
    e.request_extract("ghost")
    e.step()
    # explicit sentinel: a finished/unknown rid must be distinguishable
    # from not-yet-extracted (a None would make the worker poll its
    # full timeout for every request that finished before the sweep)
    assert e.take_extracted("ghost") == "missing"


# ---------------------------------------------------- worker-level
def test_worker_migrate_out_in_token_exact():
    """Two in-process EngineWorkers: a blocked generate() on A raises
    WorkerMigrated when its request is migrated; re-issuing the SAME
    request_id against B attaches to the adopted sequence and returns
    the full, token-exact completion."""
    import asyncio
    from resilient_llm_amd.workers.base import (GenerationRequest,
                                                WorkerMigrated)
    from resilient_llm_amd.workers.engine_worker import EngineWorker

    async def main():
        a = EngineWorker(device="cpu", model_name="tiny", num_blocks=64,
                         max_batch_size=4)
        b = EngineWorker(device="cpu", model_name="tiny", num_blocks=64,
                         max_batch_size=4)
        req = GenerationRequest(
            request_id="mig-1", model="tiny",
            messages=[{"role": "user", "content": "hello " * 30}],
            max_tokens=120, temperature=0.0)
        # reference: uninterrupted run on an identical worker
        c = EngineWorker(device="cpu", model_name="tiny", num_blocks=64,
                         max_batch_size=4)
        want = (await c.generate(dataclasses_replace(req))).text

        task = asyncio.create_task(a.generate(dataclasses_replace(req)))
        await asyncio.sleep(0.05)          # let a few tokens generate
        blob = await a.migrate_out("mig-1")
        await b.migrate_in(blob)
        await a.release_migrated("mig-1")  # adopt-then-release protocol
        with pytest.raises(WorkerMigrated):
            await task
        await asyncio.sleep(0.15)          # generation continues unattached
        res = await b.generate(dataclasses_replace(req))
        assert res.text == want, f"{res.text!r} != {want!r}"
        for w in (a, b, c):
            await w.close()

    def dataclasses_replace(req):
        import dataclasses as dc
        return dc.replace(req)

    asyncio.new_event_loop().run_until_complete(main())


# ---------------------------------------------------- gateway-level
def test_gateway_admin_migrate_end_to_end():
    """Full stack: a long request is live-migrated between two worker
    PROCESSES via POST /admin/migrate while the client blocks; the
    client transparently gets the complete, token-exact answer from the
    target worker (device header proves the re-route)."""
    import json as _json
    import os
    import subprocess
    import sys
    import tempfile
    import threading
    import time
    import urllib.error
    import urllib.request

    import asyncio
    from resilient_llm_amd.client import OpenAIClient
    from resilient_llm_amd.config import load_config
    from resilient_llm_amd.gateway.app import GatewayApp
    from resilient_llm_amd.gateway.http import HttpServer
    from resilient_llm_amd.workers.base import WorkerRegistry
    from resilient_llm_amd.workers.rpc import RpcWorkerClient
    from tests.gateway_harness import free_port

    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

    def spawn(sock, label):
        env = dict(os.environ)
        env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
        return subprocess.Popen(
            [sys.executable, "-m", "resilient_llm_amd.workers.gpu_main",
             "--device-label", label, "--model", "tiny", "--socket", sock,
             "--device", "cpu", "--num-blocks", "96"], env=env)

    run_dir = tempfile.mkdtemp(prefix="rlli-migrate-")
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
        "router_settings": {"routing_strategy": "round-robin"},
    })
    loop = asyncio.new_event_loop()
    ready = threading.Event()
    holder: dict = {}

    async def main():
        holder["stop"] = asyncio.Event()
        registry = WorkerRegistry()
        for i in range(2):
            c = RpcWorkerClient(f"gpu:{i}", {"tiny"}, socks[i])
            c.proc = spawn(socks[i], f"gpu:{i}")
            await c.connect(timeout=120)
            registry.register("gpu", str(i), c)
            holder[f"w{i}"] = c
        app = GatewayApp(cfg, registry, health_interval_s=0.5)
        server = HttpServer(app.handle, host="127.0.0.1", port=port)
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
    assert ready.wait(timeout=150)
    base = f"http://127.0.0.1:{port}"
    client = OpenAIClient(base)
    msgs = [{"role": "user", "content": "migrate me " * 20}]

    # reference: same request, untouched (greedy -> deterministic)
    ref = client.chat.completions.create(model="tiny-serve", messages=msgs,
                                         max_tokens=300, timeout=120)

    result = {}

    def do_request():
        result["res"] = client.chat.completions.create(
            model="tiny-serve", messages=msgs, max_tokens=300, timeout=120,
            extra_headers={"x-request-id": "mig-e2e"})

    t = threading.Thread(target=do_request)
    t.start()
    time.sleep(0.3)                        # a slice of tokens generated

    def post(path, body):
        r = urllib.request.Request(base + path, method="POST",
                                   data=_json.dumps(body).encode(),
                                   headers={"content-type": "application/json"})
        with urllib.request.urlopen(r, timeout=60) as resp:
            return resp.status, _json.loads(resp.read().decode())

    # the round-robin router put ref on one worker and mig-e2e on the
    # other — find the holder by trying both directions
    migrated = None
    for src, dst in (("gpu:0", "gpu:1"), ("gpu:1", "gpu:0")):
        try:
            st, body = post("/admin/migrate", {"request_id": "mig-e2e",
                                               "from": src, "to": dst})
            migrated = (src, dst, body)
            break
        except urllib.error.HTTPError:
            continue
    try:
        assert migrated is not None, "migration found no live request"
        t.join(timeout=120)
        res = result["res"]
        assert (res.choices[0].message.content
                == ref.choices[0].message.content), "migrated answer diverged"
        # the answer came from the TARGET worker
        assert res.headers.get("x-gateway-device") == migrated[1]
        assert migrated[2]["state_bytes"] > 0
    finally:
        loop.call_soon_threadsafe(holder["stop"].set)
        th.join(timeout=30)
        for i in range(2):
            w = holder.get(f"w{i}")
            if w is not None and w.proc is not None and w.proc.poll() is None:
                w.proc.kill()


def test_drain_with_migration_evacuates_live_requests():
    """POST /admin/drain {migrate_to}: every live request on the drained
    worker moves to the target and completes token-exact — zero-recompute
    evacuation for maintenance."""
    import json as _json
    import os
    import subprocess
    import sys
    import tempfile
    import threading
    import time
    import urllib.request

    import asyncio
    from resilient_llm_amd.client import OpenAIClient
    from resilient_llm_amd.config import load_config
    from resilient_llm_amd.gateway.app import GatewayApp
    from resilient_llm_amd.gateway.http import HttpServer
    from resilient_llm_amd.workers.base import WorkerRegistry
    from resilient_llm_amd.workers.rpc import RpcWorkerClient
    from tests.gateway_harness import free_port

    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

    def spawn(sock, label):
        env = dict(os.environ)
        env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
        return subprocess.Popen(
            [sys.executable, "-m", "resilient_llm_amd.workers.gpu_main",
             "--device-label", label, "--model", "tiny", "--socket", sock,
             "--device", "cpu", "--num-blocks", "96"], env=env)

    run_dir = tempfile.mkdtemp(prefix="rlli-evac-")
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
        "router_settings": {"routing_strategy": "round-robin"},
    })
    loop = asyncio.new_event_loop()
    ready = threading.Event()
    holder: dict = {}

    async def main():
        holder["stop"] = asyncio.Event()
        registry = WorkerRegistry()
        for i in range(2):
            c = RpcWorkerClient(f"gpu:{i}", {"tiny"}, socks[i])
            c.proc = spawn(socks[i], f"gpu:{i}")
            await c.connect(timeout=120)
            registry.register("gpu", str(i), c)
            holder[f"w{i}"] = c
        app = GatewayApp(cfg, registry, health_interval_s=0.5)
        server = HttpServer(app.handle, host="127.0.0.1", port=port)
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
    assert ready.wait(timeout=150)
    base = f"http://127.0.0.1:{port}"
    client = OpenAIClient(base)
    msgs = [{"role": "user", "content": "evacuate " * 25}]
    ref = client.chat.completions.create(model="tiny-serve", messages=msgs,
                                         max_tokens=250, timeout=120)
    results = {}

    def do(i):
        results[i] = client.chat.completions.create(
            model="tiny-serve", messages=msgs, max_tokens=250, timeout=120,
            extra_headers={"x-request-id": f"evac-{i}"})

    threads = [threading.Thread(target=do, args=(i,)) for i in range(3)]
    for t in threads:
        t.start()
    time.sleep(0.3)
    try:
        req = urllib.request.Request(
            base + "/admin/drain", method="POST",
            data=_json.dumps({"worker": "gpu:0",
                              "migrate_to": "gpu:1"}).encode(),
            headers={"content-type": "application/json"})
        with urllib.request.urlopen(req, timeout=120) as r:
            body = _json.loads(r.read().decode())
        assert body["draining"] is True
        assert not body["migrate_errors"], body["migrate_errors"]
        for t in threads:
            t.join(timeout=120)
        want = ref.choices[0].message.content
        for i in range(3):
            got = results[i].choices[0].message.content
            assert got == want, f"evac-{i} diverged"
        # everything that was on gpu:0 moved
        assert len(body["migrated"]) >= 1
    finally:
        loop.call_soon_threadsafe(holder["stop"].set)
        th.join(timeout=30)
        for i in range(2):
            w = holder.get(f"w{i}")
            if w is not None and w.proc is not None and w.proc.poll() is None:
                w.proc.kill()


@pytest.mark.gpu
def test_gpu_engine_migration_token_exact():
    """On-device KV extraction (D2H) + adoption (H2D) between two GPU
    engines, bf16, graphs on the target: continuation is token-exact."""
    from resilient_llm_amd.engine.graph import install_graph_runner
    cfg = get_config("tiny-128")
    m1 = LlamaForCausalLM(cfg, device="cuda:0", dtype=torch.bfloat16, seed=7)
    m2 = LlamaForCausalLM(cfg, device="cuda:0", dtype=torch.bfloat16, seed=7)

    def eng(m, graphs=False):
        kv = PagedKVCache.for_model(cfg, 128, device="cuda:0")
        e = LLMEngine(m, kv, max_batch_size=4)
        if graphs:
            install_graph_runner(e)
        return e

    params = lambda: SamplingParams(max_tokens=16, temperature=0.8,  # noqa: E731
                                    seed=5, stop_on_eos=False)
    prompt = list(range(11, 64))

    ref = eng(m1)
    ref.add_request("g", prompt, params())
    want = []
    while ref.has_work():
        for o in ref.step():
            want.append(o.token_id)

    src = eng(m1)
    dst = eng(m2, graphs=True)
    src.add_request("g", prompt, params())
    got = []
    for _ in range(7):
        for o in src.step():
            got.append(o.token_id)
    assert 0 < len(got) < 16
    src.request_extract("g")
    # the 1-deep async pipeline lands the pending token DURING the
    # extraction step — collect it (the worker path buffers it as a
    # 'pre' token the same way)
    for o in src.step():
        got.append(o.token_id)
    state = src.take_extracted("g")
    assert state is not None and state["kv"] is not None
    assert state["output_ids"] == got
    dst.queue_adopt(state)
    for _ in range(200):
        if not dst.has_work():
            break
        for o in dst.step():
            got.append(o.token_id)
    assert got == want, f"GPU migrated continuation diverged"


def test_gateway_midstream_migration_completes_stream():
    """Mid-SSE-stream migration: drain-with-evacuation moves the
    streaming request; the client's stream continues and ends with full
    usage (chunk-per-token pre replay keeps the gateway skip in sync)."""
    import json as _json
    import os
    import subprocess
    import sys
    import tempfile
    import threading
    import time
    import urllib.request

    import asyncio
    from resilient_llm_amd.client import OpenAIClient
    from resilient_llm_amd.config import load_config
    from resilient_llm_amd.gateway.app import GatewayApp
    from resilient_llm_amd.gateway.http import HttpServer
    from resilient_llm_amd.workers.base import WorkerRegistry
    from resilient_llm_amd.workers.rpc import RpcWorkerClient
    from tests.gateway_harness import free_port

    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

    def spawn(sock, label):
        env = dict(os.environ)
        env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
        return subprocess.Popen(
            [sys.executable, "-m", "resilient_llm_amd.workers.gpu_main",
             "--device-label", label, "--model", "tiny", "--socket", sock,
             "--device", "cpu", "--num-blocks", "96"], env=env)

    run_dir = tempfile.mkdtemp(prefix="rlli-smig-")
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
        "router_settings": {"routing_strategy": "round-robin"},
    })
    loop = asyncio.new_event_loop()
    ready = threading.Event()
    holder: dict = {}

    async def main():
        holder["stop"] = asyncio.Event()
        registry = WorkerRegistry()
        for i in range(2):
            c = RpcWorkerClient(f"gpu:{i}", {"tiny"}, socks[i])
            c.proc = spawn(socks[i], f"gpu:{i}")
            await c.connect(timeout=120)
            registry.register("gpu", str(i), c)
            holder[f"w{i}"] = c
        app = GatewayApp(cfg, registry, health_interval_s=0.5)
        server = HttpServer(app.handle, host="127.0.0.1", port=port)
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
    assert ready.wait(timeout=150)
    base = f"http://127.0.0.1:{port}"
    client = OpenAIClient(base)
    msgs = [{"role": "user", "content": "stream and move " * 20}]
    events: list = []
    usage: dict = {}
    done = threading.Event()
    first_dev: dict = {}

    def consume():
        stream = client.chat.completions.create(
            model="tiny-serve", messages=msgs, max_tokens=300, stream=True,
            timeout=120, stream_options={"include_usage": True},
            extra_headers={"x-request-id": "smig-1"})
        first_dev["d"] = stream.headers.get("x-gateway-device") \
            if hasattr(stream, "headers") else None
        for evt in stream:
            if "error" in evt:
                raise RuntimeError(evt["error"])
            if evt.get("usage"):
                usage.update(evt["usage"])
            events.append(evt)
        done.set()

    t = threading.Thread(target=consume)
    t.start()
    try:
        for _ in range(600):
            if len(events) >= 10 or done.is_set():
                break
            time.sleep(0.01)
        assert not done.is_set(), "stream finished before migration window"
        migrated = None
        for src, dst in (("gpu:0", "gpu:1"), ("gpu:1", "gpu:0")):
            req = urllib.request.Request(
                base + "/admin/drain", method="POST",
                data=_json.dumps({"worker": src,
                                  "migrate_to": dst}).encode(),
                headers={"content-type": "application/json"})
            with urllib.request.urlopen(req, timeout=60) as r:
                body = _json.loads(r.read().decode())
            if body["migrated"]:
                migrated = body
                break
            # undrain the wrong guess
            urllib.request.urlopen(urllib.request.Request(
                base + "/admin/drain", method="POST",
                data=_json.dumps({"worker": src, "drain": False}).encode(),
                headers={"content-type": "application/json"}), timeout=30)
        assert migrated and migrated["migrated"] == ["smig-1"], migrated
        assert not migrated["migrate_errors"]
        assert done.wait(timeout=120)
        t.join(timeout=30)
        assert usage.get("completion_tokens") == 300, usage
    finally:
        done.set()
        loop.call_soon_threadsafe(holder["stop"].set)
        th.join(timeout=30)
        for i in range(2):
            w = holder.get(f"w{i}")
            if w is not None and w.proc is not None and w.proc.poll() is None:
                w.proc.kill()


def test_migration_any_cut_point_token_exact_property():
    """Property: extracting after ANY number of steps and adopting into
    a fresh engine yields exactly the uninterrupted token sequence —
    greedy and sampled, across prompt lengths (hypothesis-lite sweep:
    deterministic grid keeps it fast and reproducible)."""
    for temp in (0.0, 0.8):
        for plen in (5, 23, 47):
            prompt = [(i * 13 + plen) % 500 for i in range(plen)]
            params = lambda: SamplingParams(max_tokens=12,  # noqa: E731
                                            temperature=temp, seed=41,
                                            stop_on_eos=False)
            ref = fresh_engine()
            ref.add_request("p", prompt, params())
            want = []
            while ref.has_work():
                for o in ref.step():
                    want.append(o.token_id)
            for cut in (0, 1, 3, 7, 11):
                src = fresh_engine()
                dst = fresh_engine()
                src.add_request("p", prompt, params())
                got = []
                for _ in range(cut):
                    for o in src.step():
                        got.append(o.token_id)
                src.request_extract("p")
                for o in src.step():
                    got.append(o.token_id)
                state = src.take_extracted("p")
                if state == "missing":       # finished before the cut
                    assert got == want
                    continue
                dst.queue_adopt(state)
                for _ in range(200):
                    if not dst.has_work():
                        break
                    for o in dst.step():
                        got.append(o.token_id)
                assert got == want, (temp, plen, cut, got, want)
                assert dst.kv.free_blocks == dst.kv.num_blocks
                assert src.kv.free_blocks == src.kv.num_blocks


def test_drain_skips_evacuation_to_dead_target():
    """Drain with migrate_to pointing at a DEAD worker: the evacuation
    is skipped after a fast health probe (no deadline burned on doomed
    RPCs), the drain itself still succeeds, and in-flight requests
    finish in place (r02 chaos-soak fix)."""
    import json as _json
    import os
    import subprocess
    import sys
    import tempfile
    import threading
    import time
    import urllib.request

    import asyncio
    from resilient_llm_amd.client import OpenAIClient
    from resilient_llm_amd.config import load_config
    from resilient_llm_amd.gateway.app import GatewayApp
    from resilient_llm_amd.gateway.http import HttpServer
    from resilient_llm_amd.workers.base import WorkerRegistry
    from resilient_llm_amd.workers.rpc import RpcWorkerClient
    from tests.gateway_harness import free_port

    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

    def spawn(sock, label):
        env = dict(os.environ)
        env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
        return subprocess.Popen(
            [sys.executable, "-m", "resilient_llm_amd.workers.gpu_main",
             "--device-label", label, "--model", "tiny", "--socket", sock,
             "--device", "cpu", "--num-blocks", "96"], env=env)

    run_dir = tempfile.mkdtemp(prefix="rlli-evacdead-")
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
        "router_settings": {"routing_strategy": "round-robin"},
    })
    loop = asyncio.new_event_loop()
    ready = threading.Event()
    holder: dict = {}

    async def main():
        holder["stop"] = asyncio.Event()
        registry = WorkerRegistry()
        for i in range(2):
            c = RpcWorkerClient(f"gpu:{i}", {"tiny"}, socks[i])
            c.proc = spawn(socks[i], f"gpu:{i}")
            await c.connect(timeout=120)
            registry.register("gpu", str(i), c)
            holder[f"w{i}"] = c
        app = GatewayApp(cfg, registry, health_interval_s=5.0)
        server = HttpServer(app.handle, host="127.0.0.1", port=port)
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
    assert ready.wait(timeout=150)
    base = f"http://127.0.0.1:{port}"
    client = OpenAIClient(base)
    msgs = [{"role": "user", "content": "finish in place " * 10}]
    results: dict = {}
    errors: list = []

    def do(i):
        try:
            results[i] = client.chat.completions.create(
                model="tiny-serve", messages=msgs, max_tokens=150,
                timeout=120, extra_headers={"x-request-id": f"ed-{i}"})
        except Exception as e:                        # noqa: BLE001
            errors.append(repr(e))

    threads = [threading.Thread(target=do, args=(i,)) for i in range(2)]
    for t in threads:
        t.start()
    time.sleep(0.3)
    try:
        # kill the TARGET, then immediately drain the source toward it
        holder["w1"].proc.kill()
        holder["w1"].proc.wait(timeout=10)
        t0 = time.time()
        req = urllib.request.Request(
            base + "/admin/drain", method="POST",
            data=_json.dumps({"worker": "gpu:0", "migrate_to": "gpu:1",
                              "timeout_s": 30}).encode(),
            headers={"content-type": "application/json"})
        with urllib.request.urlopen(req, timeout=60) as r:
            body = _json.loads(r.read().decode())
        took = time.time() - t0
        assert body["draining"] is True
        assert body["migrated"] == []
        assert any("evacuation skipped" in e for e in body["migrate_errors"]), \
            body["migrate_errors"]
        # the probe failed fast: nowhere near the 30 s deadline
        assert took < 20, f"drain took {took:.1f}s"
        # in-flight requests on the drained source still finish (the
        # requests routed to the dead gpu:1 fail over to gpu:0)
        for t in threads:
            t.join(timeout=120)
        assert len(results) == 2, f"requests lost: {errors}"
        # undrain for symmetry
        urllib.request.urlopen(urllib.request.Request(
            base + "/admin/drain", method="POST",
            data=_json.dumps({"worker": "gpu:0",
                              "drain": False}).encode(),
            headers={"content-type": "application/json"}), timeout=30)
    finally:
        loop.call_soon_threadsafe(holder["stop"].set)
        th.join(timeout=30)
        for i in range(2):
            w = holder.get(f"w{i}")
            if w is not None and w.proc is not None and w.proc.poll() is None:
                w.proc.kill()


def test_drain_survives_source_death_mid_evacuation():
    """The SOURCE worker dies while its drain-evacuation sweep is in
    flight: the admin call still returns (bounded), and every client
    request completes — either migrated, or failed over with stateless
    replay after the death."""
    import json as _json
    import os
    import subprocess
    import sys
    import tempfile
    import threading
    import time
    import urllib.request

    import asyncio
    from resilient_llm_amd.client import OpenAIClient
    from resilient_llm_amd.config import load_config
    from resilient_llm_amd.gateway.app import GatewayApp
    from resilient_llm_amd.gateway.http import HttpServer
    from resilient_llm_amd.workers.base import WorkerRegistry
    from resilient_llm_amd.workers.rpc import RpcWorkerClient
    from tests.gateway_harness import free_port

    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

    def spawn(sock, label):
        env = dict(os.environ)
        env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
        return subprocess.Popen(
            [sys.executable, "-m", "resilient_llm_amd.workers.gpu_main",
             "--device-label", label, "--model", "tiny", "--socket", sock,
             "--device", "cpu", "--num-blocks", "96"], env=env)

    run_dir = tempfile.mkdtemp(prefix="rlli-evacdie-")
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
        "router_settings": {"routing_strategy": "round-robin"},
    })
    loop = asyncio.new_event_loop()
    ready = threading.Event()
    holder: dict = {}

    async def main():
        holder["stop"] = asyncio.Event()
        registry = WorkerRegistry()
        for i in range(2):
            c = RpcWorkerClient(f"gpu:{i}", {"tiny"}, socks[i])
            c.proc = spawn(socks[i], f"gpu:{i}")
            await c.connect(timeout=120)
            registry.register("gpu", str(i), c)
            holder[f"w{i}"] = c
        app = GatewayApp(cfg, registry, health_interval_s=0.3)
        server = HttpServer(app.handle, host="127.0.0.1", port=port)
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
    assert ready.wait(timeout=150)
    base = f"http://127.0.0.1:{port}"
    client = OpenAIClient(base)
    msgs = [{"role": "user", "content": "hold the line " * 15}]
    results: dict = {}
    errors: list = []

    def do(i):
        try:
            results[i] = client.chat.completions.create(
                model="tiny-serve", messages=msgs, max_tokens=300,
                timeout=120, extra_headers={"x-request-id": f"dv-{i}"})
        except Exception as e:                        # noqa: BLE001
            errors.append(repr(e))

    threads = [threading.Thread(target=do, args=(i,)) for i in range(6)]
    for t in threads:
        t.start()
    time.sleep(0.4)
    drain_result: dict = {}

    def drain():
        req = urllib.request.Request(
            base + "/admin/drain", method="POST",
            data=_json.dumps({"worker": "gpu:0", "migrate_to": "gpu:1",
                              "timeout_s": 20}).encode(),
            headers={"content-type": "application/json"})
        try:
            with urllib.request.urlopen(req, timeout=60) as r:
                drain_result.update(_json.loads(r.read().decode()))
        except Exception as e:                        # noqa: BLE001
            drain_result["exc"] = repr(e)

    try:
        dt = threading.Thread(target=drain)
        dt.start()
        time.sleep(0.1)                # let the sweep get going
        holder["w0"].proc.kill()       # source dies MID-evacuation
        dt.join(timeout=90)
        assert not dt.is_alive(), "drain call never returned"
        assert "exc" not in drain_result, drain_result
        for t in threads:
            t.join(timeout=120)
        assert len(results) == 6, f"lost requests: {errors}"
        for i, r in results.items():
            assert r.usage.completion_tokens == 300, (i, r.usage)
        # undrain (source will be respawned by the health loop)
        urllib.request.urlopen(urllib.request.Request(
            base + "/admin/drain", method="POST",
            data=_json.dumps({"worker": "gpu:0",
                              "drain": False}).encode(),
            headers={"content-type": "application/json"}), timeout=30)
    finally:
        loop.call_soon_threadsafe(holder["stop"].set)
        th.join(timeout=30)
        for i in range(2):
            w = holder.get(f"w{i}")
            if w is not None and w.proc is not None and w.proc.poll() is None:
                w.proc.kill()
