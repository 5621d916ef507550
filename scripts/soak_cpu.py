#!/usr/bin/env python3
"""CPU-scale replica of scripts/soak.py's chaos scenario (tiny model,
2 worker processes, kills + live-migration sweeps) so regressions in the
kill/respawn/migrate interplay reproduce WITHOUT a GPU box.

Usage: python scripts/soak_cpu.py [--duration 60] [--threads 8]
"""
import argparse
import asyncio
import json
import os
import random
import subprocess
import sys
import tempfile
import threading
import time
import urllib.request

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

from resilient_llm_amd.client import OpenAIClient          # noqa: E402
from resilient_llm_amd.config import load_config           # noqa: E402
from resilient_llm_amd.gateway.app import GatewayApp       # noqa: E402
from resilient_llm_amd.gateway.http import HttpServer      # noqa: E402
from resilient_llm_amd.workers.base import WorkerRegistry  # noqa: E402
from resilient_llm_amd.workers.rpc import RpcWorkerClient  # noqa: E402


def _spawn(sock, label):
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    return subprocess.Popen(
        [sys.executable, "-m", "resilient_llm_amd.workers.gpu_main",
         "--device-label", label, "--model", "tiny", "--socket", sock,
         "--device", "cpu", "--num-blocks", "128"], env=env)


def _free_port():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def main(duration_s=60, n_threads=8):
    run_dir = tempfile.mkdtemp(prefix="rlli-soakcpu-")
    socks = [os.path.join(run_dir, f"w{i}.sock") for i in range(2)]
    port = _free_port()
    cfg = load_config(data={
        "cluster": {"port": port},
        "model_list": [
            {"model_name": "soak", "litellm_params": {"model": "gpu/0/tiny"},
             "model_info": {"id": "gpu0/soak"}},
            {"model_name": "soak", "litellm_params": {"model": "gpu/1/tiny"},
             "model_info": {"id": "gpu1/soak"}}],
        "router_settings": {"routing_strategy": "simple-shuffle",
                            "enable_pre_call_checks": False},
    })
    loop = asyncio.new_event_loop()
    ready = threading.Event()
    holder: dict = {}

    async def amain():
        holder["stop"] = asyncio.Event()
        registry = WorkerRegistry()
        for i in range(2):
            c = RpcWorkerClient(f"gpu:{i}", {"tiny"}, socks[i])
            c.proc = _spawn(socks[i], f"gpu:{i}")
            c.respawn = (lambda s=socks[i], l=f"gpu:{i}": _spawn(s, l))
            await c.connect(timeout=120)
            registry.register("gpu", str(i), c)
            holder[f"w{i}"] = c
        app = GatewayApp(cfg, registry, health_interval_s=0.5)
        app.respawn_cooldown_s = 1.0
        server = HttpServer(app.handle, host="127.0.0.1", port=port)
        await server.start()
        await app.start_background()
        ready.set()
        await holder["stop"].wait()
        await app.stop_background()
        await server.stop()
        await registry.close()

    th = threading.Thread(target=lambda: loop.run_until_complete(amain()),
                          daemon=True)
    th.start()
    assert ready.wait(timeout=150)
    base = f"http://127.0.0.1:{port}"
    stats = {"ok": 0, "err": 0, "stream_ok": 0, "stream_err": 0}
    lock = threading.Lock()
    stop = threading.Event()
    err_samples: list = []
    err_kinds: dict = {}

    def worker(tid):
        rng = random.Random(tid)
        c = OpenAIClient(base, api_key=f"sk-{tid}")
        while not stop.is_set():
            msgs = [{"role": "user",
                     "content": "x" * rng.randint(16, 400)}]
            stream = rng.random() < 0.3
            try:
                if stream:
                    s = c.chat.completions.create(
                        model="soak", messages=msgs,
                        max_tokens=rng.randint(4, 32), stream=True,
                        temperature=rng.choice([0.0, 0.8]), timeout=60)
                    s.collect_text()
                    with lock:
                        stats["stream_ok"] += 1
                else:
                    c.chat.completions.create(
                        model="soak", messages=msgs,
                        max_tokens=rng.randint(4, 32),
                        temperature=rng.choice([0.0, 0.8]), timeout=60)
                    with lock:
                        stats["ok"] += 1
            except Exception as e:                       # noqa: BLE001
                with lock:
                    stats["stream_err" if stream else "err"] += 1
                    key = f"{type(e).__name__}:{str(e)[:60]}"
                    n = err_kinds.get(key, 0)
                    err_kinds[key] = n + 1
                    if n < 2:
                        err_samples.append(f"{time.time()-t0:.0f}s {e!r:.160}")

    def _post(path, body, timeout=45):
        r = urllib.request.Request(
            base + path, method="POST", data=json.dumps(body).encode(),
            headers={"content-type": "application/json"})
        with urllib.request.urlopen(r, timeout=timeout) as resp:
            return json.loads(resp.read().decode())

    t0 = time.time()
    threads = [threading.Thread(target=worker, args=(i,), daemon=True)
               for i in range(n_threads)]
    for t in threads:
        t.start()
    killed = 0
    migrations = 0
    sweep_fails = 0
    last_mig = 0.0
    kill_at = (duration_s * 0.35, duration_s * 0.7)
    while time.time() - t0 < duration_s:
        time.sleep(1)
        el = time.time() - t0
        if killed < 2 and el > kill_at[killed]:
            dev = f"gpu:{killed}"
            print(f"[{el:.0f}s] killing {dev}", flush=True)
            try:
                _post("/admin/fault", {"device": dev, "mode": "kill"},
                      timeout=15)
            except Exception as e:                       # noqa: BLE001
                print(f"[{el:.0f}s] kill failed: {e!r:.120}", flush=True)
            killed += 1
        if el - last_mig > 3:
            last_mig = el
            src = f"gpu:{migrations % 2}"
            dst = f"gpu:{(migrations + 1) % 2}"
            ts = time.time()
            try:
                try:
                    body = _post("/admin/drain",
                                 {"worker": src, "migrate_to": dst,
                                  "timeout_s": 10})
                finally:
                    _post("/admin/drain", {"worker": src, "drain": False})
                migrations += 1
                print(f"[{el:.0f}s] sweep {src}->{dst} took "
                      f"{time.time()-ts:.1f}s migrated="
                      f"{len(body['migrated'])} "
                      f"errors={len(body['migrate_errors'])}"
                      + (f" first_err={body['migrate_errors'][0]!r:.140}"
                         if body['migrate_errors'] else ""), flush=True)
            except Exception as e:                       # noqa: BLE001
                sweep_fails += 1
                print(f"[{el:.0f}s] sweep {src}->{dst} FAILED after "
                      f"{time.time()-ts:.1f}s: {e!r:.140}", flush=True)
        if int(el) % 10 == 0:
            with lock:
                print(f"[{el:.0f}s] {dict(stats)}", flush=True)
            try:
                with urllib.request.urlopen(base + "/admin/router",
                                            timeout=10) as r:
                    deps = json.loads(r.read().decode())["deployments"]
                print("  " + "; ".join(
                    f"{d['model_id']}: h={d['healthy']} "
                    f"cd={d['cooldown_remaining']:.1f} "
                    f"ncd={d.get('total_cooldowns', '?')} "
                    f"fl={d.get('recent_fails', '?')}"
                    for d in deps), flush=True)
            except Exception:                            # noqa: BLE001
                pass
    stop.set()
    for t in threads:
        t.join(timeout=70)
    time.sleep(1)
    with urllib.request.urlopen(base + "/admin/router", timeout=15) as r:
        rows = {d["model_id"]: d
                for d in json.loads(r.read().decode())["deployments"]}
    healthy = {k: d["healthy"] for k, d in rows.items()}
    inflight = {k: d["in_flight"] for k, d in rows.items()}
    total = sum(stats.values())
    ok = stats["ok"] + stats["stream_ok"]
    print(f"SOAK-CPU DONE: {ok}/{total} ok ({100*ok/max(total,1):.1f}%), "
          f"{killed} kills, {migrations} sweeps ok, {sweep_fails} sweeps "
          f"failed; healthy={healthy}; in_flight={inflight}", flush=True)
    if err_samples:
        print("error samples (first 2 per kind):", flush=True)
        for s in err_samples[:40]:
            print("  " + s, flush=True)
        print("error kind counts:", flush=True)
        for k, v in sorted(err_kinds.items(), key=lambda kv: -kv[1]):
            print(f"  {v:6d}  {k}", flush=True)
    loop.call_soon_threadsafe(holder["stop"].set)
    th.join(timeout=30)
    for i in range(2):
        w = holder.get(f"w{i}")
        if w is not None and w.proc is not None and w.proc.poll() is None:
            w.proc.kill()
    return 0 if (ok / max(total, 1)) > 0.95 and all(healthy.values()) else 1


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--duration", type=int, default=60)
    ap.add_argument("--threads", type=int, default=8)
    a = ap.parse_args()
    raise SystemExit(main(duration_s=a.duration, n_threads=a.threads))
