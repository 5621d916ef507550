#!/usr/bin/env python3
"""Mixed-load soak: streaming + non-streaming + temperatures + mid-run
worker kill/respawn against a 2-replica single-GPU gateway.  Reports
success rates and verifies no in-flight leaks at the end."""
import os
import random
import sys
import threading
import time
from types import SimpleNamespace

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "demos"))
import importlib.util
spec = importlib.util.spec_from_file_location(
    "_common", os.path.join(REPO, "demos", "_common.py"))
common = importlib.util.module_from_spec(spec)
spec.loader.exec_module(common)

from resilient_llm_amd.client import APIError, OpenAIClient  # noqa: E402

# RLLI_SOAK_MIGRATE=1: every ~15 s, live-migrate EVERYTHING off one
# replica onto the other (drain with migrate_to) and undrain — layered
# on top of the kill/respawn chaos
def _mode_opts():
    """RLLI_SOAK_MODE=fp8 soaks the quantized-weight + fp8-KV modes."""
    if os.environ.get("RLLI_SOAK_MODE") == "fp8":
        return {"quantization": "fp8", "kv_dtype": "fp8"}
    return {}


CONFIG = {
    "cluster": {"port": 4999},
    "model_list": [
        {"model_name": "soak",
         "litellm_params": dict({"model": "gpu/0/llama-3-8b"},
                                **_mode_opts()),
         "model_info": {"id": "gpu0/soak"}},
        {"model_name": "soak",
         "litellm_params": dict({"model": "gpu/0.1/llama-3-8b"},
                                **_mode_opts()),
         "model_info": {"id": "gpu0.1/soak"}},
    ],
    "router_settings": {"routing_strategy": "simple-shuffle",
                        "enable_pre_call_checks": False},
}


def main(duration_s=420, n_threads=24):
    import yaml, tempfile
    cfgfile = os.path.join(tempfile.mkdtemp(), "soak.yaml")
    with open(cfgfile, "w") as f:
        yaml.safe_dump(CONFIG, f)
    args = SimpleNamespace(base_url=None, config=cfgfile, gpu=True)
    stats = {"ok": 0, "err": 0, "stream_ok": 0, "stream_err": 0}
    err_kinds: dict = {}
    lock = threading.Lock()
    stop = threading.Event()

    with common.gateway_session(args) as (client, config):
        base = f"http://{client.host}:{client.port}"

        def worker(tid):
            rng = random.Random(tid)
            c = OpenAIClient(base, api_key=f"sk-{tid}")
            while not stop.is_set():
                # 10% long prompts: chunked prefill + split-K decode
                # under churn, not just short-prompt steady state
                n_chars = (rng.randint(2000, 8000) if rng.random() < 0.1
                           else rng.randint(16, 400))
                msgs = [{"role": "user", "content": "x" * n_chars}]
                stream = rng.random() < 0.3
                try:
                    if stream:
                        s = c.chat.completions.create(
                            model="soak", messages=msgs,
                            max_tokens=rng.randint(4, 48), stream=True,
                            temperature=rng.choice([0.0, 0.8]), timeout=120)
                        text, _ = s.collect_text()
                        with lock:
                            stats["stream_ok"] += 1
                    else:
                        c.chat.completions.create(
                            model="soak", messages=msgs,
                            max_tokens=rng.randint(4, 48),
                            temperature=rng.choice([0.0, 0.8]),
                            top_p=rng.choice([1.0, 0.9]), timeout=120)
                        with lock:
                            stats["ok"] += 1
                except Exception as e:                 # noqa: BLE001
                    with lock:
                        stats["stream_err" if stream else "err"] += 1
                        key = f"{type(e).__name__}:{str(e)[:70]}"
                        err_kinds[key] = err_kinds.get(key, 0) + 1
                    # real clients back off on failure; without this a
                    # brief outage turns into a tight error loop that
                    # swamps the gateway (and the error counts)
                    time.sleep(0.2)

        threads = [threading.Thread(target=worker, args=(i,), daemon=True)
                   for i in range(n_threads)]
        for t in threads:
            t.start()
        t0 = time.time()
        killed = 0
        migrations = 0
        sweeps_ok = 0
        mig_on = os.environ.get("RLLI_SOAK_MIGRATE") == "1"
        last_mig = 0.0
        import json as _json
        import urllib.request as _url

        def _post(path, body):
            r = _url.Request(
                f"http://127.0.0.1:{config.cluster.port}{path}",
                method="POST", data=_json.dumps(body).encode(),
                headers={"content-type": "application/json"})
            with _url.urlopen(r, timeout=60) as resp:
                return _json.loads(resp.read().decode())

        while time.time() - t0 < duration_s:
            time.sleep(5)
            el = time.time() - t0
            if killed < 2 and el > 120 * (killed + 1):
                dev = "gpu:0" if killed == 0 else "gpu:0.1"
                print(f"[soak {el:.0f}s] killing {dev}", flush=True)
                try:
                    client.inject_fault(dev, "kill")
                except APIError:
                    pass
                killed += 1
            if mig_on and el - last_mig > 15:
                last_mig = el
                src = "gpu:0" if migrations % 2 == 0 else "gpu:0.1"
                dst = "gpu:0.1" if src == "gpu:0" else "gpu:0"
                try:
                    try:
                        body = _post("/admin/drain", {"worker": src,
                                                      "migrate_to": dst,
                                                      "timeout_s": 20})
                    finally:
                        # NEVER leave a replica draining: a failed sweep
                        # plus a later sweep of the OTHER replica would
                        # drain the whole pool
                        _post("/admin/drain", {"worker": src,
                                               "drain": False})
                    migrations += 1
                    sweeps_ok += 1
                    print(f"[soak {el:.0f}s] migrated "
                          f"{len(body['migrated'])} reqs {src}->{dst} "
                          f"(errors: {len(body['migrate_errors'])})",
                          flush=True)
                except Exception as e:             # noqa: BLE001
                    # count failed sweeps too so src alternates instead
                    # of hammering the same (possibly sick) replica
                    migrations += 1
                    print(f"[soak {el:.0f}s] migration sweep failed: {e}",
                          flush=True)
            with lock:
                print(f"[soak {el:.0f}s] {dict(stats)}", flush=True)
        stop.set()
        for t in threads:
            t.join(timeout=130)
        time.sleep(2)
        rows = {r["model_id"]: r for r in client.router_state()["deployments"]}
        inflight = {k: r["in_flight"] for k, r in rows.items()}
        healthy = {k: r["healthy"] for k, r in rows.items()}
        total = sum(stats.values())
        ok = stats["ok"] + stats["stream_ok"]
        print(f"SOAK DONE: {ok}/{total} ok ({100*ok/max(total,1):.1f}%), "
              f"{killed} kills+respawns, {sweeps_ok}/{migrations} migration sweeps ok; "
              f"in_flight={inflight}; "
              f"healthy={healthy}", flush=True)
        for k, v in sorted(err_kinds.items(), key=lambda kv: -kv[1])[:8]:
            print(f"  {v:6d}  {k}", flush=True)


if __name__ == "__main__":
    import argparse
    ap = argparse.ArgumentParser()
    ap.add_argument("--duration", type=int, default=420)
    ap.add_argument("--threads", type=int, default=24)
    a = ap.parse_args()
    main(duration_s=a.duration, n_threads=a.threads)
