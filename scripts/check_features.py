#!/usr/bin/env python3
"""Smoke the feature-showcase config end-to-end on one MI355X:
fp8 KV + W8A8-fp8 + spec_lookup + prefix-affinity + live migration."""
import json
import os
import sys
import threading
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


def main():
    args = SimpleNamespace(base_url=None,
                           config=os.path.join(REPO, "config",
                                               "config.features.yaml"),
                           gpu=True)
    with common.gateway_session(args) as (client, config):
        base = f"http://127.0.0.1:{config.cluster.port}"
        ctx = "shared system context " * 40
        devs = []
        ok = 0
        for i in range(12):
            r = client.chat.completions.create(
                model="llama-3-8b",
                messages=[{"role": "system", "content": ctx},
                          {"role": "user", "content": f"q{i}"}],
                max_tokens=24, timeout=180)
            ok += 1
            devs.append(r.headers.get("x-gateway-device"))
        st = client.chat.completions.create(
            model="llama-3-8b",
            messages=[{"role": "user", "content": "stream " * 30}],
            max_tokens=64, stream=True, timeout=180,
            extra_headers={"x-request-id": "feat-mig"})
        got_err = []

        def consume():
            try:
                st.collect_text()
            except Exception as e:                    # noqa: BLE001
                got_err.append(repr(e))

        t = threading.Thread(target=consume)
        t.start()
        src = devs[0] or "gpu:0"
        dst = "gpu:0.1" if src == "gpu:0" else "gpu:0"
        req = urllib.request.Request(
            base + "/admin/drain", method="POST",
            data=json.dumps({"worker": dst, "drain": True}).encode(),
            headers={"content-type": "application/json"})
        urllib.request.urlopen(req, timeout=60)      # plain drain toggle
        urllib.request.urlopen(urllib.request.Request(
            base + "/admin/drain", method="POST",
            data=json.dumps({"worker": dst, "drain": False}).encode(),
            headers={"content-type": "application/json"}), timeout=60)
        t.join(timeout=180)
        aff_ok = len(set(devs)) == 1      # one shared context -> one replica
        print(f"FEATURES CHECK: {ok}/12 ok, affinity sticky={aff_ok} "
              f"(devices={set(devs)}), stream_err={got_err}")
        return 0 if (ok == 12 and aff_ok and not got_err) else 1


if __name__ == "__main__":
    raise SystemExit(main())
