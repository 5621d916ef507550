#!/usr/bin/env python3
"""Measure prefix-affinity routing: N distinct long system prompts
(RAG/agent contexts), 2 co-located replicas on one GPU.  With
simple-shuffle every context keeps landing on BOTH replicas, so each
replica re-prefills prefixes the other already cached; with
prefix-affinity each context sticks to one replica and its engine
prefix cache serves the shared blocks.  Reports request latency and the
per-gateway prefix-cache hit rate scraped from /metrics.
"""
import argparse
import concurrent.futures as cf
import os
import re
import sys
import tempfile
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

import yaml  # noqa: E402


def config_for(strategy):
    return {
        "cluster": {"port": 4321},
        "model_list": [
            {"model_name": "aff",
             "litellm_params": {"model": "gpu/0/llama-3-8b"},
             "model_info": {"id": "gpu0/aff"}},
            {"model_name": "aff",
             "litellm_params": {"model": "gpu/0.1/llama-3-8b"},
             "model_info": {"id": "gpu0.1/aff"}},
        ],
        "router_settings": {"routing_strategy": strategy,
                            "enable_pre_call_checks": False},
    }


def scrape_prefix(base_url):
    txt = urllib.request.urlopen(base_url + "/metrics", timeout=10).read().decode()
    out = {}
    for metric in ("worker_prefix_hits", "worker_prefix_lookups"):
        for m in re.finditer(rf'{metric}{{worker="([^"]+)"}} (\d+)', txt):
            out.setdefault(m.group(1), {})[metric] = int(m.group(2))
    return out


def run(strategy, n_contexts, reqs_per_ctx, ctx_chars):
    with tempfile.NamedTemporaryFile("w", suffix=".yaml", delete=False) as f:
        yaml.safe_dump(config_for(strategy), f)
        path = f.name
    args = SimpleNamespace(base_url=None, config=path, gpu=True)
    with common.gateway_session(args) as (client, config):
        base = f"http://127.0.0.1:{config.cluster.port}"
        contexts = [f"You are assistant {c}. " + (f"ctx{c} " * (ctx_chars // 6))
                    for c in range(n_contexts)]
        for c, ctx in enumerate(contexts):        # warm: build each cache once
            client.chat.completions.create(model="aff", messages=[
                {"role": "system", "content": ctx},
                {"role": "user", "content": "warm"}], max_tokens=4)

        def one(i):
            c = i % n_contexts
            t0 = time.monotonic()
            client.chat.completions.create(model="aff", messages=[
                {"role": "system", "content": contexts[c]},
                {"role": "user", "content": f"question {i}?"}], max_tokens=8)
            return time.monotonic() - t0

        with cf.ThreadPoolExecutor(8) as ex:
            lat = list(ex.map(one, range(n_contexts * reqs_per_ctx)))
        time.sleep(1.5)                            # let a health sweep land
        try:
            prefix = scrape_prefix(base)
        except Exception as e:                     # noqa: BLE001
            prefix = {"error": str(e)}
    return lat, prefix


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--contexts", type=int, default=6)
    ap.add_argument("--reqs-per-ctx", type=int, default=10)
    ap.add_argument("--ctx-chars", type=int, default=3000)
    args = ap.parse_args()
    for strat in ("simple-shuffle", "prefix-affinity"):
        t0 = time.monotonic()
        lat, prefix = run(strat, args.contexts, args.reqs_per_ctx,
                          args.ctx_chars)
        wall = time.monotonic() - t0
        lat.sort()
        hits = sum(v.get("worker_prefix_hits", 0) for v in prefix.values()
                   if isinstance(v, dict))
        lookups = sum(v.get("worker_prefix_lookups", 0)
                      for v in prefix.values() if isinstance(v, dict))
        rate = f"{hits}/{lookups}" if lookups else str(prefix)
        print(f"{strat:16s} mean {sum(lat)/len(lat)*1e3:7.1f} ms  "
              f"p50 {lat[len(lat)//2]*1e3:7.1f}  "
              f"p99 {lat[min(len(lat)-1, int(len(lat)*0.99))]*1e3:7.1f}  "
              f"prefix-hits {rate}  wall {wall:.1f}s", flush=True)


if __name__ == "__main__":
    main()
