#!/usr/bin/env python3
"""Load-balancing demo — simple-shuffle over GPU replicas.

Reference analogue: src/demo_load_balancing.py (C4 in SURVEY.md §2.1) —
10 concurrent requests (0.1 s stagger) to one alias backed by two replica
deployments; the TRUE deployment comes from the response header (the
reference reads ``x-litellm-model-id``, ours is ``x-gateway-model-id`` —
X8); per-model distribution + avg latency table; optional continuous
``--loop`` mode with cumulative stats and SIGINT-graceful shutdown
(reference demo_load_balancing.py:267-345).

Usage: python demos/demo_load_balancing.py [--requests N] [--loop --interval N]
"""

from __future__ import annotations

import argparse
import concurrent.futures as cf
import signal
import sys
import threading
import time

from _common import (
    add_common_args, gateway_session, log_with_timestamp, print_router_settings,
    print_table,
)

from resilient_llm_amd.client import OpenAIClient, RateLimitError
from resilient_llm_amd.utils.logging import sanitize_error

ALIAS = "llama-loadbalance-demo"
PROMPTS = [
    "Briefly explain load balancing.", "Name three fruits.",
    "What is a GPU?", "Define latency.", "What is a replica?",
    "Explain round robin.", "What is a health check?",
    "Define throughput.", "What is a request queue?", "Explain failover.",
]


def send_request(client: OpenAIClient, req_id: int) -> dict:
    t0 = time.time()
    try:
        r = client.chat.completions.create(
            model=ALIAS,
            messages=[{"role": "user", "content": PROMPTS[req_id % len(PROMPTS)]}],
            max_tokens=16, timeout=30)
        return {"id": req_id, "status": "success",
                "model_id": r.model_id_header or r.model,
                "device": r.device_header,
                "fallback": r.was_fallback,
                "latency": time.time() - t0}
    except RateLimitError:
        return {"id": req_id, "status": "rate_limited", "latency": time.time() - t0}
    except Exception as e:
        return {"id": req_id, "status": "error", "error": sanitize_error(e),
                "latency": time.time() - t0}


def demo_load_balancing(args, client: OpenAIClient) -> dict:
    n = args.requests
    log_with_timestamp(
        f"Sending {n} staggered concurrent requests to {ALIAS!r}", "blue")
    results: list[dict] = []
    with cf.ThreadPoolExecutor(max_workers=n) as ex:
        futures = []
        for i in range(n):
            futures.append(ex.submit(send_request, client, i))
            time.sleep(0.1)      # stagger (reference demo_load_balancing.py:215)
        results = [f.result() for f in futures]

    ok = [r for r in results if r["status"] == "success"]
    by_model: dict[str, list[dict]] = {}
    for r in ok:
        by_model.setdefault(r["model_id"], []).append(r)
    rows = []
    for model_id, rs in sorted(by_model.items(), key=lambda kv: -len(kv[1])):
        avg_lat = sum(x["latency"] for x in rs) / len(rs)
        rows.append([model_id, len(rs), f"{100.0 * len(rs) / len(ok):.0f}%",
                     f"{avg_lat:.2f}s"])
    print_table(["Deployment", "Requests", "Share", "Avg latency"], rows,
                title="Per-Deployment Distribution")
    n_models = len(by_model)
    if n_models > 1:
        log_with_timestamp(
            f"LOAD BALANCING WORKING: requests distributed across "
            f"{n_models} deployments", "green")
    else:
        log_with_timestamp("All requests served by a single deployment", "yellow")
    log_with_timestamp(
        f"Summary: {len(ok)}/{n} succeeded, "
        f"{sum(1 for r in results if r['status'] == 'rate_limited')} rate-limited, "
        f"{sum(1 for r in results if r['status'] == 'error')} errors",
        "green" if len(ok) == n else "yellow")
    return {"total": n, "success": len(ok), "models": {k: len(v) for k, v in by_model.items()},
            "working": n_models > 1}


_stop = threading.Event()


def run_loop_mode(args, client: OpenAIClient) -> dict:
    """Continuous soak mode with cumulative stats (reference
    demo_load_balancing.py:267-345)."""
    signal.signal(signal.SIGINT, lambda *a: _stop.set())
    cumulative: dict[str, int] = {}
    total = ok = rounds = 0
    log_with_timestamp(
        f"Loop mode: one round every {args.interval}s (Ctrl-C to stop)", "blue")
    while not _stop.is_set():
        rounds += 1
        stats = demo_load_balancing(args, client)
        total += stats["total"]
        ok += stats["success"]
        for m, c in stats["models"].items():
            cumulative[m] = cumulative.get(m, 0) + c
        print_table(
            ["Deployment", "Cumulative requests", "Share"],
            [[m, c, f"{100.0 * c / max(1, ok):.0f}%"]
             for m, c in sorted(cumulative.items(), key=lambda kv: -kv[1])],
            title=f"Cumulative after round {rounds} "
                  f"({ok}/{total} ok)")
        if _stop.wait(timeout=args.interval):
            break
    log_with_timestamp("Loop stopped.", "yellow")
    return {"rounds": rounds, "total": total, "success": ok,
            "models": cumulative}


def main() -> int:
    ap = argparse.ArgumentParser(description="Load-balancing demo")
    ap.add_argument("--requests", type=int, default=10)
    ap.add_argument("--loop", action="store_true")
    ap.add_argument("--interval", type=int, default=30,
                    help="seconds between loop rounds (>=5)")
    add_common_args(ap)
    args = ap.parse_args()
    if args.loop and args.interval < 5:
        print("error: --interval must be >= 5", file=sys.stderr)
        return 2
    with gateway_session(args) as (client, _):
        print_router_settings(client)
        if args.loop:
            stats = run_loop_mode(args, client)
            return 0
        stats = demo_load_balancing(args, client)
        return 0 if stats["working"] and stats["success"] == stats["total"] else 1


if __name__ == "__main__":
    sys.exit(main())
