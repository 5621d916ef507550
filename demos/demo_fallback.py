#!/usr/bin/env python3
"""Fallback demo — automatic re-route to a backup deployment.

Reference analogue: src/demo_fallback.py (C3 in SURVEY.md §2.1) — 10
concurrent requests with a 0.05 s stagger to an alias whose primary
deployment has rpm=3; the router transparently fails the overflow over to
the backup (X5), the client sees success with the fallback deployment in
``response.model`` (reference README.md:156-165).  Here "primary" and
"backup" are deployments on different MI355X GPUs; with ``--fault`` the
demo additionally KILLS the primary worker mid-run and shows hot failover
(the reference could only starve quotas — SURVEY.md §5.3).

Usage: python demos/demo_fallback.py [--requests N] [--fault]
"""

from __future__ import annotations

import argparse
import concurrent.futures as cf
import sys
import threading
import time

from _common import (
    add_common_args, gateway_session, log_with_timestamp, print_router_settings,
    print_table,
)

from resilient_llm_amd.client import RateLimitError
from resilient_llm_amd.utils.logging import sanitize_error

ALIAS = "llama-fallback-demo"
QUESTIONS = [
    "What is 2+2?", "Name a color.", "What is the capital of France?",
    "Say hello.", "What is water made of?", "Name a planet.",
    "What is 10*10?", "Name an animal.", "What day follows Monday?",
    "What is the opposite of hot?",
]


def trigger_rate_limit_scenario(args) -> dict:
    with gateway_session(args) as (client, config):
        print_router_settings(client)
        n = args.requests
        log_with_timestamp(
            f"Firing {n} concurrent requests at {ALIAS!r} "
            f"(primary rpm=3 -> overflow falls back)", "blue")

        results: list[dict] = []
        model_usage: dict[str, int] = {}
        lock = threading.Lock()

        def worker(req_id: int) -> None:
            t0 = time.time()
            try:
                r = client.chat.completions.create(
                    model=ALIAS,
                    messages=[{"role": "user",
                               "content": QUESTIONS[req_id % len(QUESTIONS)]}],
                    max_tokens=16, timeout=30)
                latency = time.time() - t0
                model_used = r.model
                is_fallback = r.was_fallback
                with lock:
                    results.append({"id": req_id, "status": "success",
                                    "model": model_used, "fallback": is_fallback,
                                    "latency": latency})
                    model_usage[model_used] = model_usage.get(model_used, 0) + 1
                tag = "FALLBACK" if is_fallback else "PRIMARY"
                log_with_timestamp(
                    f"  request {req_id:2d}: {tag:8s} via {model_used} "
                    f"({latency:.2f}s)", "yellow" if is_fallback else "white")
            except RateLimitError:
                with lock:
                    results.append({"id": req_id, "status": "rate_limited"})
                log_with_timestamp(f"  request {req_id:2d}: RATE LIMITED", "red")
            except Exception as e:
                with lock:
                    results.append({"id": req_id, "status": "error",
                                    "error": sanitize_error(e)})
                log_with_timestamp(
                    f"  request {req_id:2d}: ERROR ({sanitize_error(e)})", "red")

        threads = []
        fault_injected = False
        with cf.ThreadPoolExecutor(max_workers=n) as ex:
            for i in range(n):
                threads.append(ex.submit(worker, i))
                time.sleep(0.05)        # stagger (reference demo_fallback.py:227-231)
                if args.fault and not fault_injected and i == n // 2:
                    # kill the primary's worker mid-run -> hot failover
                    primary = config.deployments_for(ALIAS)[0]
                    dev = f"{primary.backend_kind}:{primary.backend_target}"
                    log_with_timestamp(f"INJECTING FAULT: killing {dev}", "magenta")
                    client.inject_fault(dev, "kill")
                    fault_injected = True
            cf.wait(threads)
        if fault_injected:
            primary = config.deployments_for(ALIAS)[0]
            client.inject_fault(
                f"{primary.backend_kind}:{primary.backend_target}", "none")

        ok = [r for r in results if r["status"] == "success"]
        fallbacks = [r for r in ok if r["fallback"]]
        primaries = [r for r in ok if not r["fallback"]]
        print_table(
            ["Model", "Requests", "Share"],
            [[m, c, f"{100.0 * c / len(ok):.0f}%"] for m, c in
             sorted(model_usage.items(), key=lambda kv: -kv[1])] if ok else [],
            title="Model Usage Distribution")
        if fallbacks:
            print_table(
                ["Request", "Served by", "Latency"],
                [[r["id"], r["model"], f"{r['latency']:.2f}s"] for r in fallbacks],
                title="Fallback Events")
            log_with_timestamp(
                f"FALLBACK WORKING: {len(fallbacks)} requests successfully "
                f"failed over to backup", "green")
        else:
            log_with_timestamp(
                "No fallbacks triggered (primary absorbed all requests)", "yellow")
        log_with_timestamp(
            f"Summary: {len(ok)}/{n} succeeded "
            f"({len(primaries)} primary + {len(fallbacks)} fallback)",
            "green" if len(ok) == n else "yellow")
        return {"total": n, "success": len(ok), "primary": len(primaries),
                "fallback": len(fallbacks),
                "rate_limited": sum(1 for r in results if r["status"] == "rate_limited"),
                "errors": sum(1 for r in results if r["status"] == "error"),
                "fault_injected": fault_injected}


def main() -> int:
    ap = argparse.ArgumentParser(description="Fallback / hot-failover demo")
    ap.add_argument("--requests", type=int, default=10)
    ap.add_argument("--fault", action="store_true",
                    help="kill the primary worker mid-run (hot failover)")
    add_common_args(ap)
    args = ap.parse_args()
    stats = trigger_rate_limit_scenario(args)
    return 0 if stats["success"] == stats["total"] and stats["fallback"] > 0 else 1


if __name__ == "__main__":
    sys.exit(main())
