#!/usr/bin/env python3
"""Quota-isolation demo — a noisy consumer cannot starve the others.

Reference analogue: src/demo_quota_isolation.py (C5 in SURVEY.md §2.1) —
three consumers (A noisy rpm=3, B and C normal rpm=10) each fire 5
parallel requests simultaneously (nested thread pools, reference
demo_quota_isolation.py:221-239 × 150-154); consumer -> model mapping;
RateLimitError -> 429 counted per consumer; verdict
``isolation_effective = B>=80% and C>=80% success`` (demo_quota_isolation.py:309).

Usage: python demos/demo_quota_isolation.py [--loop --interval N]
"""

from __future__ import annotations

import argparse
import concurrent.futures as cf
import signal
import sys
import threading
import time

from _common import (
    add_common_args, gateway_session, log_with_timestamp, print_table,
)

from resilient_llm_amd.client import OpenAIClient, RateLimitError
from resilient_llm_amd.utils.logging import sanitize_error

# consumer -> (api_key, model alias) — mirrors reference demo_quota_isolation.py:61-65
CONSUMERS = {
    "Consumer-A (noisy)": ("sk-consumer-a", "consumer-a-model"),
    "Consumer-B": ("sk-consumer-b", "consumer-b-model"),
    "Consumer-C": ("sk-consumer-c", "consumer-c-model"),
}
REQUESTS_PER_CONSUMER = 5


def send_consumer_request(base_url: str, api_key: str, model: str,
                          req_id: int) -> dict:
    client = OpenAIClient(base_url, api_key=api_key)
    t0 = time.time()
    try:
        client.chat.completions.create(
            model=model,
            messages=[{"role": "user", "content": f"Request {req_id}: say ok."}],
            max_tokens=8, timeout=10)
        return {"status": "success", "latency": time.time() - t0}
    except RateLimitError:
        return {"status": "rate_limited", "error": "RateLimitError",
                "latency": time.time() - t0}
    except Exception as e:
        return {"status": "error", "error": sanitize_error(e),
                "latency": time.time() - t0}


def run_consumer_workload(base_url: str, name: str, api_key: str,
                          model: str) -> dict:
    with cf.ThreadPoolExecutor(max_workers=REQUESTS_PER_CONSUMER) as ex:
        results = list(ex.map(
            lambda i: send_consumer_request(base_url, api_key, model, i),
            range(REQUESTS_PER_CONSUMER)))
    ok = [r for r in results if r["status"] == "success"]
    return {
        "name": name, "model": model,
        "total": len(results), "success": len(ok),
        "rate_limited": sum(1 for r in results if r["status"] == "rate_limited"),
        "errors": sum(1 for r in results if r["status"] == "error"),
        "success_rate": 100.0 * len(ok) / len(results),
        "avg_latency": (sum(r["latency"] for r in ok) / len(ok)) if ok else None,
    }


def print_quota_table(client: OpenAIClient) -> None:
    state = client.router_state()
    rows = [[d["model_name"], d["rpm"] or "-", d["tpm"] or "-"]
            for d in state["deployments"]
            if d["model_name"].startswith("consumer-")]
    print_table(["Consumer model", "RPM limit", "TPM limit"], rows,
                title="Per-Consumer Quotas")


def demo_quota_isolation(args, client: OpenAIClient, base_url: str) -> dict:
    print_quota_table(client)
    log_with_timestamp(
        f"All 3 consumers fire {REQUESTS_PER_CONSUMER} parallel requests "
        f"simultaneously", "blue")
    t0 = time.time()
    with cf.ThreadPoolExecutor(max_workers=len(CONSUMERS)) as ex:
        futures = {ex.submit(run_consumer_workload, base_url, name, key, model): name
                   for name, (key, model) in CONSUMERS.items()}
        by_consumer = {futures[f]: f.result() for f in cf.as_completed(futures)}
    wall = time.time() - t0
    return analyze_and_display_results(by_consumer, wall)


def analyze_and_display_results(by_consumer: dict[str, dict], wall: float) -> dict:
    rows, colors = [], []
    for name in CONSUMERS:
        s = by_consumer[name]
        rows.append([name, s["model"], f"{s['success']}/{s['total']}",
                     f"{s['success_rate']:.0f}%", s["rate_limited"],
                     f"{s['avg_latency']:.2f}s" if s["avg_latency"] else "-"])
        colors.append("red" if s["success_rate"] < 80 else None)
    print_table(["Consumer", "Model", "Success", "Rate", "429s", "Avg latency"],
                rows, title="Quota Isolation Results", colors=colors)

    a = by_consumer["Consumer-A (noisy)"]
    b = by_consumer["Consumer-B"]
    c = by_consumer["Consumer-C"]
    isolation_effective = b["success_rate"] >= 80 and c["success_rate"] >= 80
    if isolation_effective:
        log_with_timestamp(
            f"QUOTA ISOLATION EFFECTIVE: noisy consumer throttled to "
            f"{a['success_rate']:.0f}% while B={b['success_rate']:.0f}% "
            f"and C={c['success_rate']:.0f}%", "green")
    else:
        log_with_timestamp(
            f"ISOLATION FAILED: B={b['success_rate']:.0f}%, "
            f"C={c['success_rate']:.0f}%", "red")
    log_with_timestamp(f"Demo wall clock: {wall:.1f}s", "grey")
    return {"consumers": by_consumer, "isolation_effective": isolation_effective,
            "wall_s": wall}


_stop = threading.Event()


def run_loop_mode(args, client: OpenAIClient, base_url: str) -> dict:
    signal.signal(signal.SIGINT, lambda *a: _stop.set())
    rounds = effective = 0
    while not _stop.is_set():
        rounds += 1
        log_with_timestamp(f"--- round {rounds} ---", "blue")
        stats = demo_quota_isolation(args, client, base_url)
        effective += int(stats["isolation_effective"])
        log_with_timestamp(
            f"Cumulative: isolation effective in {effective}/{rounds} rounds",
            "cyan")
        if _stop.wait(timeout=args.interval):
            break
    return {"rounds": rounds, "effective": effective}


def main() -> int:
    ap = argparse.ArgumentParser(description="Quota isolation demo")
    ap.add_argument("--loop", action="store_true")
    ap.add_argument("--interval", type=int, default=30,
                    help="seconds between rounds (>=5; quotas reset per minute)")
    add_common_args(ap)
    args = ap.parse_args()
    if args.loop and args.interval < 5:
        print("error: --interval must be >= 5", file=sys.stderr)
        return 2
    with gateway_session(args) as (client, config):
        base_url = f"http://{client.host}:{client.port}"
        if args.loop:
            run_loop_mode(args, client, base_url)
            return 0
        stats = demo_quota_isolation(args, client, base_url)
        return 0 if stats["isolation_effective"] else 1


if __name__ == "__main__":
    sys.exit(main())
