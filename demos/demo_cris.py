#!/usr/bin/env python3
"""Cross-GPU Inference demo — the CRIS pattern on one MI355X node.

Reference analogue: src/demo_cris.py (C1 in SURVEY.md §2.1) — N concurrent
requests to ONE endpoint, transparently spread, then a post-hoc
distribution table.  "Region" becomes GPU: the scheduler places each
request on whichever worker has the fewest in flight (capacity-driven,
not client-controlled — X10), and the CloudWatch Logs Insights query
(reference demo_cris.py:87-92, with its 60 s propagation retry ladder at
381-405) becomes ONE synchronous read of the gateway's invocation ledger
(/admin/distribution) — same table, no propagation delay.

Usage: python demos/demo_cris.py [--requests N] [--base-url URL]
"""

from __future__ import annotations

import argparse
import concurrent.futures as cf
import sys
import time

from _common import (
    add_common_args, gateway_session, log_with_timestamp, print_table,
)

from resilient_llm_amd.client import OpenAIClient, RateLimitError

QUESTIONS = [
    "Summarize the benefits of cross-device inference in one sentence.",
    "What is a token bucket?",
    "Explain weighted load balancing briefly.",
    "Why do health checks matter for serving?",
    "What is tensor parallelism?",
]


def send_request(client: OpenAIClient, model: str, req_id: int) -> dict:
    """One request; classify throttling vs hard errors like the reference
    taxonomy (demo_cris.py:261-283)."""
    t0 = time.time()
    try:
        r = client.chat.completions.create(
            model=model,
            messages=[{"role": "user", "content": QUESTIONS[req_id % len(QUESTIONS)]}],
            max_tokens=32, timeout=60)
        return {"id": req_id, "status": "success",
                "latency": time.time() - t0,
                "device": r.device_header, "model_id": r.model_id_header}
    except RateLimitError:
        return {"id": req_id, "status": "throttled", "latency": time.time() - t0}
    except Exception as e:
        from resilient_llm_amd.utils.logging import sanitize_error
        return {"id": req_id, "status": "error", "latency": time.time() - t0,
                "error": sanitize_error(e)}


def analyze_results(client: OpenAIClient, model: str, n_success: int,
                    since_s: float) -> dict:
    """The Logs-Insights analogue: count(*) by device, synchronously."""
    dist = client.distribution(by="device", alias=model, since_s=since_s)
    total = dist["total"]
    log_with_timestamp(
        f"ledger reports {total} invocations for alias {model!r}", "cyan")
    rows = [[dev, count, f"{dist['percentages'][dev]}%"]
            for dev, count in dist["distribution"].items()]
    print_table(["GPU (device)", "Invocations", "Percentage"], rows,
                title="Cross-GPU Request Distribution")
    complete = total == n_success   # completeness gate (demo_cris.py:398)
    if complete:
        log_with_timestamp(
            f"COMPLETE: ledger count matches {n_success} successful requests",
            "green")
    else:
        log_with_timestamp(
            f"INCOMPLETE: ledger has {total}, expected {n_success}", "yellow")
    return {"distribution": dist["distribution"], "complete": complete,
            "stats": dist["stats"]}


def run_cris_demo(args) -> dict:
    n = args.requests
    with gateway_session(args) as (client, config):
        model = config.cris_model or "llama-cris-demo"
        t_start = time.time()
        log_with_timestamp(
            f"Sending {n} concurrent requests to cross-GPU alias {model!r}", "blue")
        with cf.ThreadPoolExecutor(max_workers=n) as ex:
            results = list(ex.map(lambda i: send_request(client, model, i), range(n)))
        ok = [r for r in results if r["status"] == "success"]
        throttled = [r for r in results if r["status"] == "throttled"]
        errors = [r for r in results if r["status"] == "error"]
        lat = [r["latency"] for r in ok]
        log_with_timestamp(
            f"Done: {len(ok)} ok, {len(throttled)} throttled, {len(errors)} errors; "
            f"avg latency {sum(lat)/len(lat):.2f}s" if lat else "Done: no successes",
            "green" if len(ok) == n else "yellow")
        analysis = analyze_results(client, model, len(ok),
                                   since_s=time.time() - t_start + 5)
        n_devices = len(analysis["distribution"])
        if n_devices > 1:
            log_with_timestamp(
                f"CROSS-GPU INFERENCE WORKING: requests spread across "
                f"{n_devices} devices", "green")
        elif n_devices == 1:
            log_with_timestamp(
                "All requests landed on one device (single-device config?)", "yellow")
        return {"total": n, "success": len(ok), "throttled": len(throttled),
                "errors": len(errors),
                "avg_latency": (sum(lat) / len(lat)) if lat else None,
                **analysis}


def main() -> int:
    ap = argparse.ArgumentParser(description="Cross-GPU inference (CRIS) demo")
    ap.add_argument("--requests", type=int, default=20,
                    help="number of concurrent requests (1-100)")
    add_common_args(ap)
    args = ap.parse_args()
    if not 1 <= args.requests <= 100:
        print("error: --requests must be between 1 and 100", file=sys.stderr)
        return 2
    stats = run_cris_demo(args)
    return 0 if stats["success"] > 0 and stats["complete"] else 1


if __name__ == "__main__":
    sys.exit(main())
