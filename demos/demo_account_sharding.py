#!/usr/bin/env python3
"""Account-sharding demo — client-side split across two isolated GPU pools.

Reference analogue: src/demo_account_sharding.py (C2 in SURVEY.md §2.1) —
N requests split across two AWS accounts by strategy (round-robin / split /
random, reference demo_account_sharding.py:344-352), identities verified
via STS before starting (59-69, hard exit if either fails 312-317, warning
if both are the same account 319-321), then per-account success/latency
tables and per-account regional distributions queried in parallel
(467-517).

Here an "account" is an isolated GPU pool with its own quota domain
(X11): ``pool-a-model`` / ``pool-b-model`` aliases whose deployments live
on disjoint devices.  The STS identity check becomes a pool health +
disjointness probe; CloudWatch becomes the ledger distribution query.

Usage: python demos/demo_account_sharding.py [--requests N]
       [--strategy round-robin|split|random]
"""

from __future__ import annotations

import argparse
import concurrent.futures as cf
import random
import sys
import time

from _common import (
    add_common_args, gateway_session, log_with_timestamp, print_table,
)

from resilient_llm_amd.client import OpenAIClient, RateLimitError
from resilient_llm_amd.utils.logging import sanitize_error

POOLS = {"pool-a": "pool-a-model", "pool-b": "pool-b-model"}


def verify_pool_identity(client: OpenAIClient, alias: str) -> dict:
    """The STS get-caller-identity analogue: the pool must exist, have at
    least one healthy deployment, and report its device set."""
    state = client.router_state()
    deps = [d for d in state["deployments"] if d["model_name"] == alias]
    healthy = [d for d in deps if d["healthy"]]
    devices = {d["backend"].split("/")[1] for d in deps}
    return {"alias": alias, "deployments": len(deps),
            "healthy": len(healthy), "devices": devices,
            "ok": bool(healthy)}


def send_pool_request(base_url: str, alias: str, pool: str, req_id: int) -> dict:
    client = OpenAIClient(base_url, api_key=f"sk-{pool}")
    t0 = time.time()
    try:
        r = client.chat.completions.create(
            model=alias,
            messages=[{"role": "user", "content": f"Request {req_id}: reply briefly."}],
            max_tokens=16, timeout=30)
        return {"id": req_id, "pool": pool, "status": "success",
                "latency": time.time() - t0, "device": r.device_header,
                "model_id": r.model_id_header}
    except RateLimitError:
        return {"id": req_id, "pool": pool, "status": "throttled",
                "latency": time.time() - t0}
    except Exception as e:
        return {"id": req_id, "pool": pool, "status": "error",
                "error": sanitize_error(e), "latency": time.time() - t0}


def pick_pool(strategy: str, req_id: int, n: int, rng: random.Random) -> str:
    # mirrors reference demo_account_sharding.py:344-352
    pools = list(POOLS)
    if strategy == "round-robin":
        return pools[req_id % 2]
    if strategy == "split":
        return pools[0] if req_id < n // 2 else pools[1]
    return rng.choice(pools)


def run_cross_pool_demo(args) -> dict:
    n = args.requests
    rng = random.Random(args.seed)
    with gateway_session(args) as (client, config):
        base_url = f"http://{client.host}:{client.port}"

        # --- identity verification (STS analogue) --------------------
        idents = {}
        for pool, alias in POOLS.items():
            ident = verify_pool_identity(client, alias)
            idents[pool] = ident
            log_with_timestamp(
                f"{pool}: alias {alias!r} -> {ident['healthy']}/{ident['deployments']} "
                f"healthy deployments on devices {sorted(ident['devices'])}",
                "green" if ident["ok"] else "red")
        if not all(i["ok"] for i in idents.values()):
            log_with_timestamp("ABORT: a pool failed identity verification", "red")
            sys.exit(1)
        overlap = idents["pool-a"]["devices"] & idents["pool-b"]["devices"]
        if overlap:
            log_with_timestamp(
                f"WARNING: pools share devices {sorted(overlap)} — quota "
                f"domains are NOT isolated", "yellow")

        # --- fire the split workload --------------------------------
        t_start = time.time()
        log_with_timestamp(
            f"Sending {n} requests split by {args.strategy!r} across 2 pools",
            "blue")
        assignments = [pick_pool(args.strategy, i, n, rng) for i in range(n)]
        with cf.ThreadPoolExecutor(max_workers=min(n, 20)) as ex:
            results = list(ex.map(
                lambda i: send_pool_request(base_url, POOLS[assignments[i]],
                                            assignments[i], i),
                range(n)))

        # --- per-pool table (reference 404-446) ----------------------
        rows = []
        per_pool: dict[str, dict] = {}
        for pool in POOLS:
            rs = [r for r in results if r["pool"] == pool]
            ok = [r for r in rs if r["status"] == "success"]
            thr = [r for r in rs if r["status"] == "throttled"]
            err = [r for r in rs if r["status"] == "error"]
            avg = (sum(r["latency"] for r in ok) / len(ok)) if ok else None
            per_pool[pool] = {"total": len(rs), "success": len(ok),
                              "throttled": len(thr), "errors": len(err),
                              "avg_latency": avg}
            rows.append([pool, len(rs), len(ok), len(thr), len(err),
                         f"{avg:.2f}s" if avg else "-"])
        all_ok = sum(p["success"] for p in per_pool.values())
        avg_all = (sum(r["latency"] for r in results if r["status"] == "success")
                   / all_ok) if all_ok else None
        rows.append(["overall", n, all_ok,
                     sum(p["throttled"] for p in per_pool.values()),
                     sum(p["errors"] for p in per_pool.values()),
                     f"{avg_all:.2f}s" if avg_all else "-"])
        print_table(["Pool", "Sent", "Success", "Throttled", "Errors",
                     "Avg latency"], rows, title="Per-Pool Results")

        # --- per-pool device distribution, queried in parallel -------
        since = time.time() - t_start + 5
        with cf.ThreadPoolExecutor(2) as ex:
            dists = dict(zip(POOLS, ex.map(
                lambda alias: client.distribution(by="device", alias=alias,
                                                  since_s=since),
                POOLS.values())))
        complete = True
        for pool, dist in dists.items():
            print_table(
                ["Device", "Invocations", "Percentage"],
                [[dev, c, f"{dist['percentages'][dev]}%"]
                 for dev, c in dist["distribution"].items()],
                title=f"{pool} device distribution "
                      f"({dist['total']} invocations)")
            complete &= dist["total"] == per_pool[pool]["success"]
        if complete:
            log_with_timestamp("COMPLETE: ledger counts match per-pool "
                               "successes", "green")
        verdict = all_ok == n and not overlap
        log_with_timestamp(
            f"ACCOUNT SHARDING {'WORKING' if verdict else 'DEGRADED'}: "
            f"{all_ok}/{n} succeeded across 2 isolated pools",
            "green" if verdict else "yellow")
        return {"total": n, "success": all_ok, "per_pool": per_pool,
                "distributions": {p: d["distribution"] for p, d in dists.items()},
                "isolated": not overlap, "complete": complete}


def main() -> int:
    ap = argparse.ArgumentParser(description="GPU-pool (account) sharding demo")
    ap.add_argument("--requests", type=int, default=20)
    ap.add_argument("--strategy", choices=["round-robin", "split", "random"],
                    default="round-robin")
    ap.add_argument("--seed", type=int, default=None)
    add_common_args(ap)
    args = ap.parse_args()
    if not 1 <= args.requests <= 100:
        print("error: --requests must be between 1 and 100", file=sys.stderr)
        return 2
    stats = run_cross_pool_demo(args)
    return 0 if stats["success"] == stats["total"] else 1


if __name__ == "__main__":
    sys.exit(main())
