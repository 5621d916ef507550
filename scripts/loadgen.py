#!/usr/bin/env python3
"""Closed-loop load generator subprocess for bench.py.

Runs in its OWN process so the gateway's event loop (rank 0) does not
share a GIL with the client side at 8-GPU request rates.  Protocol on
stdin/stdout lines:  "ROUND" -> run one round (every concurrency slot
completes exactly one request) and print one JSON result line;
"QUIT" -> exit.
"""

from __future__ import annotations

import argparse
import asyncio
import json
import random
import string
import sys
import time


def make_prompt(rng: random.Random, n_tokens: int) -> str:
    n = max(8, n_tokens - 18)
    return "".join(rng.choice(string.ascii_lowercase + " ") for _ in range(n))


async def one_request(session, url, prompt, max_tokens, results):
    t0 = time.monotonic()
    body = {"model": "bench-model",
            "messages": [{"role": "user", "content": prompt}],
            "max_tokens": max_tokens, "temperature": 0.0}
    try:
        async with session.post(url, json=body) as resp:
            data = await resp.json()
            ok = resp.status == 200
            usage = data.get("usage", {}) if ok else {}
            results.append({
                "ok": ok, "latency": time.monotonic() - t0,
                "completion_tokens": usage.get("completion_tokens", 0),
                "prompt_tokens": usage.get("prompt_tokens", 0)})
    except Exception as e:
        results.append({"ok": False, "latency": time.monotonic() - t0,
                        "error": str(e)[:200], "completion_tokens": 0,
                        "prompt_tokens": 0})


async def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int, required=True)
    ap.add_argument("--concurrency", type=int, required=True)
    ap.add_argument("--prompt-tokens", type=int, default=128)
    ap.add_argument("--output-tokens", type=int, default=64)
    ap.add_argument("--seed", type=int, default=1234)
    args = ap.parse_args()

    import aiohttp
    rng = random.Random(args.seed)
    url = f"http://127.0.0.1:{args.port}/chat/completions"
    conn = aiohttp.TCPConnector(limit=0)
    timeout = aiohttp.ClientTimeout(total=600)
    loop = asyncio.get_running_loop()
    reader = asyncio.StreamReader()
    await loop.connect_read_pipe(
        lambda: asyncio.StreamReaderProtocol(reader), sys.stdin)

    async with aiohttp.ClientSession(connector=conn, timeout=timeout) as s:
        print("READY", flush=True)
        while True:
            line = (await reader.readline()).decode().strip()
            if not line or line == "QUIT":
                return
            assert line == "ROUND", line
            prompts = [make_prompt(rng, args.prompt_tokens)
                       for _ in range(args.concurrency)]
            results: list = []
            t0 = time.monotonic()
            await asyncio.gather(*[
                one_request(s, url, p, args.output_tokens, results)
                for p in prompts])
            out = {
                "wall": time.monotonic() - t0,
                "ok": sum(1 for r in results if r["ok"]),
                "total": len(results),
                "latencies": [round(r["latency"], 4) for r in results if r["ok"]],
                "completion_tokens": sum(r["completion_tokens"] for r in results),
                "prompt_tokens": sum(r["prompt_tokens"] for r in results),
                "errors": [r.get("error") for r in results if not r["ok"]][:3],
            }
            print(json.dumps(out), flush=True)


if __name__ == "__main__":
    asyncio.run(main())
