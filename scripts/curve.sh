#!/bin/bash
# throughput/latency curve: bench.py across concurrency levels
for C in 16 32 64 128 256; do
  timeout 420 python bench.py --steps 3 --warmup 1 --concurrency "$C" 2>/dev/null | tail -1 | python3 -c '
import json, sys
d = json.loads(sys.stdin.read())
c = d["config"]
print("%4d %7.1f req/s p50=%.2f p99=%.2f tok/s=%.0f" % (
    c["concurrency_per_gpu"], d["value"], c["p50_latency_s"],
    c["p99_latency_s"], c["total_tokens_per_s"]))'
done
