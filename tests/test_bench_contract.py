"""bench.py driver contract: run the flagship bench on CPU (tiny model,
1 step, low concurrency) and validate the single JSON line it prints —
every field the driver/judge parses must be present and well-typed.
Catches regressions to the contract before they reach a GPU run."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

pytestmark = pytest.mark.timeout(300)


def test_bench_json_contract():
    proc = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"),
         "--steps", "1", "--warmup", "0", "--model", "tiny",
         "--concurrency", "4", "--prompt-tokens", "16",
         "--output-tokens", "8", "--device", "cpu"],
        capture_output=True, text=True, timeout=280, cwd=REPO)
    assert proc.returncode == 0, proc.stderr[-2000:]
    lines = [ln for ln in proc.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, proc.stdout
    out = json.loads(lines[0])

    # metric string is BASELINE.json's, verbatim
    assert out["metric"] == ("sustained reqs/sec + success-rate + "
                             "p50 latency @ 64 concurrent, Llama-3-8B")
    assert isinstance(out["value"], (int, float)) and out["value"] > 0
    assert out["unit"] == "reqs/s"
    assert out["n_gpus"] == 1
    assert out["steps"] == 1
    assert out["warmup"] == 0
    assert isinstance(out["ms_per_step"], (int, float)) and out["ms_per_step"] > 0
    assert out["higher_is_better"] is True
    assert out["scaling"] == "weak"
    assert out["dtype"] == "bf16"
    assert out["data"] == "synthetic"
    # vs_baseline: number or null (BASELINE.md has no native-hw number)
    assert out["vs_baseline"] is None or isinstance(out["vs_baseline"], (int, float))
    cfg = out["config"]
    assert cfg["model"] == "tiny"
    assert cfg["concurrency_per_gpu"] == 4
    assert cfg["global_batch"] == 4
    assert cfg["parallelism"] == "dp1"
    assert cfg["success_rate"] == 1.0
    assert cfg["p50_latency_s"] > 0
    assert cfg["p99_latency_s"] >= cfg["p50_latency_s"]
    assert cfg["output_tokens_per_s"] > 0
    assert cfg["total_tokens_per_s"] > cfg["output_tokens_per_s"]
