"""Driver-shape regression: bench.py under torch.distributed.run with 2
ranks on CPU (gloo, tiny model) — the exact invocation shape the driver
uses for the multi-GPU scale run."""

import json
import os
import subprocess
import sys

import pytest

from tests.gateway_harness import free_port

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

pytestmark = pytest.mark.timeout(420)


def test_bench_two_ranks_cpu():
    port = free_port()
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port),
         os.path.join(REPO, "bench.py"), "--gpus", "2",
         "--steps", "1", "--warmup", "0", "--model", "tiny",
         "--concurrency", "4", "--prompt-tokens", "16",
         "--output-tokens", "8", "--device", "cpu"],
        capture_output=True, text=True, timeout=400, cwd=REPO,
        env={**os.environ, "GLOO_SOCKET_IFNAME": "lo"})
    assert proc.returncode == 0, proc.stderr[-2000:]
    lines = [ln for ln in proc.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, proc.stdout
    out = json.loads(lines[0])
    assert out["n_gpus"] == 2
    assert out["config"]["parallelism"] == "dp2"
    assert out["config"]["global_batch"] == 8        # 4 per rank, weak
    assert out["config"]["success_rate"] == 1.0
    assert out["value"] > 0
