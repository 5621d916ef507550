"""Acceptance tests: the five demo entry points run end-to-end on the
stub backend and reach the reference's verdicts (keep the
demos as self-verifying acceptance tests)."""

import os
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
DEMOS = os.path.join(REPO, "demos")


def run_demo(script, *args, timeout=180):
    # rate windows are fixed per wall-clock minute: a request burst that
    # straddles a minute boundary sees the primary's rpm budget TWICE and
    # the exact-count verdicts ("3 primary + 7 fallback") go flaky — hold
    # the launch for a few seconds when too close to the boundary
    into_minute = time.time() % 60.0
    if into_minute > 50.0:
        time.sleep(60.5 - into_minute)
    proc = subprocess.run(
        [sys.executable, os.path.join(DEMOS, script), *args],
        capture_output=True, text=True, timeout=timeout,
        env={**os.environ, "NO_COLOR": "1"})
    return proc


def test_demo_cris():
    p = run_demo("demo_cris.py", "--requests", "16")
    assert p.returncode == 0, p.stdout + p.stderr
    assert "CROSS-GPU INFERENCE WORKING" in p.stdout
    assert "COMPLETE: ledger count matches 16 successful requests" in p.stdout


def test_demo_cris_rejects_bad_request_count():
    p = run_demo("demo_cris.py", "--requests", "500")
    assert p.returncode == 2


def test_demo_fallback():
    p = run_demo("demo_fallback.py")
    assert p.returncode == 0, p.stdout + p.stderr
    assert "FALLBACK WORKING: 7 requests successfully failed over" in p.stdout
    assert "10/10 succeeded (3 primary + 7 fallback)" in p.stdout


def test_demo_fallback_with_fault_injection():
    p = run_demo("demo_fallback.py", "--fault")
    assert p.returncode == 0, p.stdout + p.stderr
    assert "INJECTING FAULT" in p.stdout
    assert "10/10 succeeded" in p.stdout


def test_demo_load_balancing():
    p = run_demo("demo_load_balancing.py")
    assert p.returncode == 0, p.stdout + p.stderr
    assert "LOAD BALANCING WORKING" in p.stdout
    assert "10/10 succeeded" in p.stdout


def test_demo_quota_isolation():
    p = run_demo("demo_quota_isolation.py")
    assert p.returncode == 0, p.stdout + p.stderr
    assert "QUOTA ISOLATION EFFECTIVE" in p.stdout
    # exact reference semantics: A 3/5 = 60%, B and C 5/5 (README.md:255-266)
    assert "3/5" in p.stdout
    assert "60%" in p.stdout


def test_demo_account_sharding_round_robin():
    p = run_demo("demo_account_sharding.py", "--requests", "20")
    assert p.returncode == 0, p.stdout + p.stderr
    assert "ACCOUNT SHARDING WORKING: 20/20 succeeded" in p.stdout
    assert "COMPLETE: ledger counts match per-pool successes" in p.stdout


def test_demo_account_sharding_split_strategy():
    p = run_demo("demo_account_sharding.py", "--requests", "10",
                 "--strategy", "split")
    assert p.returncode == 0, p.stdout + p.stderr
    assert "10/10 succeeded" in p.stdout
