"""The real gateway server process (what bin/start-gateway.sh execs):
spawn `python -m resilient_llm_amd.gateway.server` on the stub config
and serve one completion through it."""

import os
import subprocess
import sys
import time

from resilient_llm_amd.client import OpenAIClient
from tests.gateway_harness import free_port

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_server_cli_serves_stub_config():
    port = free_port()
    proc = subprocess.Popen(
        [sys.executable, "-m", "resilient_llm_amd.gateway.server",
         "--config", os.path.join(REPO, "config", "config.stub.yaml"),
         "--port", str(port)],
        env={**os.environ, "PYTHONPATH": REPO},
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    try:
        client = OpenAIClient(f"http://127.0.0.1:{port}")
        deadline = time.time() + 30
        last = None
        while time.time() < deadline:
            try:
                h = client.health()
                if h.get("status") == "ok":
                    break
            except Exception as e:
                last = e
                time.sleep(0.3)
        else:
            raise AssertionError(f"server never became healthy: {last}")
        r = client.chat.completions.create(
            model="llama-fallback-demo",
            messages=[{"role": "user", "content": "hi"}], max_tokens=4)
        assert r.usage.completion_tokens == 4
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=10)
        except Exception:
            proc.kill()


def test_server_cli_multiple_workers_reuse_port():
    """--workers N forks N SO_REUSEPORT server processes on one port
    (the LiteLLM --num_workers analogue); all requests succeed."""
    port = free_port()
    proc = subprocess.Popen(
        [sys.executable, "-m", "resilient_llm_amd.gateway.server",
         "--config", os.path.join(REPO, "config", "config.stub.yaml"),
         "--port", str(port), "--workers", "2"],
        env={**os.environ, "PYTHONPATH": REPO},
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    try:
        client = OpenAIClient(f"http://127.0.0.1:{port}")
        deadline = time.time() + 30
        while time.time() < deadline:
            try:
                if client.health().get("status") == "ok":
                    break
            except Exception:
                time.sleep(0.3)
        else:
            raise AssertionError("server never became healthy")
        for i in range(6):
            r = OpenAIClient(f"http://127.0.0.1:{port}").chat.completions.create(
                model="llama-fallback-loadbalance",
                messages=[{"role": "user", "content": f"req {i}"}],
                max_tokens=3)
            assert r.usage.completion_tokens == 3
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=10)
        except Exception:
            proc.kill()


def test_start_gateway_script():
    """bin/start-gateway.sh (C7): validate config then exec the server."""
    port = free_port()
    proc = subprocess.Popen(
        ["bash", os.path.join(REPO, "bin", "start-gateway.sh"),
         os.path.join(REPO, "config", "config.stub.yaml"), str(port)],
        env={**os.environ, "PYTHONPATH": REPO},
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL,
        start_new_session=True)
    try:
        client = OpenAIClient(f"http://127.0.0.1:{port}")
        deadline = time.time() + 30
        while time.time() < deadline:
            try:
                if client.health().get("status") == "ok":
                    break
            except Exception:
                time.sleep(0.3)
        else:
            raise AssertionError("script-launched gateway never healthy")
        r = client.chat.completions.create(
            model="llama-fallback-demo",
            messages=[{"role": "user", "content": "hi"}], max_tokens=2)
        assert r.usage.completion_tokens == 2
    finally:
        import signal
        os.killpg(proc.pid, signal.SIGTERM)
        try:
            proc.wait(timeout=10)
        except Exception:
            os.killpg(proc.pid, signal.SIGKILL)
