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
    # the forked reuse-port sibling must die with the parent (it leaked
    # an orphan per run before the daemon+terminate fix): the port must
    # stop answering entirely
    time.sleep(1.0)
    import socket as _s
    with _s.socket() as s:
        assert s.connect_ex(("127.0.0.1", port)) != 0, \
            "orphaned reuse-port worker still serving"


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


def test_graceful_drain_on_sigterm():
    """SIGTERM mid-request: the gateway stops accepting, the in-flight
    request COMPLETES, and the process exits 0 — no dropped work."""
    import json as _json
    import signal as _signal
    import threading
    import urllib.request

    port = free_port()
    # slow stub (~1.2 s per request) so SIGTERM lands mid-request
    cfg = os.path.join(REPO, "config", "config.stub.yaml")
    proc = subprocess.Popen(
        [sys.executable, "-c",
         "import sys; sys.argv=['x','--config',%r,'--port','%d']; "
         "from resilient_llm_amd.gateway import server as S; "
         "from resilient_llm_amd.config import load_config; "
         "import asyncio; "
         "cfg=load_config(%r); cfg.cluster.port=%d; "
         "reg=S.build_registry(cfg, stub_kwargs={'first_token_ms':1200}); "
         "asyncio.run(S.serve(cfg, reg))" % (cfg, port, cfg, port)],
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
                time.sleep(0.2)
        else:
            raise AssertionError("gateway never healthy")

        result = {}

        def slow_request():
            body = _json.dumps({
                "model": "llama-fallback-loadbalance",
                "messages": [{"role": "user", "content": "slow"}],
                "max_tokens": 3}).encode()
            req = urllib.request.Request(
                f"http://127.0.0.1:{port}/chat/completions", data=body,
                headers={"content-type": "application/json"})
            with urllib.request.urlopen(req, timeout=30) as r:
                result["data"] = _json.loads(r.read())

        th = threading.Thread(target=slow_request)
        th.start()
        time.sleep(0.4)                  # request is in flight
        proc.send_signal(_signal.SIGTERM)
        th.join(timeout=30)
        assert result["data"]["usage"]["completion_tokens"] == 3
        assert proc.wait(timeout=20) == 0
    finally:
        if proc.poll() is None:
            proc.kill()
