"""Shared plumbing for the five demo entry points.

The reference duplicates its logger/sanitizer/config-table helpers in
every script; here demos share this module.  Each demo can
either talk to an already-running gateway (``--base-url``, like the
reference demos talk to the LiteLLM proxy) or self-host one in-process
(``--self-host``) so the demo is runnable on any box — with stub workers
when no GPU is present, with real engine workers when there is one.
"""

from __future__ import annotations

import argparse
import asyncio
import contextlib
import os
import socket
import sys
import threading
from typing import Optional

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)

from resilient_llm_amd.client import OpenAIClient                     # noqa: E402
from resilient_llm_amd.config import Config, load_config              # noqa: E402
from resilient_llm_amd.utils.logging import (                         # noqa: E402
    colorize, log_with_timestamp,
)

DEFAULT_STUB_CONFIG = os.path.join(REPO_ROOT, "config", "config.stub.yaml")
DEFAULT_GPU_CONFIG = os.path.join(REPO_ROOT, "config", "config.yaml")


def add_common_args(ap: argparse.ArgumentParser) -> None:
    ap.add_argument("--base-url", default=None,
                    help="URL of a running gateway (default: self-host one)")
    ap.add_argument("--config", default=None,
                    help="config.yaml (default: stub config, or GPU config "
                         "when --gpu)")
    ap.add_argument("--gpu", action="store_true",
                    help="self-host with real GPU engine workers")


def free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@contextlib.contextmanager
def gateway_session(args, stub_kwargs: Optional[dict] = None):
    """Yield (client, config).  Self-hosts the gateway unless --base-url."""
    if args.base_url:
        cfg_path = args.config or (DEFAULT_GPU_CONFIG if args.gpu else DEFAULT_STUB_CONFIG)
        config = load_config(cfg_path)
        yield OpenAIClient(args.base_url), config
        return

    cfg_path = args.config or (DEFAULT_GPU_CONFIG if args.gpu else DEFAULT_STUB_CONFIG)
    config = load_config(cfg_path)
    config.cluster.port = free_port()

    from resilient_llm_amd.gateway.app import GatewayApp
    from resilient_llm_amd.gateway.http import HttpServer
    from resilient_llm_amd.gateway.server import build_registry

    registry = build_registry(config, stub_kwargs=stub_kwargs)
    if args.gpu:
        from resilient_llm_amd.workers.gpu import register_gpu_workers
        register_gpu_workers(config, registry)

    loop = asyncio.new_event_loop()
    ready = threading.Event()
    holder: dict = {}

    async def _main():
        holder["stop"] = asyncio.Event()
        app = GatewayApp(config, registry, health_interval_s=0.5)
        server = HttpServer(app.handle, host="127.0.0.1", port=config.cluster.port)
        await server.start()
        # wait for process-backed workers to come up before serving
        for w in registry.all().values():
            if hasattr(w, "connect"):
                await w.connect(timeout=900)
        await app.start_background()
        ready.set()
        await holder["stop"].wait()
        await app.stop_background()
        await server.stop()
        await registry.close()
        # let in-flight connection handlers unwind before the loop dies
        tasks = [t for t in asyncio.all_tasks() if t is not asyncio.current_task()]
        for t in tasks:
            t.cancel()
        await asyncio.gather(*tasks, return_exceptions=True)

    th = threading.Thread(target=lambda: loop.run_until_complete(_main()), daemon=True)
    th.start()
    if not ready.wait(timeout=900):
        raise RuntimeError("self-hosted gateway failed to start")
    log_with_timestamp(
        f"self-hosted gateway on http://127.0.0.1:{config.cluster.port}", "grey")
    try:
        yield OpenAIClient(f"http://127.0.0.1:{config.cluster.port}"), config
    finally:
        loop.call_soon_threadsafe(holder["stop"].set)
        th.join(timeout=15)
        loop.close()


# -------------------------------------------------------------- rendering
def print_table(headers: list[str], rows: list[list], title: str = "",
                colors: Optional[list] = None) -> None:
    widths = [len(h) for h in headers]
    str_rows = [[str(c) for c in row] for row in rows]
    for row in str_rows:
        for i, c in enumerate(row):
            widths[i] = max(widths[i], len(c))
    sep = "+" + "+".join("-" * (w + 2) for w in widths) + "+"
    if title:
        print(f"\n{title}")
    print(sep)
    print("|" + "|".join(f" {h.ljust(w)} " for h, w in zip(headers, widths)) + "|")
    print(sep)
    for j, row in enumerate(str_rows):
        line = "|" + "|".join(f" {c.ljust(w)} " for c, w in zip(row, widths)) + "|"
        if colors and colors[j]:
            line = colorize(line, colors[j])
        print(line)
    print(sep)


def print_router_settings(client: OpenAIClient) -> None:
    """Render the gateway's live router config — the analogue of the
    reference's print_router_settings (src/demo_fallback.py:45-105)."""
    state = client.router_state()
    s = state["settings"]
    print_table(
        ["Setting", "Value"],
        [["routing_strategy", s["routing_strategy"]],
         ["enable_pre_call_checks", s["enable_pre_call_checks"]],
         ["allowed_fails", s["allowed_fails"]],
         ["cooldown_time", f"{s['cooldown_time']}s"],
         ["fallbacks", s["fallbacks"]]],
        title="Router Settings")
    rows = [[d["model_name"], d["model_id"], d["backend"],
             d["rpm"] or "-", d["tpm"] or "-",
             "yes" if d["healthy"] else "NO"]
            for d in state["deployments"]]
    print_table(["Alias", "Deployment ID", "Backend", "RPM", "TPM", "Healthy"],
                rows, title="Deployments")
