"""Gateway assembly + ``python -m resilient_llm_amd.gateway.server`` CLI.

The native equivalent of the reference launcher (reference
bin/start-gateway.sh: parse/validate config, then exec the proxy).  Builds
the worker set the config names — CPU stubs in-process, one engine worker
per GPU, TP pools — then serves the OpenAI-compatible API.
"""

from __future__ import annotations

import argparse
import asyncio
import os
import signal
from typing import Optional

from ..config import Config, load_config
from ..utils.logging import log_with_timestamp
from ..workers.base import WorkerRegistry
from ..workers.stub import StubWorker
from .app import GatewayApp
from .http import HttpServer


def build_registry(config: Config, stub_kwargs: Optional[dict] = None) -> WorkerRegistry:
    """Instantiate one worker per distinct backend target in the config.

    ``stub/...`` targets become in-process :class:`StubWorker`s.
    ``gpu/...`` and ``pool/...`` targets are spawned as engine worker
    processes by :mod:`resilient_llm_amd.workers.gpu` (needs a GPU) — the
    caller registers those; this helper only auto-builds stubs so the
    plumbing path works anywhere.
    """
    registry = WorkerRegistry()
    stub_models: dict[str, set[str]] = {}
    for d in config.deployments:
        if d.backend_kind != "stub":
            continue
        targets = [d.backend_target]
        if d.backend_target == "*":
            # a spread alias needs every stub device that exists elsewhere
            targets = sorted({e.backend_target for e in config.deployments
                              if e.backend_kind == "stub" and e.backend_target != "*"}) or ["0"]
        for t in targets:
            stub_models.setdefault(t, set()).add(d.backend_model or d.model_id)
    for target, models in stub_models.items():
        registry.register("stub", target, StubWorker(target, models,
                                                     **(stub_kwargs or {})))
    return registry


async def serve(config: Config, registry: Optional[WorkerRegistry] = None,
                ready_event: Optional[asyncio.Event] = None,
                reuse_port: bool = False, drain_s: float = 30.0,
                shared_limits_path: Optional[str] = None) -> None:
    registry = registry or build_registry(config)
    app = GatewayApp(config, registry, shared_limits_path=shared_limits_path)
    server = HttpServer(app.handle, host=config.cluster.host, port=config.cluster.port)
    await server.start(reuse_port=reuse_port)
    for w in registry.all().values():
        if hasattr(w, "connect"):
            await w.connect(timeout=900)
    await app.start_background()
    log_with_timestamp(
        f"gateway listening on http://{config.cluster.host}:{config.cluster.port} "
        f"({len(config.deployments)} deployments, "
        f"{len(registry.all())} workers)", "green")
    if ready_event is not None:
        ready_event.set()
    stop = asyncio.Event()
    loop = asyncio.get_running_loop()
    for sig in (signal.SIGINT, signal.SIGTERM):
        try:
            loop.add_signal_handler(sig, stop.set)
        except NotImplementedError:
            pass
    try:
        await stop.wait()
        # graceful drain: stop accepting, let in-flight requests finish
        server.close_listener()
        deadline = loop.time() + drain_s
        while loop.time() < deadline:
            busy = sum(s.in_flight for s in app.router.states)
            if busy == 0:
                break
            await asyncio.sleep(0.1)
        else:
            log_with_timestamp(
                f"drain timeout after {drain_s}s; "
                f"{sum(s.in_flight for s in app.router.states)} still "
                f"in flight", "yellow")
    finally:
        await app.stop_background()
        await server.stop()
        await registry.close()


def main() -> None:
    ap = argparse.ArgumentParser(description="resilient_llm_amd gateway")
    ap.add_argument("--config", default="config/config.yaml")
    ap.add_argument("--port", type=int, default=None, help="override cluster.port")
    ap.add_argument("--host", default=None)
    ap.add_argument("--drain-s", type=float, default=30.0,
                    help="graceful-shutdown drain window for in-flight "
                         "requests on SIGTERM/SIGINT")
    ap.add_argument("--workers", type=int, default=1,
                    help="gateway processes sharing the port via "
                         "SO_REUSEPORT; deployment RPM/TPM windows are "
                         "SHARED across processes (mmap'd counter file), "
                         "so rpm=3 admits exactly 3 cluster-wide. "
                         "Consumer-key limits and the in-memory ledger "
                         "view stay per process (JSONL ledger is shared).")
    args = ap.parse_args()
    config = load_config(args.config)
    if args.port is not None:
        config.cluster.port = args.port
    if args.host is not None:
        config.cluster.host = args.host

    kinds = {d.backend_kind for d in config.deployments}
    registry = build_registry(config)
    if kinds & {"gpu", "pool"}:
        from ..workers.gpu import register_gpu_workers
        register_gpu_workers(config, registry)
    if args.workers > 1:
        import multiprocessing
        import tempfile
        limited = any(d.rpm or d.tpm for d in config.deployments)
        shared_path = None
        if limited:
            shared_path = os.path.join(
                tempfile.mkdtemp(prefix="rlli-gw-"), "windows.bin")
        log_with_timestamp(
            f"SO_REUSEPORT scale-out: {args.workers} gateway processes; "
            + ("RPM/TPM windows SHARED via " + shared_path if limited
               else "no rate limits configured"), "green")
        procs = []
        for _ in range(args.workers - 1):
            pr = multiprocessing.Process(
                target=lambda: asyncio.run(serve(
                    config, registry, reuse_port=True, drain_s=args.drain_s,
                    shared_limits_path=shared_path)))
            # die with the parent: SIGTERM is delivered to the parent
            # only, and an orphaned reuse-port sibling would keep the
            # port serving forever
            pr.daemon = True
            pr.start()
            procs.append(pr)
        try:
            asyncio.run(serve(config, registry, reuse_port=True,
                              drain_s=args.drain_s,
                              shared_limits_path=shared_path))
        finally:
            for pr in procs:
                pr.terminate()
            for pr in procs:
                pr.join(timeout=10)
    else:
        asyncio.run(serve(config, registry, drain_s=args.drain_s))


if __name__ == "__main__":
    main()
