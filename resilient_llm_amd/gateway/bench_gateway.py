"""Standalone gateway process for high-throughput deployments.

Several of these share ONE port via SO_REUSEPORT (the kernel spreads
connections), each holding its own connections to every worker — the
same scale-out model as the reference proxy's ``--num_workers`` (its
launcher pins 1 for exactly this reason, reference bin/start-gateway.sh:56):
RPM/TPM windows and the ledger are PER GATEWAY PROCESS, so run >1 only
for unlimited/benchmark configs or shard limits accordingly.

Usage:
  python -m resilient_llm_amd.gateway.bench_gateway --port P \
      --model llama-3-8b --sockets /tmp/w0.sock,/tmp/w1.sock [--ready-file F]
"""

from __future__ import annotations

import argparse
import asyncio


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int, required=True)
    ap.add_argument("--model", required=True)
    ap.add_argument("--sockets", required=True,
                    help="comma-separated worker unix sockets")
    ap.add_argument("--alias", default="bench-model")
    ap.add_argument("--ready-file", default=None)
    args = ap.parse_args()

    from ..config import load_config
    from .app import GatewayApp
    from .http import HttpServer
    from ..workers.base import WorkerRegistry
    from ..workers.rpc import RpcWorkerClient

    config = load_config(data={
        "cluster": {"port": args.port},
        "model_list": [
            {"model_name": args.alias,
             "litellm_params": {"model": f"gpu/*/{args.model}"},
             "model_info": {"id": f"{args.alias}-spread"}}],
        "router_settings": {"routing_strategy": "simple-shuffle",
                            "enable_pre_call_checks": False},
    })

    async def run() -> None:
        registry = WorkerRegistry()
        for i, sock in enumerate(args.sockets.split(",")):
            client = RpcWorkerClient(f"gpu:{i}", {args.model}, sock)
            await client.connect(timeout=900)
            registry.register("gpu", str(i), client)
        app = GatewayApp(config, registry, health_interval_s=5.0)
        server = HttpServer(app.handle, host="127.0.0.1", port=args.port)
        await server.start(reuse_port=True)
        await app.start_background()
        if args.ready_file:
            with open(args.ready_file, "w") as f:
                f.write("ready")
        await asyncio.Event().wait()

    try:
        asyncio.run(run())
    except KeyboardInterrupt:
        pass


if __name__ == "__main__":
    main()
