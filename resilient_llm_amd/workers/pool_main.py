"""Pool worker-process entry point (one rank of a TP pool)."""

from __future__ import annotations

import argparse
import asyncio
import os
import sys


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--rank", type=int, required=True)
    ap.add_argument("--world", type=int, required=True)
    ap.add_argument("--master-port", type=int, required=True)
    ap.add_argument("--pool-name", required=True)
    ap.add_argument("--model", default="llama-3-70b")
    ap.add_argument("--socket", required=True)
    ap.add_argument("--kv-gb", type=float, default=24.0)
    ap.add_argument("--max-batch", type=int, default=64)
    ap.add_argument("--num-blocks", type=int, default=None)
    ap.add_argument("--tp-backend", default="nccl")
    ap.add_argument("--weights", default=None,
                    help="safetensors dir — each rank loads its TP shard")
    ap.add_argument("--kv-dtype", default="bf16", choices=("bf16", "fp8"))
    ap.add_argument("--device", default=None, help="override (tests: cpu)")
    args = ap.parse_args()

    import torch
    if args.device:
        device = args.device
    else:
        assert torch.cuda.is_available()
        device = "cuda:0"       # pinned via HIP_VISIBLE_DEVICES

    if device.startswith("cuda"):
        from .. import ops
        ops.load_extension(required=True)

    from ..parallel import init_pool_groups
    from ..utils.logging import log_with_timestamp
    from ..workers.pool import TPControl, follower_loop
    from .engine_worker import EngineWorker
    from .rpc import WorkerRpcServer

    control_group, tp_group = init_pool_groups(
        args.rank, args.world, args.master_port, tp_backend=args.tp_backend,
        device=device if device.startswith("cuda") else None)
    control = TPControl(control_group, is_leader=args.rank == 0)

    worker = EngineWorker(
        device=device, model_name=args.model,
        device_label=f"pool:{args.pool_name}",
        kv_gb=args.kv_gb, max_batch_size=args.max_batch,
        kv_dtype=args.kv_dtype,
        num_blocks=args.num_blocks, weights=args.weights,
        tp_rank=args.rank, tp_world=args.world, tp_group=tp_group,
        tp_control=control)
    log_with_timestamp(
        f"pool {args.pool_name} rank {args.rank}/{args.world} ready: "
        f"{args.model} on {device}", "green")

    if args.rank == 0:
        async def run() -> None:
            server = WorkerRpcServer(worker, args.socket,
                                     on_kill=lambda: os._exit(7))
            await server.serve_forever()
        try:
            asyncio.run(run())
        except KeyboardInterrupt:
            sys.exit(0)
    else:
        follower_loop(worker.engine, control)


if __name__ == "__main__":
    main()
