#!/usr/bin/env python3
"""RCCL rehearsal smoke: multi-rank RCCL collectives on
however many GPUs exist — on a 1-GPU lease, 2 ranks co-locate on cuda:0
(RCCL, unlike stock NCCL, accepts multiple ranks per device).

Self-spawning: `python scripts/rccl_smoke.py --world 2` forks the ranks,
each initializes a `nccl` (=RCCL) process group, runs all_reduce /
broadcast / all_gather on CUDA tensors at the sizes TP decode actually
uses (batch x hidden), checks values, prints per-rank OK.  Exit 0 iff
every rank passed.
"""

from __future__ import annotations

import argparse
import os
import socket
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def rank_main(rank: int, world: int, port: int) -> None:
    import datetime

    import torch
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dev = rank % max(1, torch.cuda.device_count())
    torch.cuda.set_device(dev)
    t0 = time.time()
    dist.init_process_group(backend="nccl", rank=rank, world_size=world,
                            timeout=datetime.timedelta(seconds=180))
    print(f"[rccl r{rank}] init ok on cuda:{dev} in {time.time()-t0:.1f}s",
          flush=True)

    # decode-shaped all_reduce: [64, 4096] bf16 (one per layer in TP)
    x = torch.full((64, 4096), float(rank + 1), dtype=torch.bfloat16,
                   device=f"cuda:{dev}")
    dist.all_reduce(x)
    expect = float(world * (world + 1) // 2)
    assert torch.all(x == expect), (x[0, 0].item(), expect)

    # broadcast (weight-load path)
    b = torch.arange(1024, dtype=torch.float32, device=f"cuda:{dev}")
    if rank != 0:
        b.zero_()
    dist.broadcast(b, src=0)
    assert torch.equal(b.cpu(), torch.arange(1024, dtype=torch.float32))

    # all_gather (sampling-state exchange shape)
    g = [torch.zeros(8, device=f"cuda:{dev}") for _ in range(world)]
    dist.all_gather(g, torch.full((8,), float(rank), device=f"cuda:{dev}"))
    for r in range(world):
        assert torch.all(g[r] == float(r))

    # 100 back-to-back small all_reduces: the TP decode cadence; time them
    y = torch.ones(64, 4096, dtype=torch.bfloat16, device=f"cuda:{dev}")
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(100):
        dist.all_reduce(y)
    torch.cuda.synchronize()
    us = (time.time() - t0) / 100 * 1e6
    print(f"[rccl r{rank}] all_reduce[64x4096 bf16] x100: {us:.1f} us avg",
          flush=True)

    dist.barrier()
    dist.destroy_process_group()
    print(f"[rccl r{rank}] OK", flush=True)


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--world", type=int, default=2)
    ap.add_argument("--rank", type=int, default=None, help="(internal)")
    ap.add_argument("--port", type=int, default=None, help="(internal)")
    args = ap.parse_args()

    if args.rank is not None:
        rank_main(args.rank, args.world, args.port)
        return

    port = free_port()
    procs = []
    for r in range(args.world):
        env = dict(os.environ)
        env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
        env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
        procs.append(subprocess.Popen(
            [sys.executable, os.path.abspath(__file__),
             "--world", str(args.world), "--rank", str(r),
             "--port", str(port)], env=env))
    rcs = []
    deadline = time.time() + 420
    for p in procs:
        rcs.append(p.wait(timeout=max(5, deadline - time.time())))
    if any(rc != 0 for rc in rcs):
        print(f"FAIL: rank exit codes {rcs}", flush=True)
        sys.exit(1)
    print(f"RCCL SMOKE PASS: {args.world} ranks", flush=True)


if __name__ == "__main__":
    main()
