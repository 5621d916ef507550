"""Tensor-parallel pools over RCCL/xGMI.

A pool ("account" in the reference's analogy, X11) is tensor_parallel
GPUs serving one model: column-parallel QKV/gate-up, row-parallel
o-proj/down-proj, two all-reduces per layer over xGMI —
`torch.distributed` with backend "nccl" IS RCCL on ROCm.

Engine lockstep: every rank runs an IDENTICAL engine (same scheduler,
same seed, Gumbel-hash sampler -> same tokens from the replicated
logits), so the only cross-rank coordination is the REQUEST STREAM: the
leader broadcasts (add/abort) ops on a gloo control group once per
engine iteration; collectives then line up by construction.  No KV
migration, no tensor metadata exchange.
"""

from __future__ import annotations

import datetime
import os

import torch.distributed as dist


def init_pool_groups(rank: int, world: int, master_port: int,
                     tp_backend: str = "nccl"):
    """Returns (control_group, tp_group): gloo for request-stream
    broadcasts (CPU-side, never touches the GPU), nccl/RCCL for tensor
    collectives.  Single init_process_group + one new_group so both
    share a rendezvous."""
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ["MASTER_PORT"] = str(master_port)
    # collective timeout: must exceed the leader's idle-heartbeat period
    # (EngineWorker.TP_HEARTBEAT_S) or idle followers die in broadcast
    timeout_s = float(os.environ.get("RLLI_TP_TIMEOUT_S", "600"))
    dist.init_process_group(backend="gloo", rank=rank, world_size=world,
                            timeout=datetime.timedelta(seconds=timeout_s))
    control_group = dist.group.WORLD
    if tp_backend == "gloo":
        tp_group = dist.new_group(backend="gloo")
    else:
        tp_group = dist.new_group(backend="nccl")
    return control_group, tp_group
