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
                     tp_backend: str = "nccl", device=None):
    """Returns (control_group, tp_group): gloo for request-stream
    broadcasts (CPU-side, never touches the GPU), nccl/RCCL for tensor
    collectives.  Single init_process_group + one new_group so both
    share a rendezvous.

    When ``device`` is given and the backend is nccl, the TP group is
    VERIFIED with one eager all_reduce before serving: RCCL rejects some
    topologies outright (e.g. two ranks on one device — 'Duplicate GPU
    detected', measured r02) and the failure would otherwise surface as
    a mid-request crash on the first real collective.  All ranks vote on
    the outcome over the gloo control group and fall back to gloo TP
    together, so a topology surprise degrades throughput instead of
    taking the pool down (the first multi-GPU run is
    blind)."""
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ["MASTER_PORT"] = str(master_port)
    # collective timeout: must exceed the leader's idle-heartbeat period
    # (EngineWorker.TP_HEARTBEAT_S) or idle followers die in broadcast
    timeout_s = float(os.environ.get("RLLI_TP_TIMEOUT_S", "600"))
    dist.init_process_group(backend="gloo", rank=rank, world_size=world,
                            timeout=datetime.timedelta(seconds=timeout_s))
    control_group = dist.group.WORLD
    if tp_backend == "gloo" or world == 1:
        tp_group = dist.new_group(backend=tp_backend)
        return control_group, tp_group
    tp_group = dist.new_group(backend="nccl")
    if device is None:
        return control_group, tp_group
    import torch
    ok = 1
    try:
        probe = torch.ones(1, device=device)
        dist.all_reduce(probe, group=tp_group)
        if int(probe.item()) != world:
            ok = 0
    except Exception as e:
        from ..utils.logging import log_with_timestamp
        log_with_timestamp(f"rank {rank}: RCCL TP probe failed ({e}); "
                           f"voting for gloo fallback", "yellow")
        ok = 0
    # unanimous vote on the CPU control group (symmetric information:
    # every rank must pick the same backend or collectives deadlock)
    vote = torch.tensor([ok], dtype=torch.int32)
    dist.all_reduce(vote, op=dist.ReduceOp.MIN, group=control_group)
    if int(vote.item()) == 1:
        return control_group, tp_group
    from ..utils.logging import log_with_timestamp
    log_with_timestamp(f"rank {rank}: TP collectives fall back to gloo "
                       f"(RCCL cannot serve this topology)", "yellow")
    tp_group_gloo = dist.new_group(backend="gloo")
    return control_group, tp_group_gloo
