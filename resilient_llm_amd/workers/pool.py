"""TP pool workers: process group per pool, leader serves the socket.

The leader (pool rank 0) is a full EngineWorker behind the RPC socket;
followers run the same engine in lockstep, fed by the leader's
request-stream broadcasts (see resilient_llm_amd.parallel).  Fault
injection 'kill' hard-exits the LEADER process; the gateway sees the
pool die and fails over, mirroring a whole-account outage in the
reference's model (X11).
"""

from __future__ import annotations

import os
import socket
import subprocess
import sys
import threading
from typing import Optional

import torch.distributed as dist

from ..config import PoolDef

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


class TPControl:
    """Request-stream lockstep: leader broadcasts a list of ops per
    engine iteration; every rank applies them identically."""

    def __init__(self, group, is_leader: bool) -> None:
        self.group = group
        self.is_leader = is_leader
        self._ops: list = []
        self._lock = threading.Lock()

    def submit(self, op) -> None:
        assert self.is_leader
        with self._lock:
            self._ops.append(op)

    def pending(self) -> bool:
        with self._lock:
            return bool(self._ops)

    def sync(self, engine) -> None:
        """Called once per engine iteration on EVERY rank."""
        from ..engine.engine import SamplingParams
        if self.is_leader:
            with self._lock:
                ops, self._ops = self._ops, []
            obj = [ops]
        else:
            obj = [None]
        dist.broadcast_object_list(obj, src=0, group=self.group)
        for op in obj[0]:
            kind = op[0]
            try:
                if kind == "add":
                    _, rid, prompt_ids, params = op
                    engine.add_request(rid, prompt_ids, SamplingParams(**params))
                elif kind == "abort":
                    engine.abort(op[1])
            except Exception:
                # deterministic across ranks (same engine state) — the op
                # is dropped identically everywhere
                pass


def follower_loop(engine, control: TPControl) -> None:
    """Ranks 1..N-1: mirror the leader's iterations forever."""
    while True:
        control.sync(engine)
        if engine.has_work():
            engine.step()


# ---------------------------------------------------------------- spawn
def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def spawn_pool_worker(pool: PoolDef, model_name: str, socket_path: str,
                      kv_gb: float = 24.0, max_batch: int = 64,
                      kv_dtype: Optional[str] = None,
                      device_override: Optional[str] = None,
                      weights: Optional[str] = None,
                      tp_backend: str = "nccl") -> list:
    """Spawn pool.tensor_parallel processes; returns the Popen list
    (leader first)."""
    world = pool.tensor_parallel
    port = _free_port()
    procs = []
    for r in range(world):
        env = dict(os.environ)
        if device_override is None:
            env["HIP_VISIBLE_DEVICES"] = str(pool.gpus[r])
        env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
        env["PYTHONPATH"] = REPO_ROOT + os.pathsep + env.get("PYTHONPATH", "")
        env["MASTER_ADDR"] = "127.0.0.1"
        cmd = [sys.executable, "-m", "resilient_llm_amd.workers.pool_main",
               "--rank", str(r), "--world", str(world),
               "--master-port", str(port),
               "--pool-name", pool.name,
               "--model", model_name,
               "--socket", socket_path,
               "--kv-gb", str(kv_gb),
               "--max-batch", str(max_batch),
               "--tp-backend", tp_backend]
        if device_override is not None:
            cmd += ["--device", device_override]
        if weights:
            cmd += ["--weights", str(weights)]
        if kv_dtype:
            cmd += ["--kv-dtype", str(kv_dtype)]
        procs.append(subprocess.Popen(cmd, env=env))
    return procs
