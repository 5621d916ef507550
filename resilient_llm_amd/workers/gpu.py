"""GPU worker process management.

One worker process per MI355X (``HIP_VISIBLE_DEVICES`` pinning —
SURVEY.md §3.1 "MI355X equivalent"), serving its engine over a unix
socket; the gateway holds an RpcWorkerClient per process.  Killing the
process is REAL fault injection (mode 'kill' hard-exits it): heartbeats
fail, in-flight requests error, the router fails over — the behavior the
reference could only approximate by starving quotas.

Pool targets (``pool/<name>/<model>``) spawn one torchrun-style process
GROUP per pool with RCCL TP across the pool's GPUs; rank 0 serves the
socket (see workers/pool.py).
"""

from __future__ import annotations

import os
import subprocess
import sys
import tempfile

from ..config import Config
from ..utils.logging import log_with_timestamp
from .base import WorkerRegistry
from .rpc import RpcWorkerClient

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def spawn_gpu_worker(device_index: str, model_name: str,
                     socket_path: str, kv_gb: float = 24.0,
                     max_batch: int = 64, use_graphs: bool = True,
                     target_step_ms: float | None = None,
                     weights: str | None = None,
                     eos_ids: list | None = None,
                     kv_dtype: str | None = None,
                     quant: str | None = None,
                     spec_lookup: int | None = None,
                     extra_env: dict | None = None) -> subprocess.Popen:
    """``device_index`` may be a sub-device replica like ``0.1``: several
    worker processes co-located on physical GPU 0 — 288 GB of HBM3E holds
    many 8B replicas, and a co-located warm replica makes single-GPU hot
    failover real (SURVEY.md §5.4 warm-standby)."""
    env = dict(os.environ)
    env["HIP_VISIBLE_DEVICES"] = str(device_index).split(".")[0]
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    env["PYTHONPATH"] = REPO_ROOT + os.pathsep + env.get("PYTHONPATH", "")
    env.update(extra_env or {})
    cmd = [sys.executable, "-m", "resilient_llm_amd.workers.gpu_main",
           "--device-label", f"gpu:{device_index}",
           "--model", model_name,
           "--socket", socket_path,
           "--kv-gb", str(kv_gb),
           "--max-batch", str(max_batch)]
    if use_graphs:
        cmd.append("--graphs")
    if target_step_ms is not None:
        cmd += ["--target-step-ms", str(target_step_ms)]
    if weights:
        cmd += ["--weights", str(weights)]
    for e in eos_ids or []:
        cmd += ["--eos-id", str(e)]
    if kv_dtype:
        cmd += ["--kv-dtype", str(kv_dtype)]
    if quant:
        cmd += ["--quant", str(quant)]
    if spec_lookup:
        cmd += ["--spec-lookup", str(spec_lookup)]
    return subprocess.Popen(cmd, env=env)


def register_gpu_workers(config: Config, registry: WorkerRegistry,
                         kv_gb: float = 24.0, max_batch: int = 64,
                         use_graphs: bool = True) -> None:
    """Spawn one worker process per distinct gpu/pool target the config
    names and register proxies.  ``gpu/*`` spread targets resolve to the
    set of concrete workers; if ONLY ``*`` targets exist, one worker per
    visible GPU is spawned."""
    import torch

    gpu_models: dict[str, set[str]] = {}
    has_star = False
    star_models: set[str] = set()
    pools: dict[str, str] = {}
    opts: dict[str, dict] = {}   # target -> extra litellm_params (weights, eos_id)
    for d in config.deployments:
        if d.backend_kind == "gpu":
            if d.backend_target == "*":
                has_star = True
                star_models.add(d.backend_model)
            else:
                gpu_models.setdefault(d.backend_target, set()).add(d.backend_model)
                if d.params:
                    opts.setdefault(d.backend_target, d.params)
        elif d.backend_kind == "pool":
            pools[d.backend_target] = d.backend_model
            if d.params:
                opts.setdefault(f"pool:{d.backend_target}", d.params)
    if has_star and not gpu_models:
        n = torch.cuda.device_count()
        for i in range(n):
            gpu_models[str(i)] = set(star_models)
    elif has_star:
        for target in gpu_models:
            gpu_models[target] |= star_models

    run_dir = tempfile.mkdtemp(prefix="rlli-workers-")
    for target, models in sorted(gpu_models.items()):
        assert len(models) == 1, \
            f"one model per GPU worker in v1; gpu:{target} asked for {models}"
        (model_name,) = models
        sock = os.path.join(run_dir, f"gpu{target}.sock")
        log_with_timestamp(f"spawning worker gpu:{target} ({model_name})", "grey")
        slo = config.cluster.target_step_ms
        extra = opts.get(target, {})
        weights = extra.get("weights")
        kv_dtype = extra.get("kv_dtype")
        quant = extra.get("quantization")
        spec = extra.get("spec_lookup")
        eos = extra.get("eos_id")
        eos_list = None if eos is None else (
            [int(eos)] if isinstance(eos, int) else [int(e) for e in eos])
        proc = spawn_gpu_worker(target, model_name, sock, kv_gb=kv_gb,
                                max_batch=max_batch, use_graphs=use_graphs,
                                weights=weights, eos_ids=eos_list,
                                kv_dtype=kv_dtype, quant=quant,
                                spec_lookup=spec, target_step_ms=slo)
        client = RpcWorkerClient(f"gpu:{target}", {model_name}, sock)
        client.proc = proc
        client.respawn = (lambda t=target, m=model_name, s=sock:
                          spawn_gpu_worker(t, m, s, kv_gb=kv_gb,
                                           max_batch=max_batch,
                                           use_graphs=use_graphs,
                                           weights=weights, eos_ids=eos_list,
                                           kv_dtype=kv_dtype, quant=quant,
                                           spec_lookup=spec,
                                           target_step_ms=slo))
        registry.register("gpu", target, client)

    for pool_name, model_name in sorted(pools.items()):
        from .pool import spawn_pool_worker
        pool_def = config.cluster.pools[pool_name]
        sock = os.path.join(run_dir, f"pool-{pool_name}.sock")
        log_with_timestamp(
            f"spawning TP={pool_def.tensor_parallel} pool worker "
            f"pool:{pool_name} on GPUs {pool_def.gpus} ({model_name})", "grey")
        pool_opts = opts.get(f"pool:{pool_name}", {})
        weights = pool_opts.get("weights")
        pool_kvd = pool_opts.get("kv_dtype")
        procs = spawn_pool_worker(pool_def, model_name, sock, kv_gb=kv_gb,
                                  max_batch=max_batch, weights=weights,
                                  kv_dtype=pool_kvd)
        client = RpcWorkerClient(f"pool:{pool_name}", {model_name}, sock)
        client.proc = procs[0]
        client.proc_group = procs
        client.respawn = (lambda pd=pool_def, m=model_name, s=sock:
                          spawn_pool_worker(pd, m, s, kv_gb=kv_gb,
                                            max_batch=max_batch,
                                            weights=weights,
                                            kv_dtype=pool_kvd))
        registry.register("pool", pool_name, client)
