#!/usr/bin/env python3
"""Flagship serving benchmark — the driver contract.

Measures BASELINE.json's metric: sustained reqs/sec + success-rate + p50
latency @ 64 concurrent, Llama-3-8B bf16, synthetic prompts, random-init
weights, on N GPUs of one node (weak scaling: 64 concurrent clients per
GPU).

Full-stack path: every rank hosts one EngineWorker on its GPU behind the
unix-socket RPC; rank 0 additionally runs the OpenAI-compatible gateway
(router + ledger) and the closed-loop load generator over HTTP.  One
"step" = every concurrency slot completes exactly one request
(prompt 128 tokens -> 64 new tokens, greedy).  W untimed warmup steps,
then K timed steps bracketed by gloo barrier + torch.cuda.synchronize on
both sides; elapsed is the MAX over ranks; rank 0 prints ONE JSON line.

Launch (driver):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
"""

from __future__ import annotations

import argparse
import asyncio
import json
import os
import socket
import sys
import tempfile
import threading
import time

REPO_ROOT = os.path.dirname(os.path.abspath(__file__))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)

PROMPT_TOKENS = 128
OUTPUT_TOKENS = 64
CONCURRENCY_PER_GPU = 64

# distributed hardening (the first 8-GPU run happens
# blind — a dead rank must produce a diagnosis, not a silent hang):
# rendezvous + collective timeout, and a faulthandler watchdog that dumps
# all thread stacks and exits non-zero if a barrier never completes.
RENDEZVOUS_TIMEOUT_S = float(os.environ.get("RLLI_BENCH_RDZV_TIMEOUT_S", "300"))
WATCHDOG_S = float(os.environ.get("RLLI_BENCH_WATCHDOG_S", "1200"))

_RANK = os.environ.get("RANK", "0")


def log(msg):
    print(f"[bench r{_RANK}] {msg}", file=sys.stderr, flush=True)


def arm_watchdog(tag: str, seconds: float = WATCHDOG_S):
    """If this rank is still stuck here after ``seconds``, dump every
    thread's stack to stderr and hard-exit so torchrun tears the job
    down with a readable cause instead of hanging the driver."""
    import faulthandler
    log(f"watchdog armed: {tag} ({seconds:.0f}s)")
    faulthandler.dump_traceback_later(seconds, exit=True)


def disarm_watchdog():
    import faulthandler
    faulthandler.cancel_dump_traceback_later()


def free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


class LoopThread:
    """An asyncio loop on a daemon thread; run() schedules coroutines."""

    def __init__(self) -> None:
        self.loop = asyncio.new_event_loop()
        self._thread = threading.Thread(
            target=self.loop.run_forever, daemon=True)
        self._thread.start()

    def run(self, coro, timeout=None):
        return asyncio.run_coroutine_threadsafe(coro, self.loop).result(timeout)


def start_worker(loopth: LoopThread, device: str, label: str, sock: str,
                 model: str, max_batch: int, num_blocks: int,
                 use_graphs: bool, kv_dtype: str = "bf16",
                 quant=None):
    from resilient_llm_amd.workers.engine_worker import EngineWorker
    from resilient_llm_amd.workers.rpc import WorkerRpcServer

    async def _start():
        worker = EngineWorker(device=device, model_name=model,
                              device_label=label,
                              max_batch_size=max_batch,
                              max_queue=max_batch * 4,
                              num_blocks=num_blocks,
                              kv_dtype=kv_dtype,
                              quant=quant,
                              use_graphs=use_graphs)
        server = WorkerRpcServer(worker, sock)
        await server.start()
        return worker

    return loopth.run(_start(), timeout=600)


def spawn_gateways(n_gpus: int, model: str, port: int, run_dir: str,
                   n_procs: int) -> list:
    """Gateway PROCESSES sharing the port via SO_REUSEPORT (the same
    scale-out LiteLLM's --num_workers provides); the bench config has no
    rate limits, so per-process windows are irrelevant here."""
    import subprocess
    sockets = ",".join(os.path.join(run_dir, f"w{r}.sock")
                       for r in range(n_gpus))
    procs = []
    ready_files = []
    for g in range(n_procs):
        rf = os.path.join(run_dir, f"gw{g}.ready")
        ready_files.append(rf)
        env = dict(os.environ)
        env["PYTHONPATH"] = REPO_ROOT + os.pathsep + env.get("PYTHONPATH", "")
        procs.append(subprocess.Popen(
            [sys.executable, "-m", "resilient_llm_amd.gateway.bench_gateway",
             "--port", str(port), "--model", model, "--sockets", sockets,
             "--ready-file", rf], env=env))
    deadline = time.time() + 900
    while time.time() < deadline:
        if all(os.path.exists(rf) for rf in ready_files):
            return procs
        for p in procs:
            assert p.poll() is None, "gateway process died during startup"
        time.sleep(0.25)
    raise RuntimeError("gateways failed to come up")


class LoadGen:
    """Driver for the loadgen subprocess (scripts/loadgen.py) — the
    client side runs in its own process so rank 0's gateway loop does
    not share a GIL with it at 8-GPU request rates."""

    def __init__(self, port: int, concurrency: int, prompt_tokens: int,
                 output_tokens: int) -> None:
        import subprocess
        self.proc = subprocess.Popen(
            [sys.executable, os.path.join(REPO_ROOT, "scripts", "loadgen.py"),
             "--port", str(port), "--concurrency", str(concurrency),
             "--prompt-tokens", str(prompt_tokens),
             "--output-tokens", str(output_tokens)],
            stdin=subprocess.PIPE, stdout=subprocess.PIPE, text=True)
        ready = self.proc.stdout.readline().strip()
        assert ready == "READY", ready

    def round(self) -> dict:
        self.proc.stdin.write("ROUND\n")
        self.proc.stdin.flush()
        line = self.proc.stdout.readline()
        return json.loads(line)

    def close(self) -> None:
        try:
            self.proc.stdin.write("QUIT\n")
            self.proc.stdin.flush()
            self.proc.wait(timeout=10)
        except Exception:
            self.proc.kill()


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=4)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--concurrency", type=int, default=CONCURRENCY_PER_GPU,
                    help="concurrent clients per GPU (weak scaling)")
    ap.add_argument("--prompt-tokens", type=int, default=PROMPT_TOKENS)
    ap.add_argument("--output-tokens", type=int, default=OUTPUT_TOKENS)
    ap.add_argument("--no-graphs", action="store_true")
    ap.add_argument("--kv-dtype", default="bf16", choices=("bf16", "fp8"),
                    help="paged-KV element type (fp8 = opt-in e4m3 cache; "
                         "the headline metric stays bf16)")
    ap.add_argument("--quant", default=None, choices=("fp8",),
                    help="W8A8-fp8 weight quantization (opt-in mode; the "
                         "headline metric stays bf16)")
    ap.add_argument("--device", default=None,
                    help="override torch device (tests: cpu)")
    args = ap.parse_args()
    globals()["OUTPUT_TOKENS"] = args.output_tokens

    import torch
    import torch.distributed as dist

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    n_gpus = args.gpus
    assert world == n_gpus or world == 1, \
        f"WORLD_SIZE {world} must equal --gpus {n_gpus}"
    distributed = world > 1
    if distributed:
        import datetime
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        arm_watchdog("rendezvous", RENDEZVOUS_TIMEOUT_S + 60)
        dist.init_process_group(
            backend="gloo", rank=rank, world_size=world,
            timeout=datetime.timedelta(seconds=RENDEZVOUS_TIMEOUT_S))
        disarm_watchdog()
        log(f"rendezvous complete: rank {rank}/{world}")

    on_gpu = args.device is None and torch.cuda.is_available()
    # co-located ranks (rehearsal: 2 ranks on a 1-GPU box) wrap onto the
    # devices that exist; on a full node this is the identity mapping
    dev_index = local_rank % max(1, torch.cuda.device_count()) if on_gpu else 0
    device = args.device or (f"cuda:{dev_index}" if on_gpu else "cpu")
    if on_gpu:
        torch.cuda.set_device(dev_index)
        if dev_index != local_rank:
            log(f"co-located: local_rank {local_rank} -> cuda:{dev_index}")

    def sync(tag="sync"):
        arm_watchdog(tag)
        if on_gpu:
            torch.cuda.synchronize()
        if distributed:
            dist.barrier()
        disarm_watchdog()

    # shared run dir for worker sockets (rank 0 creates, broadcasts)
    if distributed:
        holder = [tempfile.mkdtemp(prefix="rlli-bench-")] if rank == 0 else [None]
        dist.broadcast_object_list(holder, src=0)
        run_dir = holder[0]
    else:
        run_dir = tempfile.mkdtemp(prefix="rlli-bench-")

    conc_total = args.concurrency * n_gpus
    max_seq_blocks = -(-(args.prompt_tokens + args.output_tokens + 8) // 16)
    num_blocks = args.concurrency * (max_seq_blocks + 1) + 64

    loopth = LoopThread()
    t_init = time.time()
    worker = start_worker(loopth, device, f"gpu:{rank}",
                          os.path.join(run_dir, f"w{rank}.sock"),
                          args.model, max_batch=args.concurrency,
                          num_blocks=num_blocks,
                          use_graphs=on_gpu and not args.no_graphs,
                          kv_dtype=args.kv_dtype, quant=args.quant)
    log(f"rank {rank}: worker up on {device} in {time.time() - t_init:.1f}s "
        f"({num_blocks} KV blocks)")
    sync("post-worker-startup barrier")

    port = None
    loadgen = None
    gateways: list = []
    rounds_timed: list = []
    if rank == 0:
        port = free_port()
        n_gw = max(1, min(4, (n_gpus + 1) // 2))
        gateways = spawn_gateways(n_gpus if distributed else 1, args.model,
                                  port, run_dir, n_gw)
        loadgen = LoadGen(port, conc_total, args.prompt_tokens,
                          args.output_tokens)
        log(f"{n_gw} gateway process(es) on :{port}, driving {conc_total} "
            f"concurrent clients (loadgen subprocess)")

    # ---- warmup ----
    for w in range(args.warmup):
        if rank == 0:
            res = loadgen.round()
            log(f"warmup {w}: {res['ok']}/{res['total']} ok")
        sync(f"warmup {w} barrier")

    # ---- timed ----
    sync("pre-timed barrier")
    t0 = time.monotonic()
    arm_watchdog("timed rounds", WATCHDOG_S)
    for k in range(args.steps):
        if rank == 0:
            rounds_timed.append(loadgen.round())
    disarm_watchdog()
    sync("post-timed barrier")
    elapsed = time.monotonic() - t0
    try:
        h = loopth.run(worker.health(), timeout=30)
        stats = h.get("engine_stats", {})
        log(f"rank {rank} engine stats: {stats}")
        admits = worker.engine.stats.get("admit_events", [])
        log(f"rank {rank} admits (t, n, still_waiting): {admits[:40]}")
        if worker.engine._trace:
            tr = worker.engine._trace[-220:]
            lines = []
            prev = None
            for t0, kind, ms in tr:
                gap = 0.0 if prev is None else (t0 - prev) * 1e3
                lines.append(f"{t0:.4f} +{gap:6.2f}ms {kind} host={ms}ms")
                prev = t0
            log("step trace (last 220):\n" + "\n".join(lines))
    except Exception:
        pass
    if distributed:
        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if rank == 0:
        loadgen.close()
        for g in gateways:
            g.terminate()
        n_req = sum(r["total"] for r in rounds_timed)
        n_ok = sum(r["ok"] for r in rounds_timed)
        lats = sorted(l for r in rounds_timed for l in r["latencies"])
        p50 = lats[len(lats) // 2] if lats else None
        p99 = lats[min(len(lats) - 1, int(0.99 * len(lats)))] if lats else None
        out_toks = sum(r["completion_tokens"] for r in rounds_timed)
        total_toks = out_toks + sum(r["prompt_tokens"] for r in rounds_timed)
        reqs_per_s = n_ok / elapsed
        result = {
            "metric": "sustained reqs/sec + success-rate + p50 latency "
                      "@ 64 concurrent, Llama-3-8B",
            "value": round(reqs_per_s, 3),
            "unit": "reqs/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000.0, 1),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp8-w8a8(gate_up/down/lm_head)+bf16"
                     if args.quant == "fp8" else "bf16",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": conc_total,
                "seq_len": args.prompt_tokens + args.output_tokens,
                "parallelism": f"dp{n_gpus}",
                "concurrency_per_gpu": args.concurrency,
                "prompt_tokens": args.prompt_tokens,
                "output_tokens": args.output_tokens,
                "success_rate": round(n_ok / n_req, 4) if n_req else None,
                "p50_latency_s": round(p50, 3) if p50 else None,
                "p99_latency_s": round(p99, 3) if p99 else None,
                "output_tokens_per_s": round(out_toks / elapsed, 1),
                "total_tokens_per_s": round(total_toks / elapsed, 1),
                "temperature": 0.0,
            },
        }
        print(json.dumps(result), flush=True)
    if distributed:
        dist.barrier()
        dist.destroy_process_group()
    os._exit(0)   # daemon loops/threads: exit hard once results are printed


if __name__ == "__main__":
    main()
