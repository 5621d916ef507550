"""Engine-backed worker: the per-GPU serving unit.

Owns one :class:`LLMEngine` (model + paged KV on one device) and exposes
the async Worker interface the gateway dispatches onto.  The engine's
blocking step loop runs on a dedicated thread; step outputs stream back
to per-request asyncio queues (time-to-first-token measured here).

Capacity pressure maps to the typed taxonomy (X13): CapacityExceeded ->
WorkerThrottled.  Fault injection supports 'error' / 'kill' / 'hang'
in-process; the subprocess wrapper (workers/gpu.py) escalates 'kill' to
killing the actual GPU process.
"""

from __future__ import annotations

import asyncio
import dataclasses
import os
import threading
import time
from typing import AsyncIterator, Optional

import torch

from ..engine import LLMEngine, PagedKVCache, SamplingParams
from ..engine.engine import CapacityExceeded
from ..models import LlamaForCausalLM, get_config
from ..utils.tokenizer import load_tokenizer
from .base import (
    GenerationChunk, GenerationRequest, GenerationResult, Worker,
    WorkerDead, WorkerError, WorkerMigrated, WorkerThrottled,
)

DEFAULT_KV_GB = 24.0


def default_num_blocks(config, kv_gb: float = DEFAULT_KV_GB,
                       block_size: int = 16, tp_world: int = 1) -> int:
    per_block = (config.n_layers * 2 * (config.n_kv_heads // tp_world)
                 * block_size * config.head_dim * 2)
    # cap: tiny test models would otherwise turn a GB budget into
    # millions of blocks (the allocator free list + zeros get silly)
    return max(64, min(65536, int(kv_gb * (1 << 30) / per_block)))


def render_prompt(messages: list) -> str:
    parts = []
    for m in messages:
        parts.append(f"{m.get('role', 'user')}: {m.get('content', '')}")
    parts.append("assistant:")
    return "\n".join(parts)


def _earliest_stop(text: str, stops) -> int:
    """Index of the earliest stop-sequence occurrence, or -1."""
    cut = -1
    for s in stops or []:
        i = text.find(s)
        if i >= 0 and (cut < 0 or i < cut):
            cut = i
    return cut


class EngineWorker(Worker):
    def __init__(self, device: str, model_name: str = "llama-3-8b",
                 device_label: Optional[str] = None,
                 kv_gb: float = DEFAULT_KV_GB,
                 max_batch_size: int = 64,
                 max_queue: int = 256,
                 max_prefill_tokens: int = 16384,
                 chunk_size: int = 512,
                 dtype: Optional[torch.dtype] = None,
                 num_blocks: Optional[int] = None,
                 use_graphs: bool = False,
                 tp_rank: int = 0, tp_world: int = 1, tp_group=None,
                 tp_control=None,
                 target_step_ms: Optional[float] = None,
                 weights: Optional[str] = None,
                 eos_id=None,
                 kv_dtype: str = "bf16",
                 quant=None,
                 spec_lookup: int = 0,
                 seed: int = 0) -> None:
        super().__init__(device=device_label or f"gpu:{device}",
                         models={model_name})
        self.model_name = model_name
        torch_device = device if ":" in str(device) or device == "cpu" \
            else f"cuda:{device}"
        config = get_config(model_name)
        if dtype is None:
            dtype = torch.bfloat16 if torch_device != "cpu" else torch.float32
        self.config = config
        self.model = LlamaForCausalLM(config, device=torch_device, dtype=dtype,
                                      tp_rank=tp_rank, tp_world=tp_world,
                                      tp_group=tp_group, seed=seed,
                                      quant=quant)
        if weights:
            n = self.model.load_safetensors(weights)
            from ..utils.logging import log_with_timestamp
            log_with_timestamp(f"{self.device}: loaded {n} tensors from "
                               f"{weights} (tp {tp_rank}/{tp_world})", "grey")
        self.tokenizer = load_tokenizer(weights, config.vocab_size)
        # EOS precedence: explicit deployment override > tokenizer-declared
        # ids (real HF tokenizer) > model-preset ids
        if eos_id is not None:
            eos_ids = [eos_id] if isinstance(eos_id, int) else list(eos_id)
        elif getattr(self.tokenizer, "eos_ids", None):
            eos_ids = list(self.tokenizer.eos_ids)
        else:
            eos_ids = list(getattr(config, "eos_ids", (2,)))
        self.eos_ids = frozenset(eos_ids)
        # opt-in fp8-e4m3 KV cache: half the bytes per token -> double
        # the resident KV capacity of the same budget (and half the
        # decode-attention HBM traffic); attention math stays f32 via
        # the gfx950 cvt_pk conversions (ops/csrc/common.h codecs)
        if kv_dtype not in ("bf16", "fp8"):
            raise ValueError(f"kv_dtype must be bf16 or fp8, got {kv_dtype!r}")
        use_fp8 = kv_dtype == "fp8"
        kv_torch_dtype = torch.float8_e4m3fn if use_fp8 else torch.bfloat16
        elem_bytes = 1 if use_fp8 else 2
        nb = num_blocks or (default_num_blocks(config, kv_gb,
                                               tp_world=tp_world)
                            * 2 // elem_bytes // 1)
        kv = PagedKVCache.for_model(config, nb, device=torch_device,
                                    tp_world=tp_world,
                                    kv_dtype=kv_torch_dtype)
        if dtype != torch.bfloat16 and not use_fp8:
            kv.k = kv.k.to(dtype)
            kv.v = kv.v.to(dtype)
        self.engine = LLMEngine(self.model, kv, max_batch_size=max_batch_size,
                                max_queue=max_queue, seed=seed,
                                max_prefill_tokens=max_prefill_tokens,
                                chunk_size=chunk_size,
                                eos_ids=self.eos_ids,
                                admission_window_s=0.0 if tp_control is not None
                                else 0.006,
                                # wall-clock tuning diverges under TP
                                # lockstep, like admission_window_s
                                target_step_ms=None if tp_control is not None
                                else target_step_ms,
                                spec_lookup=spec_lookup)
        if use_graphs and torch_device != "cpu":
            from ..engine.graph import install_graph_runner
            install_graph_runner(self.engine)
        self.tp_control = tp_control
        self.fault_mode = "none"
        self.total_served = 0
        self._req_counter = 0
        self._in_flight = 0
        # req_id -> [queue, loop, mode, buffer, first_token_t]
        # mode "stream": one queue item per token; mode "final": tokens
        # buffer engine-side, ONE queue item at finish (64x fewer loop
        # wakeups for non-streaming traffic)
        self._sinks: dict[str, list] = {}
        # live migration: adopted-but-unattached requests (gateway
        # request_id -> {"rid", "pre", "prompt_len", "buf", "done"});
        # outputs produced before the client re-attaches buffer here
        self._adopted: dict[str, dict] = {}
        self._adopted_by_rid: dict[str, dict] = {}
        self._adopt_lock = threading.Lock()
        self._work_event = threading.Event()
        self._stop = False
        self._thread = None
        # TP followers run the loop in their own main thread (pool_main)
        if tp_control is None or tp_control.is_leader:
            self._thread = threading.Thread(target=self._engine_loop,
                                            daemon=True,
                                            name=f"engine-{self.device}")
            self._thread.start()
        # stall watchdog: if the engine has work but step_count hasn't
        # moved for RLLI_STEP_WATCHDOG_S, dump every thread's stack to
        # stderr ONCE — turns a silent GPU/thread wedge into a
        # diagnosable trace (ops feature; cheap: one counter compare/s)
        wd_s = float(os.environ.get("RLLI_STEP_WATCHDOG_S", "60"))
        if wd_s > 0 and (tp_control is None or tp_control.is_leader):
            def _watchdog():
                import faulthandler
                import sys as _sys
                last = -1
                stuck_since = None
                dumped = False
                while not self._stop:
                    time.sleep(1.0)
                    sc = self.engine.step_count
                    if not self.engine.has_work():
                        stuck_since = None
                        dumped = False
                    elif sc != last:
                        stuck_since = None
                        dumped = False
                    elif stuck_since is None:
                        stuck_since = time.monotonic()
                    elif (not dumped
                          and time.monotonic() - stuck_since > wd_s):
                        print(f"[watchdog] {self.device}: engine stalled "
                              f">{wd_s:.0f}s at step {sc} "
                              f"(waiting={len(self.engine.waiting)} "
                              f"prefilling={len(self.engine.prefilling)} "
                              f"running={len(self.engine.running)})",
                              file=_sys.stderr, flush=True)
                        faulthandler.dump_traceback(file=_sys.stderr)
                        dumped = True
                    last = sc
            threading.Thread(target=_watchdog, daemon=True,
                             name=f"watchdog-{self.device}").start()

    # --------------------------------------------------------- engine loop
    # TP lockstep heartbeat: followers BLOCK inside the control-group
    # broadcast; without periodic empty syncs an IDLE pool would hit the
    # gloo collective timeout (parallel.init_pool_groups) and the
    # followers would crash.  Env-tunable for tests.
    TP_HEARTBEAT_S = float(os.environ.get("RLLI_TP_HEARTBEAT_S", "30"))

    def _engine_loop(self) -> None:
        last_sync = time.monotonic()
        while not self._stop:
            if self.fault_mode == "hang":
                time.sleep(0.05)
                continue
            has_work = self.engine.has_work() or (
                self.tp_control is not None and self.tp_control.pending())
            if not has_work:
                if (self.tp_control is not None and self.tp_control.is_leader
                        and time.monotonic() - last_sync > self.TP_HEARTBEAT_S):
                    self.tp_control.sync(self.engine)
                    last_sync = time.monotonic()
                self._work_event.wait(timeout=0.01)
                self._work_event.clear()
                continue
            if self.tp_control is not None:
                self.tp_control.sync(self.engine)
                last_sync = time.monotonic()
                if not self.engine.has_work():
                    continue
            try:
                outputs = self.engine.step()
            except Exception as e:  # engine-level failure -> fail all in flight
                import traceback
                from ..utils.logging import log_with_timestamp
                log_with_timestamp(
                    f"{self.device}: engine step failed: {e!r}", "red")
                traceback.print_exc()
                self._broadcast_error(e)
                continue
            now = time.monotonic()
            for out in outputs:
                sink = self._sinks.get(out.req_id)
                if sink is None:
                    # adopted-but-unattached: buffer under the adopt
                    # lock; if the attach raced us and registered the
                    # sink in between, fall through and deliver to it —
                    # dropping here would lose the token forever
                    with self._adopt_lock:
                        sink = self._sinks.get(out.req_id)
                        if sink is None:
                            entry = self._adopted_by_rid.get(out.req_id)
                            if entry is not None:
                                entry["buf"].append(out)
                                if out.finished:
                                    entry["done"] = True
                            continue
                q, loop, mode, buf, first_t = sink
                if sink[4] is None:
                    sink[4] = now
                if mode == "stream":
                    loop.call_soon_threadsafe(q.put_nowait, out)
                else:
                    buf.append(out)
                    if out.finished:
                        loop.call_soon_threadsafe(q.put_nowait,
                                                  (list(buf), sink[4]))

    def _broadcast_error(self, e: Exception) -> None:
        for q, loop, *_ in list(self._sinks.values()):
            loop.call_soon_threadsafe(q.put_nowait, e)

    # ----------------------------------------------------------- helpers
    def _check_fault(self) -> None:
        if self.fault_mode == "kill":
            raise WorkerDead(f"{self.device} injected fault: killed")
        if self.fault_mode == "error":
            raise WorkerError(f"{self.device} injected fault: error")

    def _enqueue(self, req: GenerationRequest,
                 mode: str = "stream") -> tuple[str, asyncio.Queue, int]:
        self._check_fault()
        prompt_ids = self.tokenizer.encode(render_prompt(req.messages))
        params = SamplingParams(max_tokens=req.max_tokens,
                                temperature=req.temperature,
                                top_p=req.top_p,
                                presence_penalty=req.presence_penalty,
                                frequency_penalty=req.frequency_penalty,
                                seed=req.seed)
        self._req_counter += 1
        rid = f"{req.request_id}-{self._req_counter}"
        q: asyncio.Queue = asyncio.Queue()
        loop = asyncio.get_running_loop()
        self._sinks[rid] = [q, loop, mode, [], None]
        try:
            if self.tp_control is not None:
                # leader: validate locally, then lockstep-broadcast the op
                blocks_needed = -(-(len(prompt_ids) + params.max_tokens)
                                  // self.engine.block_size)
                if blocks_needed > self.engine.kv.num_blocks:
                    raise CapacityExceeded(
                        f"request needs {blocks_needed} KV blocks, cache has "
                        f"{self.engine.kv.num_blocks}")
                if len(self.engine.waiting) >= self.engine.max_queue:
                    raise CapacityExceeded(f"queue full ({self.engine.max_queue})")
                # the FULL sampling params ride the broadcast: a dropped
                # field (r01 dropped the penalties) silently no-ops that
                # feature on TP pools and can diverge rank sampling
                self.tp_control.submit(
                    ("add", rid, prompt_ids, dataclasses.asdict(params)))
            else:
                self.engine.add_request(rid, prompt_ids, params)
        except CapacityExceeded as e:
            del self._sinks[rid]
            raise WorkerThrottled(str(e)) from e
        self._work_event.set()
        return rid, q, len(prompt_ids)

    def _cleanup(self, rid: str) -> None:
        self._sinks.pop(rid, None)

    @property
    def in_flight(self) -> int:
        return self._in_flight

    # --------------------------------------------------------------- API
    def _abort(self, rid: str) -> None:
        if self.tp_control is not None:
            self.tp_control.submit(("abort", rid))
            self._work_event.set()
        else:
            self.engine.abort(rid)

    # ------------------------------------------------------ live migration
    def _find_rid(self, request_id: str) -> Optional[str]:
        for rid in list(self._sinks):
            if rid.rsplit("-", 1)[0] == request_id:
                return rid
        with self._adopt_lock:
            e = self._adopted.get(request_id)
        return e["rid"] if e else None

    def list_requests(self) -> list:
        """Gateway-side request ids of everything live on this worker
        (attached sinks + adopted-unattached) — the evacuation list for
        drain-with-migration."""
        ids = {rid.rsplit("-", 1)[0] for rid in self._sinks}
        with self._adopt_lock:
            ids |= set(self._adopted)
        return sorted(ids)

    async def migrate_out(self, request_id: str) -> bytes:
        """Extract a live request (tokens + sampling identity + KV) and
        return it serialized; the blocked generate/stream for it raises
        WorkerMigrated so the gateway re-routes.  TP pools are excluded
        (lockstep state lives on every rank)."""
        if self.tp_control is not None:
            raise WorkerError("live migration is not supported on TP pools")
        rid = self._find_rid(request_id)
        if rid is None:
            raise WorkerError(f"no live request {request_id!r}")
        self.engine.request_extract(rid)
        self._work_event.set()
        state = None
        for _ in range(250):
            state = self.engine.take_extracted(rid)
            if state is not None:
                break
            await asyncio.sleep(0.02)
        if state is None:
            raise WorkerError(f"extraction of {request_id!r} timed out")
        if state == "missing":
            # finished/unknown in the engine: also drop any stale
            # adopted-entry so the NEXT sweep's list_requests no longer
            # offers this id (an r02 soak retried the same ghost id in
            # every sweep for its full 600 s prune window)
            with self._adopt_lock:
                e = self._adopted.pop(request_id, None)
                if e:
                    self._adopted_by_rid.pop(e["rid"], None)
            raise WorkerError(f"no live request {request_id!r} "
                              "(finished before extraction)")
        # the state has LEFT this engine: an adopted-unattached entry
        # here is now a ghost (its tokens travel in the blob; the target
        # registers a fresh entry on migrate_in) — drop it so later
        # sweeps/attaches don't find a dead id
        with self._adopt_lock:
            e = self._adopted.pop(request_id, None)
            if e:
                self._adopted_by_rid.pop(e["rid"], None)
        # NOTE: the blocked client is NOT released here — the caller
        # releases it via release_migrated() AFTER the target has
        # adopted, so the re-routed attach always finds the state
        # (releasing first raced the adoption and caused duplicate
        # regeneration on the target)
        import io

        def _pack():
            bio = io.BytesIO()
            torch.save(state, bio)
            return bio.getvalue()
        # serialize OFF the event loop: a multi-MB torch.save would
        # stall health/stream RPCs and flap this worker unhealthy
        return await asyncio.to_thread(_pack)

    async def release_migrated(self, request_id: str) -> None:
        """Raise WorkerMigrated into the request's blocked
        generate/stream so the gateway re-routes (called once the
        target holds the state — or on migration failure, in which
        case the re-route regenerates identically from the seed)."""
        rid = self._find_rid(request_id)
        sink = self._sinks.get(rid) if rid else None
        if sink is not None:
            sink[0].put_nowait(WorkerMigrated(
                f"request {request_id} migrated off {self.device}"))

    async def migrate_in(self, blob: bytes) -> None:
        """Adopt a migrated request: KV blocks land in this worker's
        cache and generation continues token-exact; outputs buffer until
        the client re-attaches (generate/stream with the same
        request_id)."""
        if self.tp_control is not None:
            raise WorkerError("live migration is not supported on TP pools")
        import io
        state = await asyncio.to_thread(
            lambda: torch.load(io.BytesIO(blob), weights_only=False))
        rid = state["rid"]
        # register the catch buffer BEFORE the engine can run the
        # adopted sequence — outputs between adoption and registration
        # would otherwise drop on the floor (no sink, no entry)
        entry = {"rid": rid, "pre": list(state["output_ids"]),
                 "prompt_len": len(state["prompt_ids"]),
                 "buf": [], "done": False, "t": time.monotonic()}
        with self._adopt_lock:
            # prune abandoned adoptions (client never re-attached):
            # finished entries older than 10 min, or oldest when large
            now = time.monotonic()
            stale = [k for k, e in self._adopted.items()
                     if (e.get("done") and now - e.get("t", now) > 600)
                     or len(self._adopted) > 1024]
            for k in stale:
                e = self._adopted.pop(k, None)
                if e:
                    self._adopted_by_rid.pop(e["rid"], None)
            self._adopted[rid.rsplit("-", 1)[0]] = entry
            self._adopted_by_rid[rid] = entry
        self.engine.queue_adopt(state)
        self._work_event.set()
        res = None
        for _ in range(250):
            res = self.engine.take_adopt_result(rid)
            if res is not None:
                break
            await asyncio.sleep(0.02)
        if res is None or isinstance(res, Exception):
            with self._adopt_lock:
                self._adopted.pop(rid.rsplit("-", 1)[0], None)
                self._adopted_by_rid.pop(rid, None)
            if res is None:
                raise WorkerError("adoption timed out")
            raise WorkerError(f"adoption failed: {res}")

    def _attach(self, req: GenerationRequest, mode: str):
        """Re-attach a client to an adopted request: returns the sink
        plus the tokens generated BEFORE attachment (migrated 'pre'
        tokens + anything buffered since adoption), or None when the
        request_id is not adopted here."""
        with self._adopt_lock:
            entry = self._adopted.pop(req.request_id, None)
            if entry is None:
                return None
            self._adopted_by_rid.pop(entry["rid"], None)
            rid = entry["rid"]
            q: asyncio.Queue = asyncio.Queue()
            loop = asyncio.get_running_loop()
            sink = [q, loop, mode, [], None]
            self._sinks[rid] = sink
            buffered = list(entry["buf"])
        pre = list(entry["pre"])
        if mode == "stream":
            for out in buffered:
                q.put_nowait(out)
        else:
            sink[3].extend(buffered)
            if buffered and buffered[-1].finished:
                q.put_nowait((list(sink[3]), time.monotonic()))
        return rid, q, entry["prompt_len"], pre

    async def generate(self, req: GenerationRequest) -> GenerationResult:
        t0 = time.monotonic()
        attached = self._attach(req, mode="final")
        if attached is not None:
            rid, q, n_prompt, pre = attached
        else:
            rid, q, n_prompt = self._enqueue(req, mode="final")
            pre = []
        self._in_flight += 1
        finished = False
        try:
            got = await q.get()
            self._check_fault()
            if isinstance(got, WorkerError):
                raise got                       # typed (e.g. WorkerMigrated)
            if isinstance(got, Exception):
                raise WorkerError(f"engine error: {got}") from got
            outs, first_t = got
            finished = True    # engine freed the sequence itself: no abort
            token_ids = pre + [o.token_id for o in outs]
            finish = outs[-1].finish_reason or "stop"
            text = self.tokenizer.decode(token_ids)
            cut = _earliest_stop(text, req.stop)
            if cut >= 0:
                text = text[:cut]
                finish = "stop"
            self.total_served += 1
            return GenerationResult(
                text=text,
                prompt_tokens=n_prompt,
                completion_tokens=len(token_ids),
                finish_reason=finish,
                ttft_ms=max(0.0, (first_t - t0) * 1000.0))
        finally:
            self._in_flight -= 1
            self._cleanup(rid)
            if not finished:   # abort only requests the engine still owns
                self._abort(rid)

    async def _stream_impl(self, req: GenerationRequest) -> AsyncIterator[GenerationChunk]:
        attached = self._attach(req, mode="stream")
        if attached is not None:
            rid, q, n_prompt, pre = attached
        else:
            rid, q, n_prompt = self._enqueue(req)
            pre = []
        self._in_flight += 1
        finished = False
        # migrated 'pre' tokens replay as ONE CHUNK PER TOKEN so the
        # gateway's failover replay-skip (which counts chunks == output
        # tokens from index 0) stays in units with a fresh regeneration
        import types as _types
        pre_outs = [_types.SimpleNamespace(token_id=t, finished=False,
                                           finish_reason=None) for t in pre]
        pre_idx = 0
        try:
            emitted = ""
            token_ids: list[int] = []
            stops = req.stop or []
            # hold back enough text that a stop sequence spanning chunk
            # boundaries is never partially emitted
            hold = max((len(s) - 1 for s in stops), default=0)
            while True:
                if pre_idx < len(pre_outs):
                    out = pre_outs[pre_idx]
                    pre_idx += 1
                else:
                    out = await q.get()
                self._check_fault()
                if isinstance(out, WorkerError):
                    raise out                   # typed (e.g. WorkerMigrated)
                if isinstance(out, Exception):
                    raise WorkerError(f"engine error: {out}") from out
                token_ids.append(out.token_id)
                full = self.tokenizer.decode_stream(token_ids, out.finished)
                cut = _earliest_stop(full, stops)
                if cut >= 0:
                    delta = full[len(emitted):cut]
                    emitted = full[:cut]
                    yield GenerationChunk(text=delta, token_id=out.token_id,
                                          finish_reason="stop",
                                          prompt_tokens=n_prompt
                                          if len(token_ids) == 1 else None)
                    break
                visible = len(full) if out.finished else max(
                    len(emitted), len(full) - hold)
                delta = full[len(emitted):visible]
                emitted = full[:visible]
                yield GenerationChunk(
                    text=delta, token_id=out.token_id,
                    finish_reason=out.finish_reason if out.finished else None,
                    prompt_tokens=n_prompt if len(token_ids) == 1 else None)
                if out.finished:
                    finished = True   # engine-side finish: no abort needed
                    break
            self.total_served += 1
        finally:
            self._in_flight -= 1
            self._cleanup(rid)
            # a stop-sequence cut or client disconnect leaves the engine
            # still generating: abort it; a normal finish needs nothing
            if not finished:
                self._abort(rid)

    def generate_stream(self, req: GenerationRequest) -> AsyncIterator[GenerationChunk]:
        return self._stream_impl(req)

    async def health(self) -> dict:
        if self.fault_mode in ("kill", "hang"):
            raise WorkerDead(f"{self.device} unhealthy (fault={self.fault_mode})")
        return {"device": self.device, "status": "ok",
                "models": sorted(self.models),
                "model": self.model_name,
                "in_flight": self._in_flight,
                "queued": len(self.engine.waiting),
                "running": len(self.engine.running),
                "kv_free_blocks": self.engine.kv.free_blocks,
                "prefix_cache": {
                    "hits": self.engine.kv.prefix_hits,
                    "lookups": self.engine.kv.prefix_lookups,
                },
                "total_served": self.total_served,
                "engine_stats": {k: v for k, v in self.engine.stats.items()
                                 if k != "admit_events"}}

    async def inject_fault(self, mode: str) -> None:
        assert mode in ("none", "kill", "hang", "error")
        self.fault_mode = mode

    async def close(self) -> None:
        self._stop = True
        self._work_event.set()
        self._thread.join(timeout=5)
