"""Continuous-batching inference engine (one per GPU worker).

The serving hot loop (SURVEY.md §3.2 "MI355X equivalent"): admitted
sequences share every decode step (one fused forward for the whole
batch); new arrivals are prefilled in batched varlen form and join the
running batch the same iteration.  Prefill-priority scheduling, paged KV
with full-reservation admission (a sequence admitted never dies of OOM —
over-capacity surfaces at admission as Throttled, X13).

The decode step is deliberately shaped for hipGraph capture (static
tensor shapes per batch-size bucket — see engine/graph.py).
"""

from __future__ import annotations

import collections
import dataclasses
import os
import threading
import time
from typing import Optional

import torch

from .. import ops
from .kv_cache import PagedKVCache, block_hash_chain

# roctx phase markers for rocprofv3 --marker-trace:
# torch.cuda.nvtx lowers to roctx on ROCm.  Opt-in: RLLI_ROCTX=1.
_ROCTX = os.environ.get("RLLI_ROCTX") == "1"


@dataclasses.dataclass
class SamplingParams:
    max_tokens: int = 128
    temperature: float = 0.0
    top_p: float = 1.0
    presence_penalty: float = 0.0     # subtract once per seen output token
    frequency_penalty: float = 0.0    # subtract per occurrence
    seed: Optional[int] = None
    stop_on_eos: bool = True

    @property
    def needs_torch_sampling(self) -> bool:
        """Rows the fused gumbel-max kernel cannot serve: nucleus
        filtering or repetition penalties (penalties need the row's
        output-token histogram, kept host-side)."""
        return ((self.temperature > 0 and self.top_p < 1.0)
                or self.presence_penalty != 0.0
                or self.frequency_penalty != 0.0)


@dataclasses.dataclass
class StepOutput:
    req_id: str
    token_id: int
    finished: bool
    finish_reason: Optional[str] = None   # stop | length
    n_output_tokens: int = 0


@dataclasses.dataclass
class SeqState:
    req_id: str
    prompt_ids: list
    params: SamplingParams
    output_ids: list = dataclasses.field(default_factory=list)
    blocks: list = dataclasses.field(default_factory=list)
    block_keys: Optional[list] = None      # prefix-cache content keys
    default_seed: int = 0                  # deterministic per arrival order
    n_cached: int = 0
    arrived_at: float = dataclasses.field(default_factory=time.monotonic)

    def blocks_needed(self, block_size: int) -> int:
        return -(-(len(self.prompt_ids) + self.params.max_tokens) // block_size)


class CapacityExceeded(RuntimeError):
    """Queue + KV budget cannot take this request now (-> 429/Throttled)."""


class LLMEngine:
    def __init__(self, model, kv_cache: PagedKVCache,
                 max_batch_size: int = 64,
                 max_prefill_tokens: int = 8192,
                 max_queue: int = 256,
                 max_model_len: Optional[int] = None,
                 chunk_size: int = 2048,
                 admission_window_s: float = 0.0,
                 enable_prefix_caching: bool = True,
                 target_step_ms: Optional[float] = None,
                 eos_ids=None,
                 spec_lookup: int = 0,
                 seed: int = 0) -> None:
        self.model = model
        self.kv = kv_cache
        self.device = model.device
        self.max_batch_size = max_batch_size
        self.max_prefill_tokens = max_prefill_tokens
        self.max_queue = max_queue
        self.max_model_len = max_model_len or getattr(
            model.config, "max_position", 8192)
        self.seed = seed
        # per-model EOS: taken from the model config
        # (Llama-3 presets carry 128001/128009) unless the caller
        # overrides — the worker passes the tokenizer's/deployment's ids
        if eos_ids is not None:
            self.eos_ids = frozenset(int(e) for e in eos_ids)
        else:
            self.eos_ids = frozenset(getattr(model.config, "eos_ids", (2,)))
        # waiting/abort are the ONLY cross-thread surfaces: guarded by a
        # small queue lock so enqueue NEVER waits on a running step (a
        # coarse lock here serialized arrivals behind 100ms prefills and
        # fragmented bursts into waves)
        self._queue_lock = threading.Lock()
        self.waiting: collections.deque[SeqState] = collections.deque()
        self.prefilling: list[SeqState] = []   # admitted, prompt not fully cached
        self.running: list[SeqState] = []
        self.chunk_size = chunk_size
        self.enable_prefix_caching = enable_prefix_caching
        # coalesce request bursts into one prefill batch: hold admission
        # while arrivals are still landing (MUST be 0 under TP lockstep —
        # wall-clock decisions would diverge across ranks)
        self.admission_window_s = admission_window_s
        # latency-targeted scheduler: when set, the per-step prefill
        # token budget is AIMD-tuned so mixed steps stay near this bound
        # (decode cadence / inter-token-latency SLO).  MUST be None
        # under TP lockstep — wall-clock decisions diverge across ranks
        # (same rule as admission_window_s).
        self.target_step_ms = target_step_ms
        # prompt-lookup speculative decoding (opt-in, greedy-only):
        # propose up to N continuation tokens by n-gram match against
        # the sequence's own context and verify them in ONE multi-row
        # forward (the chunked-prefill machinery — proposals attend the
        # paged prefix + themselves causally).  Rejected proposals cost
        # nothing to roll back: their KV slots are simply rewritten when
        # the real tokens are fed.
        self.spec_lookup = spec_lookup
        self._prefill_budget = max_prefill_tokens
        self._budget_floor = min(max_prefill_tokens, 256)
        self.step_count = 0
        self.block_size = kv_cache.block_size
        self._aborted: set[str] = set()
        self._live: set[str] = set()   # rids currently owned by the engine
        # live migration (SURVEY.md §5.8 state-migration): extraction and
        # adoption are queued control ops executed BETWEEN steps on the
        # engine thread — the only thread that may touch KV/seq state
        self._extract_reqs: set[str] = set()
        self._extracted: dict[str, dict] = {}
        self._adopt_queue: list[dict] = []
        self._adopt_results: dict[str, object] = {}
        self._arrival_counter = 0
        self._batch_dirty = True
        self._graph_loaded = False
        self._pending = None          # (seqs, event, host_tokens) 1-deep
        self._pinned = None
        self.stats = {"prefill_steps": 0, "decode_steps": 0,
                      "prefill_time": 0.0, "decode_time": 0.0,
                      "prefill_tokens": 0, "decode_tokens": 0,
                      # host-side phase split of the mixed/prefill step
                      # (prep = python+tensor build, fwd = launch, sample
                      # = kernel + tolist sync): finds host-bound stalls
                      "prefill_prep_time": 0.0, "prefill_fwd_time": 0.0,
                      "prefill_sample_time": 0.0,
                      "admit_events": []}
        # decode graph runner installed by engine/graph.py (GPU only)
        self.graph_runner = None
        # RLLI_STEP_TRACE=1: per-step wall-clock ring (t_start, kind,
        # batch, ms) — finds serving-context stalls the aggregate
        # counters average away
        self._trace = [] if os.environ.get("RLLI_STEP_TRACE") == "1" else None

    # ------------------------------------------------------------- admin
    @property
    def n_active(self) -> int:
        return len(self.waiting) + len(self.prefilling) + len(self.running)

    def has_work(self) -> bool:
        return bool(self.waiting or self.prefilling or self.running
                    or self._pending or self._adopt_queue
                    or self._extract_reqs)

    def add_request(self, req_id: str, prompt_ids: list,
                    params: Optional[SamplingParams] = None) -> None:
        params = params or SamplingParams()
        # out-of-vocab ids would fault the GPU inside the embedding
        # gather (hardware exception, not a python error) — reject at
        # the API boundary instead
        vocab = getattr(self.model.config, "vocab_size", None)
        if vocab and prompt_ids:
            lo, hi = min(prompt_ids), max(prompt_ids)
            if lo < 0 or hi >= vocab:
                raise ValueError(
                    f"prompt token id out of range [0, {vocab}): "
                    f"min={lo} max={hi}")
        if len(self.waiting) >= self.max_queue:
            raise CapacityExceeded(f"queue full ({self.max_queue})")
        seq = SeqState(req_id=req_id, prompt_ids=list(prompt_ids), params=params)
        self._arrival_counter += 1
        seq.default_seed = (self.seed * 1000003
                            + self._arrival_counter) & 0x7fffffffffffffff
        if self.enable_prefix_caching:
            seq.block_keys = block_hash_chain(seq.prompt_ids, self.block_size)
        if len(seq.prompt_ids) + params.max_tokens > self.max_model_len:
            raise CapacityExceeded(
                f"prompt+max_tokens = "
                f"{len(seq.prompt_ids) + params.max_tokens} exceeds "
                f"max_model_len {self.max_model_len}")
        if seq.blocks_needed(self.block_size) > self.kv.num_blocks:
            raise CapacityExceeded(
                f"request needs {seq.blocks_needed(self.block_size)} KV "
                f"blocks, cache has {self.kv.num_blocks}")
        with self._queue_lock:
            self.waiting.append(seq)
            self._live.add(req_id)

    def abort(self, req_id: str) -> None:
        """Abort is a no-op for rids the engine no longer owns (finished /
        never added), so post-completion aborts cannot grow ``_aborted``
        without bound (one leaked entry per request)."""
        with self._queue_lock:
            if req_id in self._live:
                self._aborted.add(req_id)

    # --------------------------------------------------------- scheduling
    def _admit(self) -> list[SeqState]:
        with self._queue_lock:
            return self._admit_locked()

    def _admit_locked(self) -> list[SeqState]:
        admitted: list[SeqState] = []
        if (self.waiting and self.admission_window_s > 0
                and time.monotonic() - self.waiting[-1].arrived_at
                < self.admission_window_s
                and len(self.waiting) + len(self.running)
                + len(self.prefilling) < self.max_batch_size):
            return admitted
        tokens = 0
        while (self.waiting
               and len(self.running) + len(self.prefilling) + len(admitted)
               < self.max_batch_size):
            seq = self.waiting[0]
            if seq.req_id in self._aborted:
                self.waiting.popleft()
                self._aborted.discard(seq.req_id)
                self._live.discard(seq.req_id)
                continue
            if tokens + len(seq.prompt_ids) > self.max_prefill_tokens and admitted:
                break
            # prefix cache: acquire the longest cached chain of FULL
            # prompt blocks (keep >= 1 suffix token so the first-output
            # logits are always computed)
            reused: list = []
            if seq.block_keys:
                max_reuse = (len(seq.prompt_ids) - 1) // self.block_size
                reused = self.kv.lookup_prefix(seq.block_keys[:max_reuse])
            needed = seq.blocks_needed(self.block_size) - len(reused)
            if self.spec_lookup:
                # speculative verification writes KV up to spec_lookup
                # positions past the committed length
                needed += -(-self.spec_lookup // self.block_size)
            if needed > self.kv.free_blocks:
                if reused:
                    self.kv.free(reused)
                break
            seq.blocks = reused + self.kv.allocate(needed)
            seq.n_cached = len(reused) * self.block_size
            tokens += len(seq.prompt_ids) - seq.n_cached
            admitted.append(self.waiting.popleft())
        if admitted and len(self.stats["admit_events"]) < 1000:
            self.stats["admit_events"].append(
                (round(time.monotonic(), 4), len(admitted), len(self.waiting)))
        return admitted

    def _slot(self, seq: SeqState, pos: int) -> int:
        return seq.blocks[pos // self.block_size] * self.block_size \
            + pos % self.block_size

    def _finish(self, seq: SeqState, outs: list[StepOutput], token: int) -> bool:
        p = seq.params
        reason = None
        if p.stop_on_eos and token in self.eos_ids:
            reason = "stop"
        elif len(seq.output_ids) >= p.max_tokens:
            reason = "length"
        outs.append(StepOutput(req_id=seq.req_id, token_id=token,
                               finished=reason is not None,
                               finish_reason=reason,
                               n_output_tokens=len(seq.output_ids)))
        if reason is not None:
            self.kv.free(seq.blocks)
            seq.blocks = []
            self._live.discard(seq.req_id)
            return True
        return False

    # -------------------------------------------------------------- steps
    def step(self) -> list[StepOutput]:
        """One engine iteration: drain the async sampling pipeline, then a
        prefill batch if arrivals are waiting and fit, else one decode for
        the running batch.

        On GPU the decode loop is pipelined one step deep: the forward
        consumes the previous step's sampled tokens directly on device,
        while their host copy (needed only for EOS/finish bookkeeping)
        arrives asynchronously and is processed at the NEXT step() — so
        steady-state decode has no host<->device synchronization at all."""
        self.step_count += 1
        outs = self._process_pending()
        self._drop_aborted()
        self._do_extracts()
        self._do_adopts()
        self.prefilling.extend(self._admit())
        if self.prefilling:
            t0 = time.monotonic()
            if _ROCTX:
                torch.cuda.nvtx.range_push(
                    f"mixed d{len(self.running)}+p{len(self.prefilling)}")
            nd, np_ = len(self.running), len(self.prefilling)
            n_chunk_tokens, new_outs = self._mixed_step()
            outs += new_outs
            if _ROCTX:
                torch.cuda.nvtx.range_pop()
            elapsed = time.monotonic() - t0
            self.stats["prefill_steps"] += 1
            self.stats["prefill_time"] += elapsed
            self.stats["prefill_tokens"] += n_chunk_tokens
            if self._trace is not None:
                self._trace.append((t0, f"mix d{nd}+p{np_}({n_chunk_tokens}t)",
                                    round(elapsed * 1e3, 2)))
            self._tune_budget(elapsed)
            return outs
        if self.running:
            t0 = time.monotonic()
            if _ROCTX:
                torch.cuda.nvtx.range_push(f"decode b{len(self.running)}")
            nb = len(self.running)
            spec_outs = None
            if self.spec_lookup > 0 and self._spec_eligible():
                spec_outs = self._decode_step_spec()
            if spec_outs is not None:
                outs += spec_outs
            else:
                outs += self._decode_step()
            if _ROCTX:
                torch.cuda.nvtx.range_pop()
            self.stats["decode_steps"] += 1
            elapsed = time.monotonic() - t0
            self.stats["decode_time"] += elapsed
            if self._trace is not None:
                self._trace.append((t0, f"dec b{nb}", round(elapsed * 1e3, 2)))
            return outs
        return outs

    def _process_pending(self) -> list[StepOutput]:
        if self._pending is None:
            return []
        seqs, event, buf = self._pending
        self._pending = None
        if event is not None:
            event.synchronize()
        tokens = buf.tolist() if isinstance(buf, torch.Tensor) else buf
        outs: list[StepOutput] = []
        still: list[SeqState] = []
        running_ids = {id(s) for s in self.running}
        for seq, tok in zip(seqs, tokens):
            if id(seq) not in running_ids:
                continue               # aborted between forward and landing
            seq.output_ids.append(tok)
            if not self._finish(seq, outs, tok):
                still.append(seq)
        if len(still) != len(seqs):
            self._batch_dirty = True
        self.running = still
        self.stats["decode_tokens"] += len(outs)
        return outs

    def _drop_aborted(self) -> None:
        if not self._aborted:
            return
        for attr in ("running", "prefilling"):
            keep = []
            for seq in getattr(self, attr):
                if seq.req_id in self._aborted:
                    self.kv.free(seq.blocks)
                    seq.blocks = []
                    self._aborted.discard(seq.req_id)
                    self._live.discard(seq.req_id)
                else:
                    keep.append(seq)
            if len(keep) != len(getattr(self, attr)):
                self._batch_dirty = True
            setattr(self, attr, keep)
        # an abort that raced a normal finish matches no live sequence:
        # clear it here so _aborted cannot accumulate stale rids
        with self._queue_lock:
            self._aborted &= self._live

    # ------------------------------------------------------ live migration
    def request_extract(self, req_id: str) -> None:
        """Ask the engine thread to extract a live request's full state
        (tokens + sampling identity + KV blocks) at the next step
        boundary; collect it with :meth:`take_extracted`."""
        with self._queue_lock:
            self._extract_reqs.add(req_id)

    def take_extracted(self, req_id: str):
        with self._queue_lock:
            return self._extracted.pop(req_id, None)

    def queue_adopt(self, state: dict) -> None:
        """Hand an extracted state to this engine; adoption happens on
        the engine thread at the next step boundary.  Outcome via
        :meth:`take_adopt_result` ("ok" or an exception)."""
        with self._queue_lock:
            self._adopt_queue.append(state)

    def take_adopt_result(self, req_id: str):
        with self._queue_lock:
            return self._adopt_results.pop(req_id, None)

    def _do_extracts(self) -> None:
        if not self._extract_reqs:      # racy-read fast path: a newly
            return                      # queued rid lands next step
        with self._queue_lock:
            if not self._extract_reqs:
                return
            rids = self._extract_reqs
            self._extract_reqs = set()
        for rid in rids:
            state = None
            for attr in ("running", "prefilling", "waiting"):
                pool = getattr(self, attr)
                if attr == "waiting":
                    # waiting is the cross-thread surface (add_request
                    # appends under the queue lock) — mutate it there
                    with self._queue_lock:
                        seq = next((x for x in pool if x.req_id == rid),
                                   None)
                        if seq is not None:
                            pool.remove(seq)
                    if seq is None:
                        continue
                    state = {"rid": rid,
                             "prompt_ids": list(seq.prompt_ids),
                             "output_ids": [],
                             "params": dataclasses.asdict(seq.params),
                             "default_seed": seq.default_seed,
                             "n_cached": 0,
                             "block_size": self.block_size,
                             "kv": None}
                    if seq.blocks:
                        self.kv.free(seq.blocks)
                        seq.blocks = []
                    break
                for seq in list(pool):
                    if seq.req_id != rid:
                        continue
                    state = {"rid": rid,
                             "prompt_ids": list(seq.prompt_ids),
                             "output_ids": list(seq.output_ids),
                             "params": dataclasses.asdict(seq.params),
                             "default_seed": seq.default_seed,
                             "n_cached": seq.n_cached,
                             "block_size": self.block_size,
                             "kv": None}
                    # KV rides along only for decode-steady sequences;
                    # a mid-prefill extraction re-prefills on the target
                    # (prefix caching there may still serve it)
                    if attr == "running" and seq.n_cached > 0:
                        nb = -(-seq.n_cached // self.block_size)
                        idx = torch.tensor(seq.blocks[:nb],
                                           dtype=torch.long,
                                           device=self.kv.k.device)
                        state["kv"] = (
                            self.kv.k.index_select(1, idx).to("cpu"),
                            self.kv.v.index_select(1, idx).to("cpu"))
                    else:
                        state["n_cached"] = 0
                        state["output_ids"] = []
                    if seq.blocks:
                        self.kv.free(seq.blocks)
                        seq.blocks = []
                    if isinstance(pool, list):
                        pool.remove(seq)
                    else:
                        pool.remove(seq)   # deque supports remove too
                    self._batch_dirty = True
                    break
                if state is not None:
                    break
            with self._queue_lock:
                self._live.discard(rid)
                # prune unclaimed extractions (caller timed out before
                # the state landed): they hold CPU copies of KV
                now = time.monotonic()
                for k in [k for k, v in self._extracted.items()
                          if isinstance(v, dict)
                          and now - v.get("_t", now) > 120]:
                    self._extracted.pop(k, None)
                if isinstance(state, dict):
                    state["_t"] = now
                # "missing" (not None): a finished/unknown rid must be
                # distinguishable from not-yet-extracted, or the
                # worker's poll waits its full timeout for every
                # request that finished just before the sweep reached
                # it — which is exactly what a drain under load hits
                self._extracted[rid] = state if state is not None \
                    else "missing"

    def _do_adopts(self) -> None:
        if not self._adopt_queue:       # racy-read fast path
            return
        with self._queue_lock:
            if not self._adopt_queue:
                return
            batch = self._adopt_queue
            self._adopt_queue = []
        for state in batch:
            rid = state["rid"]
            try:
                params = SamplingParams(**state["params"])
                seq = SeqState(req_id=rid,
                               prompt_ids=list(state["prompt_ids"]),
                               params=params,
                               output_ids=list(state["output_ids"]),
                               default_seed=state["default_seed"],
                               n_cached=int(state["n_cached"]))
                kv_pair = state.get("kv")
                usable = (kv_pair is not None
                          and state.get("block_size") == self.block_size
                          and seq.output_ids
                          and seq.n_cached >= len(seq.prompt_ids))
                if usable:
                    # allocate ONLY on the direct-to-running path; the
                    # waiting path re-allocates in _admit (allocating
                    # in both places leaked the first reservation)
                    nb_need = seq.blocks_needed(self.block_size)
                    if self.spec_lookup:
                        nb_need += -(-self.spec_lookup // self.block_size)
                    seq.blocks = self.kv.allocate(nb_need)
                    k_src, v_src = kv_pair
                    nb = k_src.shape[1]
                    idx = torch.tensor(seq.blocks[:nb], dtype=torch.long,
                                       device=self.kv.k.device)
                    self.kv.k.index_copy_(
                        1, idx, k_src.to(self.kv.k.device, self.kv.k.dtype))
                    self.kv.v.index_copy_(
                        1, idx, v_src.to(self.kv.v.device, self.kv.v.dtype))
                    self.running.append(seq)
                else:
                    seq.n_cached = 0
                    seq.output_ids = []
                    if self.enable_prefix_caching:
                        seq.block_keys = block_hash_chain(
                            seq.prompt_ids, self.block_size)
                    with self._queue_lock:
                        self.waiting.append(seq)
                with self._queue_lock:
                    self._live.add(rid)
                    if len(self._adopt_results) > 256:
                        # drop the OLDEST unclaimed half (insertion
                        # order): clearing everything could wipe a
                        # result a concurrent migrate_in is polling
                        # for, making a successful adoption report
                        # "timed out" while the sequence runs headless
                        for k in list(self._adopt_results)[:128]:
                            self._adopt_results.pop(k, None)
                    self._adopt_results[rid] = "ok"
                self._batch_dirty = True
            except Exception as e:                        # noqa: BLE001
                with self._queue_lock:
                    self._adopt_results[rid] = e

    @staticmethod
    def _seq_seed(s: SeqState) -> int:
        """Per-(request, absolute output index) seed, wrapped to SIGNED
        int64 (full 64-bit value — a 63-bit mask here would discard the
        bit-63 carry at whatever step _rebuild_batch happened to run,
        while the sampling kernel's later ``+ GOLDEN*step`` additions wrap
        mod 2^64: tokens would depend on rebuild timing).
        Both the kernel (uint64 reinterpret) and the torch fallback see
        the same premix regardless of where the step offset was added."""
        base = s.params.seed if s.params.seed is not None else s.default_seed
        v = (base + 0x9e3779b97f4a7c15 * len(s.output_ids)) & 0xffffffffffffffff
        return v - 0x10000000000000000 if v >= 0x8000000000000000 else v

    def _sample(self, logits: torch.Tensor, seqs: list[SeqState]) -> list[int]:
        temps = torch.tensor([s.params.temperature for s in seqs],
                             dtype=torch.float32, device=self.device)
        seeds = torch.tensor([self._seq_seed(s) for s in seqs],
                             dtype=torch.int64, device=self.device)
        toks = ops.sample(logits, temps, seeds, 0)
        special = [i for i, s in enumerate(seqs)
                   if s.params.needs_torch_sampling]
        if special:
            toks = toks.clone()
            for i in special:
                toks[i] = self._sample_torch(logits[i], seqs[i])
        return toks.tolist()

    def _sample_torch(self, row_logits: torch.Tensor, s: SeqState) -> int:
        """Torch sampling path for rows the fused kernel cannot serve:
        repetition penalties (OpenAI presence/frequency semantics) and/or
        nucleus (top_p) filtering.  Greedy rows with penalties argmax
        the penalized logits."""
        p = s.params
        lg = row_logits.float()
        if p.presence_penalty or p.frequency_penalty:
            lg = lg.clone()
            counts: dict[int, int] = {}
            for t in s.output_ids:
                counts[t] = counts.get(t, 0) + 1
            for t, c in counts.items():
                lg[t] -= p.presence_penalty + p.frequency_penalty * c
        if p.temperature <= 0:
            return int(lg.argmax())
        probs = torch.softmax(lg / p.temperature, -1)
        if p.top_p < 1.0:
            sp, idx = probs.sort(descending=True)
            keep = int((sp.cumsum(0) < p.top_p).sum()) + 1
            sp = sp[:keep] / sp[:keep].sum()
        else:
            sp, idx = probs, torch.arange(probs.shape[0],
                                          device=probs.device)
        gen = torch.Generator(device=row_logits.device)
        gen.manual_seed(self._seq_seed(s))
        return int(idx[int(torch.multinomial(sp, 1, generator=gen))])

    def _tune_budget(self, elapsed_s: float) -> None:
        """AIMD latency-targeted scheduling: multiplicative decrease of
        the per-step prefill token budget proportional to overshoot of
        target_step_ms, additive increase while comfortably under it —
        long prompt bursts cannot stall the decode cadence for longer
        than (roughly) the SLO."""
        if self.target_step_ms is None:
            return
        ms = elapsed_s * 1e3
        if ms > self.target_step_ms:
            scaled = int(self._prefill_budget * self.target_step_ms / ms * 0.9)
            self._prefill_budget = max(self._budget_floor, scaled)
        elif ms < 0.8 * self.target_step_ms:
            self._prefill_budget = min(
                self.max_prefill_tokens,
                self._prefill_budget + max(64, self.chunk_size // 4))

    def _mixed_step(self) -> tuple[int, list[StepOutput]]:
        """One forward over [decode rows | prefill-chunk rows]: running
        sequences decode while prefilling sequences advance by up to
        chunk_size prompt tokens each (budget max_prefill_tokens/step).
        Chunk rows attend over the paged cache (prefix + the chunk
        itself), so a long prompt never blocks the decode batch for more
        than one bounded step."""
        dev = self.device
        seqs_d = list(self.running)
        B_d = len(seqs_d)
        input_ids = [s.output_ids[-1] for s in seqs_d]
        positions = [s.n_cached for s in seqs_d]
        slots = [self._slot(s, s.n_cached) for s in seqs_d]
        seq_lens_d = [s.n_cached + 1 for s in seqs_d]

        budget = self._prefill_budget
        chunk_plan: list = []
        for seq in self.prefilling:
            if budget <= 0:
                break
            n = min(len(seq.prompt_ids) - seq.n_cached, self.chunk_size, budget)
            if n <= 0:
                continue
            chunk_plan.append((seq, seq.n_cached, n))
            budget -= n

        c_row0, c_pos0, c_nrows, c_btrow = [], [], [], []
        bt_rows: list[list[int]] = []
        sample_idx = list(range(B_d))
        sampled_seqs = list(seqs_d)
        row = B_d
        for seq, start, n in chunk_plan:
            btr = len(bt_rows)
            bt_rows.append(seq.blocks)
            for q0 in range(0, n, 32):
                c_row0.append(row + q0)
                c_pos0.append(start + q0)
                c_nrows.append(min(32, n - q0))
                c_btrow.append(btr)
            input_ids.extend(seq.prompt_ids[start:start + n])
            positions.extend(range(start, start + n))
            slots.extend(self._slot(seq, p) for p in range(start, start + n))
            if start + n == len(seq.prompt_ids):
                sample_idx.append(row + n - 1)
                sampled_seqs.append(seq)
            row += n

        def i32(x):
            return torch.tensor(x, dtype=torch.int32, device=dev)

        def bt_tensor(rows):
            w = max((len(r) for r in rows), default=1)
            t = torch.zeros((max(len(rows), 1), w), dtype=torch.int32)
            for i, r in enumerate(rows):
                t[i, :len(r)] = torch.tensor(r, dtype=torch.int32)
            return t.to(dev)

        t_prep = time.monotonic()
        # fast path: no decode rows and every chunk is a whole prompt ->
        # classic batched varlen prefill over in-batch K/V (no paged
        # reads, no row-copy) — this is the bench/burst-arrival shape
        if B_d == 0 and all(start == 0 and start + n == len(s.prompt_ids)
                            for s, start, n in chunk_plan):
            cu = [0]
            for s, _, n in chunk_plan:
                cu.append(cu[-1] + n)
            ids_t, pos_t, slots_t, cu_t = (i32(input_ids), i32(positions),
                                           i32(slots), i32(cu))
            self.stats["prefill_prep_time"] += time.monotonic() - t_prep
            t_fwd = time.monotonic()
            logits = self.model.forward_prefill(ids_t, pos_t, self.kv,
                                                slots_t, cu_t)
            self.stats["prefill_fwd_time"] += time.monotonic() - t_fwd
            return self._mixed_finish(chunk_plan, sampled_seqs, logits)

        args = (i32(input_ids), i32(positions), self.kv, i32(slots), B_d,
                bt_tensor([s.blocks for s in seqs_d]) if B_d else None,
                i32(seq_lens_d) if B_d else None,
                i32(c_row0), i32(c_pos0), i32(c_nrows), i32(c_btrow),
                bt_tensor(bt_rows),
                torch.tensor(sample_idx, dtype=torch.long, device=dev))
        self.stats["prefill_prep_time"] += time.monotonic() - t_prep
        t_fwd = time.monotonic()
        logits = self.model.forward_mixed(*args)
        self.stats["prefill_fwd_time"] += time.monotonic() - t_fwd
        return self._mixed_finish(chunk_plan, sampled_seqs, logits,
                                  decode_seqs=seqs_d)

    def _mixed_finish(self, chunk_plan, sampled_seqs, logits,
                      decode_seqs=()) -> tuple[int, list[StepOutput]]:
        t_s = time.monotonic()
        tokens = self._sample(logits, sampled_seqs)
        self.stats["prefill_sample_time"] += time.monotonic() - t_s
        n_chunk_tokens = 0
        for seq, start, n in chunk_plan:
            seq.n_cached = start + n
            n_chunk_tokens += n
        outs: list[StepOutput] = []
        still_running: list[SeqState] = []
        decode_set = {id(s) for s in decode_seqs}
        for seq, tok in zip(sampled_seqs, tokens):
            if id(seq) in decode_set:
                seq.n_cached += 1
            else:
                # prompt fully prefilled: publish its full blocks to the
                # prefix cache (idempotent for reused blocks)
                if seq.block_keys:
                    for i, key in enumerate(
                            seq.block_keys[:len(seq.prompt_ids)
                                           // self.block_size]):
                        self.kv.register_block(seq.blocks[i], key)
            seq.output_ids.append(tok)
            if not self._finish(seq, outs, tok):
                still_running.append(seq)
        self.running = [s for s in self.running
                        if id(s) not in decode_set] + still_running
        done = {id(s) for s, start, n in chunk_plan
                if start + n == len(s.prompt_ids)}
        self.prefilling = [s for s in self.prefilling if id(s) not in done]
        self._batch_dirty = True
        return n_chunk_tokens, outs

    # -------------------------------------- speculative decoding (lookup)
    def _spec_eligible(self) -> bool:
        return self._pending is None and all(
            s.params.temperature == 0 and not s.params.needs_torch_sampling
            for s in self.running)

    @staticmethod
    def _propose_lookup(seq: SeqState, k: int) -> list:
        """Prompt-lookup: continuation after the most recent prior
        occurrence of the context's last bigram."""
        if k <= 0:
            return []
        ctx = seq.prompt_ids + seq.output_ids
        if len(ctx) < 4:
            return []
        a, b = ctx[-2], ctx[-1]
        for i in range(len(ctx) - 3, 0, -1):
            if ctx[i - 1] == a and ctx[i] == b:
                return list(ctx[i + 1:i + 1 + k])
        return []

    def _decode_step_spec(self) -> Optional[list]:
        seqs = self.running
        dev = self.device
        proposals = []
        any_prop = False
        for s in seqs:
            cap = min(self.spec_lookup,
                      s.params.max_tokens - len(s.output_ids) - 1)
            prop = self._propose_lookup(s, cap)
            proposals.append(prop)
            any_prop = any_prop or bool(prop)
        if not any_prop:
            return None                     # caller runs the normal step

        input_ids: list = []
        positions: list = []
        slots: list = []
        c_row0, c_pos0, c_nrows, c_btrow = [], [], [], []
        bt_rows: list[list] = []
        metas = []
        row = 0
        for s, prop in zip(seqs, proposals):
            toks = [s.output_ids[-1]] + prop
            btr = len(bt_rows)
            bt_rows.append(s.blocks)
            c_row0.append(row)
            c_pos0.append(s.n_cached)
            c_nrows.append(len(toks))
            c_btrow.append(btr)
            input_ids.extend(toks)
            positions.extend(range(s.n_cached, s.n_cached + len(toks)))
            slots.extend(self._slot(s, s.n_cached + j)
                         for j in range(len(toks)))
            metas.append((s, prop, row, len(toks)))
            row += len(toks)

        def i32(x):
            return torch.tensor(x, dtype=torch.int32, device=dev)

        def bt_tensor(rows):
            w = max((len(r) for r in rows), default=1)
            t = torch.zeros((max(len(rows), 1), w), dtype=torch.int32)
            for i, r in enumerate(rows):
                t[i, :len(r)] = torch.tensor(r, dtype=torch.int32)
            return t.to(dev)

        logits = self.model.forward_mixed(
            i32(input_ids), i32(positions), self.kv, i32(slots), 0,
            None, None, i32(c_row0), i32(c_pos0), i32(c_nrows), i32(c_btrow),
            bt_tensor(bt_rows),
            torch.arange(row, dtype=torch.long, device=dev))
        greedy = logits.argmax(-1).tolist()           # host sync (spec mode)

        outs: list[StepOutput] = []
        still: list[SeqState] = []
        accepted_total = 0
        for s, prop, r0, n in metas:
            g = greedy[r0:r0 + n]
            a = 0
            while a < len(prop) and prop[a] == g[a]:
                a += 1
            accepted_total += a
            finished = False
            for t in prop[:a] + [g[a]]:
                s.output_ids.append(t)
                if self._finish(s, outs, t):
                    finished = True
                    break
            # KV is real through the last ACCEPTED fed position; the
            # rejected proposal's slot gets rewritten when the real
            # token is fed next step
            s.n_cached += 1 + a
            if not finished:
                still.append(s)
        self.running = still
        self.stats["decode_tokens"] += len(outs)
        self.stats.setdefault("spec_steps", 0)
        self.stats.setdefault("spec_accepted", 0)
        self.stats["spec_steps"] += 1
        self.stats["spec_accepted"] += accepted_total
        self._batch_dirty = True           # positions advanced multi-token
        return outs

    def _rebuild_batch(self) -> None:
        """(Re)build persistent device tensors for the running batch.
        Block tables are complete at admission (full reservation), so
        between composition changes every decode step is device-side."""
        dev = self.device
        seqs = self.running
        max_blocks = max(len(s.blocks) for s in seqs)
        bt = torch.zeros((len(seqs), max_blocks), dtype=torch.int32)
        for i, s in enumerate(seqs):
            bt[i, :len(s.blocks)] = torch.tensor(s.blocks, dtype=torch.int32)
        self._b_ids = torch.tensor([s.output_ids[-1] for s in seqs],
                                   dtype=torch.int32, device=dev)
        self._b_pos = torch.tensor([s.n_cached for s in seqs],
                                   dtype=torch.int32, device=dev)
        self._b_bt = bt.to(dev)
        self._b_temps = torch.tensor([s.params.temperature for s in seqs],
                                     dtype=torch.float32, device=dev)
        self._b_seeds = torch.tensor([self._seq_seed(s) for s in seqs],
                                     dtype=torch.int64, device=dev)
        self._b_step_off = 0
        self._b_torch_sampling = any(s.params.needs_torch_sampling for s in seqs)
        self._batch_dirty = False
        self._graph_loaded = False

    def _decode_step(self) -> list[StepOutput]:
        seqs = self.running
        if getattr(self, "_batch_dirty", True) or self._b_ids.shape[0] != len(seqs):
            self._rebuild_batch()
        gr = self.graph_runner
        if gr is not None and gr.steady_ok(len(seqs), self._b_bt.shape[1]):
            # steady path: the WHOLE step (slots, forward, sample, state
            # advance) is one graph replay; the static buffers are loaded
            # only when the batch composition changed (graph.py)
            if not self._graph_loaded:
                gr.load(self._b_ids, self._b_pos, self._b_bt, self._b_temps,
                        self._b_seeds)
                self._graph_loaded = True
            tok_dev, logits = gr.step()
            if self._b_torch_sampling:
                # top-p / penalty rows: recompute on the host from the
                # captured logits and correct the fed tokens before the
                # next replay (_sample reproduces the fused kernel's
                # choice for plain rows — same seeds, same premix)
                tokens = self._sample(logits, seqs)
                tok_dev = torch.tensor(tokens, dtype=torch.int32,
                                       device=self.device)
                gr.override_tokens(tok_dev)
            for seq in seqs:
                seq.n_cached += 1      # KV of the fed token was appended
            return self._queue_pending(seqs, tok_dev)
        # eager path (no graphs, torch-side sampling, or out-of-envelope
        # block tables)
        pos = self._b_pos
        bt = self._b_bt
        block_idx = (pos // self.block_size).long()
        slots = bt.gather(1, block_idx.unsqueeze(1)).squeeze(1) \
            * self.block_size + (pos - block_idx.int() * self.block_size)
        seq_lens = pos + 1
        logits = self.model.forward_decode(self._b_ids, pos, self.kv,
                                           slots, bt, seq_lens)
        if self._b_torch_sampling:
            tokens = self._sample(logits, seqs)     # torch top-p path (sync)
            tok_dev = torch.tensor(tokens, dtype=torch.int32,
                                   device=self.device)
        else:
            tok_dev = ops.sample(logits, self._b_temps, self._b_seeds,
                                 self._b_step_off)
            self._b_step_off += 1
        for seq in seqs:
            seq.n_cached += 1          # KV of the fed token was appended
        # advance device state in place: the NEXT forward feeds tok_dev
        # without the host ever seeing it (a rebuild overrides if the
        # batch composition changes when the pending tokens land)
        self._b_ids.copy_(tok_dev)
        self._b_pos += 1
        return self._queue_pending(seqs, tok_dev)

    def _queue_pending(self, seqs: list, tok_dev: torch.Tensor) \
            -> list[StepOutput]:
        """Hand the sampled tokens to the async landing pipeline: D2H
        into the pinned ring, event-fenced; processed at the NEXT step
        so steady decode never synchronizes."""
        B = len(seqs)
        if self.device.type == "cuda":
            if self._pinned is None or self._pinned.shape[0] < self.max_batch_size:
                self._pinned = torch.empty(self.max_batch_size,
                                           dtype=torch.int32, pin_memory=True)
            host = self._pinned[:B]
            host.copy_(tok_dev, non_blocking=True)
            ev = torch.cuda.Event()
            ev.record()
            self._pending = (list(seqs), ev, host)
            return []
        self._pending = (list(seqs), None, tok_dev.tolist())
        return self._process_pending()
