"""Worker abstraction: the gateway-side view of a backend.

The reference's "backend" is Amazon Bedrock behind an HTTPS call
(reference src/demo_cris.py:233-238).  Here a worker is a per-GPU engine
(one process per MI355X, ``workers/gpu.py``), a TP pool of GPUs
(``parallel/``), or an in-process CPU stub (``workers/stub.py`` — the fake
backend for plumbing tests, SURVEY.md §4 implication (a)).

Error taxonomy mirrors X13: :class:`WorkerThrottled` (backend over
capacity — queue/KV budget full) is distinguishable from hard errors
(:class:`WorkerError`) and dead workers (:class:`WorkerDead`), like
Bedrock's ThrottlingException vs other failures (demo_cris.py:261-283).
"""

from __future__ import annotations

import dataclasses
import random
import time
from typing import AsyncIterator, Optional


class WorkerError(Exception):
    """Hard failure inside a worker (counts toward cooldown)."""


class WorkerMigrated(WorkerError):
    """The request's state was live-migrated to a peer worker; the
    caller should re-route (the adopting worker continues token-exact,
    any other replica regenerates identically from the seed)."""


class WorkerThrottled(WorkerError):
    """Backend over capacity — typed Throttled status (X13)."""


class WorkerDead(WorkerError):
    """Worker process is gone / unresponsive (drives hot failover)."""


@dataclasses.dataclass
class GenerationRequest:
    request_id: str
    model: str                       # backend model name, e.g. llama-3-8b
    messages: list                   # OpenAI chat messages
    max_tokens: int = 128
    temperature: float = 0.0
    top_p: float = 1.0
    presence_penalty: float = 0.0
    frequency_penalty: float = 0.0
    seed: Optional[int] = None
    stream: bool = False
    consumer: str = "anonymous"
    stop: Optional[list] = None      # OpenAI stop sequences (<= 4 strings)


@dataclasses.dataclass
class GenerationChunk:
    text: str
    token_id: Optional[int] = None
    finish_reason: Optional[str] = None   # set on the last chunk
    prompt_tokens: Optional[int] = None   # set on the FIRST chunk


@dataclasses.dataclass
class GenerationResult:
    text: str
    prompt_tokens: int
    completion_tokens: int
    finish_reason: str = "stop"
    ttft_ms: Optional[float] = None


class Worker:
    """Interface the gateway dispatches onto."""

    def __init__(self, device: str, models: set[str]) -> None:
        self.device = device          # "gpu:0" | "pool:a" | "stub:0"
        self.models = models
        self.started_at = time.time()

    async def generate(self, req: GenerationRequest) -> GenerationResult:
        raise NotImplementedError

    def generate_stream(self, req: GenerationRequest) -> AsyncIterator[GenerationChunk]:
        raise NotImplementedError

    async def health(self) -> dict:
        return {"device": self.device, "status": "ok", "models": sorted(self.models)}

    async def inject_fault(self, mode: str) -> None:
        """Fault-injection hook: 'kill' | 'hang' | 'error' | 'none'."""
        raise NotImplementedError

    async def close(self) -> None:
        pass

    @property
    def in_flight(self) -> int:
        return 0


class WorkerRegistry:
    """device-spec -> worker resolution, including the ``*`` spread target.

    A deployment whose backend is ``gpu/*/<model>`` ("cross-GPU inference",
    X10) resolves to whichever worker holding that model currently has the
    fewest requests in flight — the scheduler-chooses-by-live-load
    analogue of Bedrock CRIS capacity-driven spreading.
    """

    def __init__(self) -> None:
        self._workers: dict[str, Worker] = {}

    def register(self, kind: str, target: str, worker: Worker) -> None:
        self._workers[f"{kind}:{target}"] = worker

    def get(self, kind: str, target: str, model: str) -> Worker:
        if target == "*":
            candidates = [w for k, w in self._workers.items()
                          if k.startswith(f"{kind}:") and model in w.models]
            if not candidates:
                raise WorkerDead(f"no {kind} worker holds model {model!r}")
            # admin drain: spread targets skip draining workers while
            # any alternative exists
            active = [w for w in candidates
                      if not getattr(w, "draining", False)]
            if active:
                candidates = active
            # capacity-driven spread: least in flight, random among ties —
            # mirrors CRIS's "distribution is capacity-driven, not
            # client-controlled" 
            lowest = min(w.in_flight for w in candidates)
            return random.choice([w for w in candidates if w.in_flight == lowest])
        try:
            w = self._workers[f"{kind}:{target}"]
        except KeyError:
            raise WorkerDead(f"no worker registered for {kind}:{target}") from None
        if model not in w.models:
            raise WorkerError(f"worker {kind}:{target} does not hold model {model!r}")
        return w

    def all(self) -> dict[str, Worker]:
        return dict(self._workers)

    async def close(self) -> None:
        for w in self._workers.values():
            await w.close()
