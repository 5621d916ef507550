"""In-process CPU stub worker — the fake backend for plumbing tests.

Exercises every gateway/router path with no GPU (BASELINE.json config 1;
SURVEY.md §4 implication (a)).  Echoes canned tokens with configurable
latency, supports streaming, typed throttling, and the fault-injection
hook ('kill' / 'hang' / 'error') that drives the failover tests.
"""

from __future__ import annotations

import asyncio
import time
from typing import AsyncIterator

from .base import (
    GenerationChunk, GenerationRequest, GenerationResult, Worker,
    WorkerDead, WorkerError, WorkerThrottled,
)

_CANNED = ("Resilient inference on MI355X: the scheduler places each request "
           "on a healthy GPU and the ledger records where it ran.").split()


def estimate_tokens(text: str) -> int:
    """Cheap whitespace+punctuation token estimate used by the stub and by
    the gateway's pre-call charge (≈chars/4 floor)."""
    return max(1, len(text) // 4)


class StubWorker(Worker):
    def __init__(self, device_index: str, models: set[str],
                 token_delay_ms: float = 0.0, first_token_ms: float = 0.0,
                 max_concurrency: int = 0) -> None:
        super().__init__(device=f"stub:{device_index}", models=models)
        self.token_delay_ms = token_delay_ms
        self.first_token_ms = first_token_ms
        self.max_concurrency = max_concurrency   # 0 = unlimited
        self.fault_mode = "none"
        self._in_flight = 0
        self.total_served = 0

    @property
    def in_flight(self) -> int:
        return self._in_flight

    def _check_fault(self) -> None:
        if self.fault_mode == "kill":
            raise WorkerDead(f"{self.device} injected fault: killed")
        if self.fault_mode == "error":
            raise WorkerError(f"{self.device} injected fault: error")

    async def _maybe_hang(self) -> None:
        while self.fault_mode == "hang":
            await asyncio.sleep(0.05)

    def _admit(self) -> None:
        if self.max_concurrency and self._in_flight >= self.max_concurrency:
            raise WorkerThrottled(f"{self.device} queue full")

    def _prompt_tokens(self, req: GenerationRequest) -> int:
        return sum(estimate_tokens(str(m.get("content", ""))) for m in req.messages)

    async def generate(self, req: GenerationRequest) -> GenerationResult:
        self._check_fault()
        self._admit()
        self._in_flight += 1
        try:
            t0 = time.monotonic()
            await self._maybe_hang()
            self._check_fault()
            if self.first_token_ms:
                await asyncio.sleep(self.first_token_ms / 1000.0)
            n = min(req.max_tokens, len(_CANNED))
            words = _CANNED[:n]
            if self.token_delay_ms:
                await asyncio.sleep(self.token_delay_ms * n / 1000.0)
            self._check_fault()
            self.total_served += 1
            text = " ".join(words)
            finish = "length" if n == req.max_tokens else "stop"
            from .engine_worker import _earliest_stop
            cut = _earliest_stop(text, req.stop)
            if cut >= 0:
                text, finish = text[:cut], "stop"
            return GenerationResult(
                text=text,
                prompt_tokens=self._prompt_tokens(req),
                completion_tokens=n,
                finish_reason=finish,
                ttft_ms=(time.monotonic() - t0) * 1000.0,
            )
        finally:
            self._in_flight -= 1

    async def _stream_impl(self, req: GenerationRequest) -> AsyncIterator[GenerationChunk]:
        self._check_fault()
        self._admit()
        self._in_flight += 1
        try:
            await self._maybe_hang()
            if self.first_token_ms:
                await asyncio.sleep(self.first_token_ms / 1000.0)
            n = min(req.max_tokens, len(_CANNED))
            for i, w in enumerate(_CANNED[:n]):
                self._check_fault()   # mid-stream fault -> failover path
                if self.token_delay_ms:
                    await asyncio.sleep(self.token_delay_ms / 1000.0)
                last = i == n - 1
                yield GenerationChunk(
                    text=(w if i == 0 else " " + w),
                    finish_reason=("length" if n == req.max_tokens else "stop") if last else None,
                    prompt_tokens=self._prompt_tokens(req) if i == 0 else None,
                )
            self.total_served += 1
        finally:
            self._in_flight -= 1

    def generate_stream(self, req: GenerationRequest) -> AsyncIterator[GenerationChunk]:
        return self._stream_impl(req)

    async def health(self) -> dict:
        if self.fault_mode in ("kill", "hang"):
            raise WorkerDead(f"{self.device} unhealthy (fault={self.fault_mode})")
        return {"device": self.device, "status": "ok", "models": sorted(self.models),
                "in_flight": self._in_flight, "total_served": self.total_served}

    async def inject_fault(self, mode: str) -> None:
        assert mode in ("none", "kill", "hang", "error")
        self.fault_mode = mode
