"""Per-deployment RPM/TPM rate limiting (capability X4).

The observable contract comes from the reference's demo outputs: with
``rpm: 3`` exactly 3 of 5 simultaneous requests succeed and 2 get HTTP 429
(reference README.md:255-266, config.yaml:81-100).  That is fixed-window
per-minute semantics — a counter that resets each clock minute — not a
leaky bucket, so that is what we implement.  TPM is charged optimistically
at admission (prompt estimate + max_tokens) and reconciled to actual usage
at completion, matching pre-call checks (X7: filter deployments that WOULD
breach before dispatch, reference config.yaml:108).
"""

from __future__ import annotations

import dataclasses
import threading
import time
from typing import Callable, Optional


@dataclasses.dataclass
class RateSnapshot:
    rpm_limit: Optional[int]
    tpm_limit: Optional[int]
    rpm_used: int
    tpm_used: int

    @property
    def rpm_remaining(self) -> Optional[int]:
        return None if self.rpm_limit is None else max(0, self.rpm_limit - self.rpm_used)

    @property
    def tpm_remaining(self) -> Optional[int]:
        return None if self.tpm_limit is None else max(0, self.tpm_limit - self.tpm_used)


class MinuteWindowLimiter:
    """Fixed-window requests/minute + tokens/minute counter.

    Thread-safe: the gateway is a single asyncio loop, but workers report
    completions from executor threads.  ``clock`` is injectable for tests.
    """

    def __init__(self, rpm: Optional[int] = None, tpm: Optional[int] = None,
                 clock: Callable[[], float] = time.monotonic) -> None:
        self.rpm = rpm
        self.tpm = tpm
        self._clock = clock
        self._lock = threading.Lock()
        self._window_start = self._window_of(clock())
        self._req_count = 0
        self._tok_count = 0

    @staticmethod
    def _window_of(now: float) -> int:
        return int(now // 60.0)

    def _roll(self, now: float) -> None:
        w = self._window_of(now)
        if w != self._window_start:
            self._window_start = w
            self._req_count = 0
            self._tok_count = 0

    def would_admit(self, tokens: int = 0) -> bool:
        """Pre-call check (X7): True iff one more request of ``tokens``
        estimated tokens fits in the current window."""
        with self._lock:
            self._roll(self._clock())
            if self.rpm is not None and self._req_count + 1 > self.rpm:
                return False
            if self.tpm is not None and self._tok_count + tokens > self.tpm:
                return False
            return True

    def try_acquire(self, tokens: int = 0) -> bool:
        """Atomically admit one request charging ``tokens`` (estimate)."""
        with self._lock:
            self._roll(self._clock())
            if self.rpm is not None and self._req_count + 1 > self.rpm:
                return False
            if self.tpm is not None and self._tok_count + tokens > self.tpm:
                return False
            self._req_count += 1
            self._tok_count += tokens
            return True

    def reconcile(self, estimated: int, actual: int) -> None:
        """Replace an admission-time token estimate with actual usage.

        Only adjusts the current window; if the window rolled since
        admission the estimate is already gone (and so is the budget it
        consumed — matching fixed-window semantics).
        """
        with self._lock:
            self._roll(self._clock())
            delta = actual - estimated
            self._tok_count = max(0, self._tok_count + delta)

    def release(self, tokens: int = 0) -> None:
        """Un-charge a request that was admitted but never dispatched
        (e.g. its worker died before starting — it must not double-bill
        on the fallback, SURVEY.md §7 hard-part 2)."""
        with self._lock:
            self._roll(self._clock())
            self._req_count = max(0, self._req_count - 1)
            self._tok_count = max(0, self._tok_count - tokens)

    def snapshot(self) -> RateSnapshot:
        with self._lock:
            self._roll(self._clock())
            return RateSnapshot(self.rpm, self.tpm, self._req_count, self._tok_count)
