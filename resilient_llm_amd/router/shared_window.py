"""Cross-process minute-window rate limiting.

The reference pins ``--num_workers 1`` because LiteLLM's in-memory RPM/TPM
windows are per process (reference bin/start-gateway.sh:56); round 1
reproduced that limitation for SO_REUSEPORT scale-out.  This module beats
it: window counters live in one mmap'd file, mutated under an fcntl file
lock, so N gateway processes sharing a port enforce ONE budget — rpm=3
admits exactly 3 across all of them.

Layout: 16-byte header (magic, slot count) + one 24-byte slot per
deployment ``(window:int64, req_count:int64, tok_count:int64)``.  Slots
are assigned by deployment order, which is identical in every process
(they all load the same config).  Windows roll on WALL clock (time.time)
— processes must agree on the epoch, so the per-process monotonic clock
cannot be used here.

Costs: one flock + read-modify-write per admission (~2-5 us).  Only
rate-limited deployments go through it; unlimited deployments never touch
the file (the router only engages shared slots when rpm/tpm is set).
"""

from __future__ import annotations

import fcntl
import mmap
import os
import struct
import time
from typing import Callable, Optional

from .token_bucket import RateSnapshot

_MAGIC = 0x524C4C49        # "RLLI"
_HDR = struct.Struct("<qq")        # magic, n_slots
_SLOT = struct.Struct("<qqq")      # window, req_count, tok_count
SLOT_SIZE = _SLOT.size


class SharedWindowFile:
    """One mmap'd counter file shared by every gateway process."""

    def __init__(self, path: str, n_slots: int) -> None:
        self.path = path
        self.n_slots = n_slots
        size = _HDR.size + n_slots * SLOT_SIZE
        # O_CREAT without truncation: every process opens the same file;
        # the first to win the init lock writes the header
        self.fd = os.open(path, os.O_RDWR | os.O_CREAT, 0o600)
        with self._locked():
            if os.fstat(self.fd).st_size < size:
                os.ftruncate(self.fd, size)
                os.pwrite(self.fd, _HDR.pack(_MAGIC, n_slots), 0)
        self.mm = mmap.mmap(self.fd, size)
        magic, slots = _HDR.unpack_from(self.mm, 0)
        assert magic == _MAGIC and slots == n_slots, \
            f"shared window file {path} does not match this config " \
            f"(slots {slots} != {n_slots})"

    def _locked(self):
        class _L:
            def __enter__(_s):
                fcntl.flock(self.fd, fcntl.LOCK_EX)

            def __exit__(_s, *a):
                fcntl.flock(self.fd, fcntl.LOCK_UN)
        return _L()

    def _off(self, slot: int) -> int:
        assert 0 <= slot < self.n_slots
        return _HDR.size + slot * SLOT_SIZE

    def read(self, slot: int) -> tuple[int, int, int]:
        return _SLOT.unpack_from(self.mm, self._off(slot))

    def write(self, slot: int, window: int, reqs: int, toks: int) -> None:
        _SLOT.pack_into(self.mm, self._off(slot), window, reqs, toks)

    def update(self, slot: int, fn):
        """fn(window, reqs, toks) -> (window, reqs, toks, result);
        executed under the file lock."""
        with self._locked():
            w, r, t = self.read(slot)
            w, r, t, result = fn(w, r, t)
            self.write(slot, w, r, t)
            return result

    def close(self) -> None:
        try:
            self.mm.close()
        finally:
            os.close(self.fd)


class SharedMinuteWindowLimiter:
    """Drop-in for :class:`MinuteWindowLimiter` backed by a shared slot.

    Same fixed-window semantics (the exact 3-of-5 contract, reference
    README.md:255-266), enforced across every process that opened the
    file."""

    def __init__(self, rpm: Optional[int], tpm: Optional[int],
                 shared: SharedWindowFile, slot: int,
                 clock: Callable[[], float] = time.time) -> None:
        self.rpm = rpm
        self.tpm = tpm
        self._shared = shared
        self._slot = slot
        self._clock = clock

    @staticmethod
    def _window_of(now: float) -> int:
        return int(now // 60.0)

    def _rolled(self, w, r, t):
        cur = self._window_of(self._clock())
        if w != cur:
            return cur, 0, 0
        return w, r, t

    def would_admit(self, tokens: int = 0) -> bool:
        if self.rpm is None and self.tpm is None:
            return True

        def fn(w, r, t):
            w, r, t = self._rolled(w, r, t)
            ok = not (self.rpm is not None and r + 1 > self.rpm) and \
                 not (self.tpm is not None and t + tokens > self.tpm)
            return w, r, t, ok
        return self._shared.update(self._slot, fn)

    def try_acquire(self, tokens: int = 0) -> bool:
        if self.rpm is None and self.tpm is None:
            return True

        def fn(w, r, t):
            w, r, t = self._rolled(w, r, t)
            if (self.rpm is not None and r + 1 > self.rpm) or \
               (self.tpm is not None and t + tokens > self.tpm):
                return w, r, t, False
            return w, r + 1, t + tokens, True
        return self._shared.update(self._slot, fn)

    def reconcile(self, estimated: int, actual: int) -> None:
        if self.rpm is None and self.tpm is None:
            return

        def fn(w, r, t):
            w, r, t = self._rolled(w, r, t)
            return w, r, max(0, t + actual - estimated), None
        self._shared.update(self._slot, fn)

    def release(self, tokens: int = 0) -> None:
        if self.rpm is None and self.tpm is None:
            return

        def fn(w, r, t):
            w, r, t = self._rolled(w, r, t)
            return w, max(0, r - 1), max(0, t - tokens), None
        self._shared.update(self._slot, fn)

    def snapshot(self) -> RateSnapshot:
        if self.rpm is None and self.tpm is None:
            return RateSnapshot(None, None, 0, 0)

        def fn(w, r, t):
            w, r, t = self._rolled(w, r, t)
            return w, r, t, (r, t)
        r, t = self._shared.update(self._slot, fn)
        return RateSnapshot(self.rpm, self.tpm, r, t)
