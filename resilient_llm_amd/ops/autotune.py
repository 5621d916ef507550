"""Measured per-shape GEMM algorithm selection (decode projections).

torch's F.linear lets hipBLASLt pick a kernel via torch's heuristic;
probing the library's full candidate list showed the best candidate
beats that pick by 13-40% on the skinny decode shapes of Llama-8B
(o-proj at batch 64: 19.5 -> 11.7 us) while LOSING on others (gate/up)
— so the only safe policy is to race them: at first sight of a
(M, N, K) shape, time torch's pick and the library's top candidates,
cache the winner, and route every later call accordingly.

Tuning needs device synchronization, so it cannot run inside hipGraph
capture: `tuned_linear` falls back to F.linear (without caching) when
the stream is capturing, and engines pre-tune their decode shapes
eagerly before graphs are captured (models/llama.py warmup_gemms).
"""

from __future__ import annotations

import os
import time

import torch
import torch.nn.functional as F

_cache: dict[tuple[int, int, int], int | None] = {}
_candidates: dict[tuple[int, int, int], list[int]] = {}
_MAX_CANDIDATES = 24
_TIME_ITERS = 20
# Default OFF: the tuner's per-GEMM wins (13-40% in isolation, see
# scripts/probe_hipblaslt.cpp) did NOT survive in context — the full
# serving bench measured 129.4 reqs/s tuned vs 139.1 with torch's picks
# on the same box.  The fast isolated algos are split-K-heavy and their
# workspace traffic competes with the rest of the decode stream.  Kept
# as an A/B lever (RLLI_LT=1) and for other model geometries.
_DISABLED = os.environ.get("RLLI_LT") != "1"


def _time_fn(fn, iters: int = _TIME_ITERS) -> float:
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.monotonic() - t0) / iters


def _tune(x: torch.Tensor, w: torch.Tensor) -> int | None:
    """Race torch's pick against the library's heuristic candidates;
    returns the winning algo index, or None when torch wins.  The top
    few candidates are remembered for the in-context pass."""
    key = (x.shape[0], w.shape[0], w.shape[1])
    torch_t = _time_fn(lambda: F.linear(x, w))
    best_algo, best_t = None, torch_t
    timed: list[tuple[float, int]] = []
    try:
        idxs = torch.ops.rlli.lt_heuristics(x, w, _MAX_CANDIDATES).tolist()
    except Exception:
        return None
    for idx in idxs:
        try:
            t = _time_fn(lambda: torch.ops.rlli.lt_linear(x, w, idx))
        except Exception:
            continue
        timed.append((t, idx))
        if t < best_t:
            best_algo, best_t = idx, t
    _candidates[key] = [i for _, i in sorted(timed)[:6]]
    # require a real margin over torch before pinning an algo: the
    # tuning sample is small and torch's pick is the safe default
    if best_algo is not None and best_t < 0.97 * torch_t:
        return int(best_algo)
    return None


def tuned_linear(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """F.linear with measured hipBLASLt algo selection (bf16, no bias)."""
    if _DISABLED:
        return F.linear(x, w)
    key = (x.shape[0], w.shape[0], w.shape[1])
    algo = _cache.get(key, -1)
    if algo == -1:
        if torch.cuda.is_current_stream_capturing():
            # cannot tune mid-capture; stay on torch for THIS call but
            # leave the shape untuned for a later eager visit
            return F.linear(x, w)
        algo = _tune(x, w)
        _cache[key] = algo
    if algo is None:
        return F.linear(x, w)
    try:
        return torch.ops.rlli.lt_linear(x, w, algo)
    except Exception:
        # an algo propagated to a row count it does not support
        _cache[key] = None
        return F.linear(x, w)


def enabled() -> bool:
    return not _DISABLED


def tune_in_context(run_fn, keys=None, iters: int = 6) -> dict:
    """Second tuning pass with the CORRECT objective: isolated per-GEMM
    wins did not survive the real decode stream (129.4 vs 139.1 reqs/s
    e2e — split-K workspace traffic competes with neighbouring
    kernels), so re-decide each shape by timing ``run_fn`` — a full
    decode step — with the cache pinned to torch vs each remembered
    candidate (greedy coordinate descent, one pass).  Call before graph
    capture; needs device syncs."""
    if _DISABLED:
        return {}
    report: dict = {}
    for key in (keys if keys is not None else list(_candidates)):
        if key not in _candidates:
            continue
        best, best_t = None, None
        for opt in [None] + _candidates[key]:
            _cache[key] = opt
            t = _time_fn(run_fn, iters)
            if best_t is None or t < best_t:
                best, best_t = opt, t
        _cache[key] = best
        report[key] = (best, best_t)
    return report


def propagate(tuned_keys) -> None:
    """Copy each in-context (N, K) decision to every other cached row
    count of the same weight (decode graph buckets share weights)."""
    by_nk = {(k[1], k[2]): _cache[k] for k in tuned_keys if k in _cache}
    for key in list(_cache):
        nk = (key[1], key[2])
        if nk in by_nk:
            _cache[key] = by_nk[nk]


def tuned_shapes() -> dict:
    """Snapshot of the tuning cache (observability/debugging)."""
    return dict(_cache)
