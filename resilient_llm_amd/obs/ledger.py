"""Per-GPU invocation ledger — the CloudWatch Logs Insights replacement.

The reference's observability plane writes Bedrock invocation logs to
CloudWatch and aggregates them with a Logs Insights query
(``stats count(*) by inferenceRegion`` — reference src/demo_cris.py:87-92)
behind a 3-phase polling protocol with a 60 s propagation retry ladder
(demo_cris.py:112-152, 381-405).  Here the ledger is in-process: a
synchronous read of per-GPU counters with no propagation delay (SURVEY.md
§3.3), queryable by the same dimensions the demos need — device ("region"),
pool ("account"), deployment id, consumer — plus serving-quality fields
(latency, time-to-first-token, token counts) CloudWatch never had.

Optionally appends JSONL for post-hoc analysis, and feeds the gateway's
Prometheus exposition (``GET /metrics`` — gateway/app.py
prometheus_metrics).
"""

from __future__ import annotations

import collections
import dataclasses
import json
import threading
import time
from typing import Iterable, Optional


@dataclasses.dataclass
class InvocationRecord:
    ts: float                       # unix time at completion
    request_id: str
    alias: str                      # what the client asked for
    model_id: str                   # deployment that served it (X8)
    device: str                     # "gpu:3" / "pool:a" / "stub:0" — the "region"
    consumer: str                   # api-key identity ("account")
    status: str                     # ok | error | throttled
    is_fallback: bool = False
    prompt_tokens: int = 0
    completion_tokens: int = 0
    latency_ms: float = 0.0
    ttft_ms: Optional[float] = None  # time to first token (streaming)

    def to_json(self) -> str:
        return json.dumps(dataclasses.asdict(self), separators=(",", ":"))


class InvocationLedger:
    def __init__(self, maxlen: int = 200_000, jsonl_path: Optional[str] = None) -> None:
        self._records: collections.deque[InvocationRecord] = collections.deque(maxlen=maxlen)
        self._lock = threading.Lock()
        self._jsonl_path = jsonl_path
        self._jsonl_file = open(jsonl_path, "a") if jsonl_path else None
        self.started_at = time.time()

    def record(self, rec: InvocationRecord) -> None:
        with self._lock:
            self._records.append(rec)
            if self._jsonl_file is not None:
                self._jsonl_file.write(rec.to_json() + "\n")
                self._jsonl_file.flush()

    # ------------------------------------------------------------ queries
    def _select(self, since: Optional[float], alias: Optional[str],
                status: Optional[str]) -> Iterable[InvocationRecord]:
        with self._lock:
            recs = list(self._records)
        for r in recs:
            if since is not None and r.ts < since:
                continue
            if alias is not None and r.alias != alias:
                continue
            if status is not None and r.status != status:
                continue
            yield r

    def distribution(self, by: str = "device", since: Optional[float] = None,
                     alias: Optional[str] = None,
                     status: Optional[str] = "ok") -> dict[str, int]:
        """``stats count(*) by <dim>`` — the Logs Insights analogue.

        ``by`` is one of device / model_id / alias / consumer / status.
        """
        counts: dict[str, int] = {}
        for r in self._select(since, alias, status):
            key = getattr(r, by)
            counts[str(key)] = counts.get(str(key), 0) + 1
        return dict(sorted(counts.items(), key=lambda kv: -kv[1]))

    def stats(self, since: Optional[float] = None,
              alias: Optional[str] = None) -> dict:
        lat: list[float] = []
        ttft: list[float] = []
        n_ok = n_err = n_thr = 0
        p_tok = c_tok = 0
        fallbacks = 0
        t_min = t_max = None
        for r in self._select(since, alias, None):
            if r.status == "ok":
                n_ok += 1
                lat.append(r.latency_ms)
                if r.ttft_ms is not None:
                    ttft.append(r.ttft_ms)
            elif r.status == "throttled":
                n_thr += 1
            else:
                n_err += 1
            p_tok += r.prompt_tokens
            c_tok += r.completion_tokens
            fallbacks += int(r.is_fallback)
            t_min = r.ts if t_min is None else min(t_min, r.ts)
            t_max = r.ts if t_max is None else max(t_max, r.ts)
        lat.sort()
        ttft.sort()

        def pct(xs: list[float], q: float) -> Optional[float]:
            if not xs:
                return None
            i = min(len(xs) - 1, int(round(q * (len(xs) - 1))))
            return xs[i]

        total = n_ok + n_err + n_thr
        span = (t_max - t_min) if (t_min is not None and t_max and t_max > t_min) else None
        return {
            "total": total, "ok": n_ok, "errors": n_err, "throttled": n_thr,
            "success_rate": (n_ok / total) if total else None,
            "fallbacks": fallbacks,
            "prompt_tokens": p_tok, "completion_tokens": c_tok,
            "latency_ms": {"p50": pct(lat, 0.50), "p90": pct(lat, 0.90),
                           "p99": pct(lat, 0.99),
                           "avg": (sum(lat) / len(lat)) if lat else None},
            "ttft_ms": {"p50": pct(ttft, 0.50), "p99": pct(ttft, 0.99)},
            "reqs_per_sec": (n_ok / span) if span else None,
        }

    def recent(self, n: int = 50) -> list[dict]:
        import dataclasses as _dc
        with self._lock:
            recs = list(self._records)[-n:]
        return [_dc.asdict(r) for r in reversed(recs)]

    def close(self) -> None:
        if self._jsonl_file is not None:
            self._jsonl_file.close()
            self._jsonl_file = None
