"""InvocationLedger unit tests — the Logs Insights analogue (X12):
distribution by each dimension, since/alias/status filters, stats
percentiles, recent ordering, JSONL append, and the maxlen bound."""

import json
import os
import tempfile

from resilient_llm_amd.obs.ledger import InvocationLedger, InvocationRecord


def rec(ts, device="gpu:0", alias="m", status="ok", consumer="k1",
        model_id="id0", lat=10.0, fallback=False, ct=4):
    return InvocationRecord(ts=ts, request_id=f"r{ts}", alias=alias,
                            model_id=model_id, device=device,
                            consumer=consumer, status=status,
                            is_fallback=fallback, prompt_tokens=8,
                            completion_tokens=ct, latency_ms=lat)


def test_distribution_dimensions_and_filters():
    led = InvocationLedger()
    led.record(rec(1.0, device="gpu:0"))
    led.record(rec(2.0, device="gpu:0", consumer="k2"))
    led.record(rec(3.0, device="gpu:1", alias="other"))
    led.record(rec(4.0, device="gpu:1", status="error"))

    assert led.distribution(by="device") == {"gpu:0": 2, "gpu:1": 1}
    assert led.distribution(by="device", status=None) == {"gpu:1": 2, "gpu:0": 2}
    assert led.distribution(by="consumer") == {"k1": 2, "k2": 1}
    assert led.distribution(by="alias", status=None)["m"] == 3
    assert led.distribution(by="device", alias="other") == {"gpu:1": 1}
    assert led.distribution(by="device", since=2.5) == {"gpu:1": 1}


def test_stats_percentiles_and_rates():
    led = InvocationLedger()
    for i in range(10):
        led.record(rec(float(i), lat=float(i * 10)))
    led.record(rec(10.0, status="throttled"))
    led.record(rec(11.0, status="error", fallback=True))
    s = led.stats()
    assert s["total"] == 12 and s["ok"] == 10
    assert s["throttled"] == 1 and s["errors"] == 1
    assert s["fallbacks"] == 1
    assert abs(s["success_rate"] - 10 / 12) < 1e-9
    assert s["latency_ms"]["p50"] == 40.0 or s["latency_ms"]["p50"] == 50.0
    assert s["latency_ms"]["p99"] == 90.0
    assert s["reqs_per_sec"] is not None


def test_recent_order_and_maxlen():
    led = InvocationLedger(maxlen=5)
    for i in range(8):
        led.record(rec(float(i)))
    r = led.recent(10)
    assert len(r) == 5
    assert r[0]["ts"] == 7.0 and r[-1]["ts"] == 3.0   # newest first


def test_jsonl_append():
    path = os.path.join(tempfile.mkdtemp(prefix="rlli-led-"), "led.jsonl")
    led = InvocationLedger(jsonl_path=path)
    led.record(rec(1.0))
    led.record(rec(2.0, status="error"))
    led.close()
    lines = [json.loads(l) for l in open(path)]
    assert len(lines) == 2
    assert lines[0]["request_id"] == "r1.0"
    assert lines[1]["status"] == "error"
