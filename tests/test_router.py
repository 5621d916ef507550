import random

import pytest

from resilient_llm_amd.config import Deployment, RouterSettings
from resilient_llm_amd.router.core import (
    Router, RouterRateLimit, UnknownAlias,
)


class FakeClock:
    def __init__(self, t=0.0):
        self.t = t

    def __call__(self):
        return self.t


def make_deps():
    return [
        Deployment("primary", "gpu/0/llama-3-8b", "gpu0/primary", rpm=3, tpm=100000, weight=3),
        Deployment("lb", "gpu/0/llama-3-8b", "gpu0/r0", rpm=3, tpm=100000, weight=3),
        Deployment("lb", "gpu/1/llama-3-8b", "gpu1/r1", rpm=3, tpm=100000, weight=3),
        Deployment("backup", "gpu/1/llama-3-8b", "gpu1/fb", rpm=25, tpm=250000, weight=25),
    ]


def make_router(clock=None, fallbacks="default", **kw):
    if fallbacks == "default":
        fallbacks = {"primary": ["backup"]}
    settings = RouterSettings(fallbacks=fallbacks, **kw)
    return Router(make_deps(), settings, clock=clock or FakeClock(0.0),
                  rng=random.Random(1234))


def test_unknown_alias():
    r = make_router()
    with pytest.raises(UnknownAlias):
        r.acquire("nope")


def test_fallback_after_rpm_exhaustion():
    """Reference fallback pattern: rpm=3 primary, requests 4..N land on the
    backup alias transparently (reference README.md:167-180)."""
    r = make_router()
    picks = [r.acquire("primary", 10) for _ in range(10)]
    primary = [t for t in picks if not t.is_fallback]
    fallback = [t for t in picks if t.is_fallback]
    assert len(primary) == 3
    assert len(fallback) == 7
    assert all(t.deployment.model_id == "gpu1/fb" for t in fallback)


def test_rate_limit_when_no_fallback():
    """Quota isolation: alias with no fallback returns 429 semantics (X9)."""
    r = make_router(fallbacks={})
    for _ in range(3):
        r.acquire("primary", 10)
    with pytest.raises(RouterRateLimit):
        r.acquire("primary", 10)


def test_shuffle_spreads_over_replicas():
    r = make_router()
    seen = set()
    for _ in range(6):
        t = r.acquire("lb", 1)
        seen.add(t.deployment.model_id)
    assert seen == {"gpu0/r0", "gpu1/r1"}


def test_tpm_pre_call_check_filters():
    r = make_router(fallbacks={})
    t = r.acquire("primary", 99999)
    r.complete(t, actual_tokens=99999)
    with pytest.raises(RouterRateLimit):
        r.acquire("primary", 10)  # tpm window full (100000)


def test_cooldown_after_allowed_fails():
    clk = FakeClock(0.0)
    r = make_router(clock=clk, fallbacks={}, allowed_fails=2, cooldown_time=15.0)
    t1 = r.acquire("primary", 1)
    r.fail(t1)
    t2 = r.acquire("primary", 1)
    r.fail(t2)
    # two fails -> cooldown.  With no alternative, the LAST-RESORT pass
    # still serves the cooling (healthy, within-limits) deployment
    # rather than guaranteeing a client-visible 429 (r02 chaos-soak
    # lesson: cooling out the last replica turned blips into outages).
    before = r.last_resort_total
    t3 = r.acquire("primary", 1)
    assert t3.deployment.model_id == "gpu0/primary"
    assert r.last_resort_total == before + 1
    r.complete(t3)
    clk.t = 16.0
    t4 = r.acquire("primary", 1)
    assert t4.deployment.model_id == "gpu0/primary"
    assert r.last_resort_total == before + 1  # cooldown over: normal path


def test_cooldown_skips_when_peer_available():
    clk = FakeClock(0.0)
    r = make_router(clock=clk, fallbacks={}, allowed_fails=1,
                    cooldown_time=15.0)
    t = r.acquire("lb", 1)
    cooled = t.deployment.model_id
    r.fail(t)
    # the cooling replica is out of rotation while its peer serves
    # (2 < the peer's rpm window: stays on the NORMAL path throughout)
    for _ in range(2):
        t2 = r.acquire("lb", 1)
        assert t2.deployment.model_id != cooled
        r.complete(t2)
    assert r.last_resort_total == 0


def test_unhealthy_still_raises():
    r = make_router(fallbacks={})
    r.set_healthy("gpu0/primary", False)
    with pytest.raises(RouterRateLimit):
        r.acquire("primary", 1)  # last-resort never routes to unhealthy


def test_fail_refunds_rate_window():
    clk = FakeClock(0.0)
    r = make_router(clock=clk, fallbacks={}, allowed_fails=10)
    tickets = [r.acquire("primary", 1) for _ in range(3)]
    r.fail(tickets[0])  # refund -> one more admission fits this window
    r.acquire("primary", 1)


def test_failed_deployment_excluded_on_retry():
    r = make_router()
    t = r.acquire("lb", 1)
    first = t.deployment.model_id
    r.fail(t)
    t2 = r.acquire("lb", 1, exclude={id(t.state)})
    assert t2.deployment.model_id != first


def test_unhealthy_routes_to_fallback():
    """Hot failover: primary marked unhealthy (worker died) -> requests go
    to the backup without consuming primary budget (X6)."""
    r = make_router()
    r.set_healthy("gpu0/primary", False)
    t = r.acquire("primary", 1)
    assert t.is_fallback and t.deployment.model_id == "gpu1/fb"
    r.set_healthy("gpu0/primary", True)
    assert not r.acquire("primary", 1).is_fallback


def test_reconcile_frees_tpm_budget():
    r = make_router(fallbacks={})
    t = r.acquire("primary", 90000)
    r.complete(t, actual_tokens=50)
    t2 = r.acquire("primary", 90000)
    assert t2.deployment.model_id == "gpu0/primary"


def test_describe_shape():
    r = make_router()
    rows = r.describe()
    assert len(rows) == 4
    assert {"model_name", "model_id", "rpm_used", "healthy",
            "cooldown_remaining"} <= set(rows[0])


def test_least_busy_strategy():
    settings = RouterSettings(routing_strategy="least-busy", fallbacks={})
    r = Router(make_deps(), settings, clock=FakeClock(0.0))
    t1 = r.acquire("lb", 1)
    t2 = r.acquire("lb", 1)
    assert {t1.deployment.model_id, t2.deployment.model_id} == {"gpu0/r0", "gpu1/r1"}


# ------------------------------------------------------- prefix affinity
def _affinity_router(n=3, rpm=1000):
    deps = [Deployment("aff", f"gpu/{i}/llama-3-8b", f"gpu{i}/r{i}",
                       rpm=rpm, tpm=10**7, weight=1) for i in range(n)]
    settings = RouterSettings(routing_strategy="prefix-affinity", fallbacks={})
    return Router(deps, settings, clock=FakeClock(0.0),
                  rng=random.Random(7))


def test_prefix_affinity_sticky_and_spread():
    """Identical prompt prefixes always land on the same deployment
    (prefix-cache locality); distinct prefixes spread across replicas."""
    r = _affinity_router()
    picks = set()
    for _ in range(10):
        t = r.acquire("aff", 10, affinity_key="system: you are a bot\x1eQ1")
        picks.add(t.state.dep.model_id)
        r.complete(t, 10)
    assert len(picks) == 1
    spread = set()
    for i in range(40):
        t = r.acquire("aff", 10, affinity_key=f"prefix-{i}")
        spread.add(t.state.dep.model_id)
        r.complete(t, 10)
    assert len(spread) == 3   # rendezvous hashing uses every replica


def test_prefix_affinity_failover_and_restore():
    """When the preferred replica is down, the SAME alternate is chosen
    every time (deterministic next-highest score); affinity returns to
    the original replica on recovery."""
    r = _affinity_router()
    key = "shared-context"
    t = r.acquire("aff", 10, affinity_key=key)
    preferred = t.state
    r.complete(t, 10)
    preferred.healthy = False
    alts = set()
    for _ in range(8):
        t = r.acquire("aff", 10, affinity_key=key)
        alts.add(t.state.dep.model_id)
        r.complete(t, 10)
    assert len(alts) == 1 and preferred.dep.model_id not in alts
    preferred.healthy = True
    t = r.acquire("aff", 10, affinity_key=key)
    assert t.state is preferred
    r.complete(t, 10)


def test_prefix_affinity_without_key_falls_back_to_shuffle():
    r = _affinity_router()
    picks = set()
    for _ in range(40):
        t = r.acquire("aff", 10)
        picks.add(t.state.dep.model_id)
        r.complete(t, 10)
    assert len(picks) == 3


def test_property_acquire_never_starves_while_served():
    """Hypothesis: under ANY sequence of drain/cooldown/heal operations,
    acquire('lb') succeeds as long as at least one lb replica is healthy
    and rate-limit-admissible — drain and cooldown alone can never
    produce a client-visible 'no deployment' (the last-resort
    invariant)."""
    from hypothesis import given, settings as hsettings, strategies as st

    ops = st.lists(
        st.tuples(
            st.sampled_from(["drain", "undrain", "cool", "heal", "sick",
                             "acquire"]),
            st.sampled_from(["gpu0/r0", "gpu1/r1"])),
        min_size=1, max_size=40)

    @hsettings(max_examples=200, deadline=None)
    @given(ops)
    def run(seq):
        clk = FakeClock(0.0)
        r = make_router(clock=clk, fallbacks={}, allowed_fails=1,
                        cooldown_time=30.0)
        for op, dep in seq:
            if op == "drain":
                r.set_draining(dep, True)
            elif op == "undrain":
                r.set_draining(dep, False)
            elif op == "cool":
                s = r.state_for_id(dep)
                s.cooldown_until = clk() + 30.0
            elif op == "heal":
                r.set_healthy(dep, True)
            elif op == "sick":
                r.set_healthy(dep, False)
            else:
                healthy = [s for s in r.alias_states("lb")
                           if s.healthy and s.limiter.would_admit(1)]
                if healthy:
                    t = r.acquire("lb", 1)     # must NOT raise
                    r.complete(t, actual_tokens=0)  # refund: rpm stays open
                else:
                    with pytest.raises(RouterRateLimit):
                        r.acquire("lb", 1)

    run()
