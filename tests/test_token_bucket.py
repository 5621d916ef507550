from resilient_llm_amd.router.token_bucket import MinuteWindowLimiter


class FakeClock:
    def __init__(self, t=0.0):
        self.t = t

    def __call__(self):
        return self.t


def test_rpm_exact_window_semantics():
    """The reference contract: rpm=3 admits exactly 3 of 5 simultaneous
    requests (reference README.md:255-266)."""
    clk = FakeClock(10.0)
    lim = MinuteWindowLimiter(rpm=3, clock=clk)
    results = [lim.try_acquire() for _ in range(5)]
    assert results == [True, True, True, False, False]


def test_window_resets_each_minute():
    clk = FakeClock(10.0)
    lim = MinuteWindowLimiter(rpm=2, clock=clk)
    assert lim.try_acquire() and lim.try_acquire()
    assert not lim.try_acquire()
    clk.t = 61.0  # next minute window
    assert lim.try_acquire()


def test_tpm_pre_call_check():
    clk = FakeClock(0.0)
    lim = MinuteWindowLimiter(tpm=1000, clock=clk)
    assert lim.would_admit(900)
    assert lim.try_acquire(900)
    assert not lim.would_admit(200)
    assert not lim.try_acquire(200)
    assert lim.try_acquire(100)


def test_reconcile_replaces_estimate():
    clk = FakeClock(0.0)
    lim = MinuteWindowLimiter(tpm=1000, clock=clk)
    assert lim.try_acquire(800)
    lim.reconcile(estimated=800, actual=100)
    snap = lim.snapshot()
    assert snap.tpm_used == 100
    assert lim.try_acquire(800)


def test_release_refunds_admission():
    clk = FakeClock(0.0)
    lim = MinuteWindowLimiter(rpm=1, tpm=500, clock=clk)
    assert lim.try_acquire(500)
    lim.release(500)
    assert lim.try_acquire(500)


def test_unlimited_when_none():
    lim = MinuteWindowLimiter()
    assert all(lim.try_acquire(10**6) for _ in range(100))


def test_snapshot_remaining():
    clk = FakeClock(0.0)
    lim = MinuteWindowLimiter(rpm=5, tpm=100, clock=clk)
    lim.try_acquire(30)
    snap = lim.snapshot()
    assert snap.rpm_remaining == 4
    assert snap.tpm_remaining == 70
