"""GEMM autotuner plumbing (CPU-safe paths; algo racing itself is a GPU
concern covered by tests/test_ops_gpu.py)."""

import torch

from resilient_llm_amd.ops import autotune


def test_disabled_by_default_and_bypasses_cache():
    assert autotune._DISABLED, "tuner must be opt-in (RLLI_LT=1)"
    x = torch.randn(4, 8, dtype=torch.float32)
    w = torch.randn(6, 8, dtype=torch.float32)
    y = autotune.tuned_linear(x, w)
    torch.testing.assert_close(y, torch.nn.functional.linear(x, w))
    assert (4, 6, 8) not in autotune.tuned_shapes()


def test_propagate_copies_per_weight_decision():
    autotune._cache[(64, 100, 200)] = 1234
    autotune._cache[(8, 100, 200)] = None
    autotune._cache[(8, 999, 200)] = None
    try:
        autotune.propagate([(64, 100, 200)])
        assert autotune._cache[(8, 100, 200)] == 1234
        assert autotune._cache[(8, 999, 200)] is None
    finally:
        for k in [(64, 100, 200), (8, 100, 200), (8, 999, 200)]:
            autotune._cache.pop(k, None)
