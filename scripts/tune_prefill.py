#!/usr/bin/env python3
"""Prefill GEMM algo experiment (r02): torch's hipBLASLt heuristic sends
ALL four M=8192 layer GEMMs to one stream-K kernel (596 us avg, 76.6% of
prefill GPU time, ~1.68 PF/s — gpurun_out/r02_prefill_kernel_stats.csv).
Race the library's candidate list per shape, then validate the winners
against the FULL prefill forward (ms-scale objective, unlike the decode
tuning where tens-of-us deltas drowned in noise).
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("RLLI_LT", "1")   # enable the autotune cache

import torch
import torch.nn.functional as F

from resilient_llm_amd import ops
from resilient_llm_amd.ops import autotune


def timeit(fn, reps=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.monotonic() - t0) / reps * 1e6


def main():
    ops.load_extension(required=True)
    dev = "cuda:0"
    M = int(sys.argv[1]) if len(sys.argv) > 1 else 8192
    shapes = [("qkv", 6144, 4096), ("o", 4096, 4096),
              ("gate_up", 28672, 4096), ("down", 4096, 14336)]
    torch.manual_seed(0)
    flops = {}
    for name, N, K in shapes:
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16) * 0.1
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
        t_torch = timeit(lambda: F.linear(x, w))
        idxs = torch.ops.rlli.lt_heuristics(x, w, 24).tolist()
        best_t, best_i = t_torch, None
        for i in idxs:
            try:
                t = timeit(lambda: torch.ops.rlli.lt_linear(x, w, i), reps=10)
            except Exception:
                continue
            if t < best_t:
                best_t, best_i = t, i
        fl = 2.0 * M * N * K
        flops[name] = fl
        print(f"{name:8s} M={M}: torch={t_torch:7.1f}us "
              f"({fl / t_torch / 1e9:.2f} PF/s)  best=algo{best_i} "
              f"{best_t:7.1f}us ({fl / best_t / 1e9:.2f} PF/s)  "
              f"{t_torch / best_t:.2f}x", flush=True)
        autotune._cache[(M, N, K)] = best_i

    # ---- e2e prefill forward: torch picks vs pinned winners ----
    from resilient_llm_amd.engine import PagedKVCache
    from resilient_llm_amd.models import LlamaForCausalLM, get_config
    import resilient_llm_amd.models.llama as llama_mod

    cfg = get_config("llama-3-8b")
    model = LlamaForCausalLM(cfg, device=dev, dtype=torch.bfloat16, seed=0)
    kv = PagedKVCache.for_model(cfg, 600, device=dev)
    B, L = M // 128, 128
    ids = torch.randint(10, 28000, (M,), dtype=torch.int32, device=dev)
    pos = torch.arange(L, dtype=torch.int32, device=dev).repeat(B)
    slots = torch.arange(M, dtype=torch.int32, device=dev)
    cu = torch.arange(0, M + 1, L, dtype=torch.int32, device=dev)

    def prefill():
        return model.forward_prefill(ids, pos, kv, slots, cu)

    pinned = dict(autotune._cache)
    autotune._cache.clear()
    saved_rows = llama_mod._LT_ROWS
    llama_mod._LT_ROWS = frozenset()      # force pure torch picks
    t_base = timeit(prefill, reps=10, warmup=3)
    llama_mod._LT_ROWS = frozenset({M})   # route M through the cache
    autotune._cache.update(pinned)
    t_tuned = timeit(prefill, reps=10, warmup=3)
    total_fl = sum(flops.values()) / 4096 * (M / 8192) * 0   # unused
    fl_fwd = 2.0 * 8.03e9 * M
    print(f"e2e prefill M={M}: torch={t_base / 1e3:.2f}ms "
          f"({fl_fwd / t_base / 1e9:.2f} PF/s) vs tuned={t_tuned / 1e3:.2f}ms "
          f"({fl_fwd / t_tuned / 1e9:.2f} PF/s)  {t_base / t_tuned:.3f}x",
          flush=True)
    llama_mod._LT_ROWS = saved_rows


if __name__ == "__main__":
    main()
