#!/usr/bin/env python3
"""Within-process A/B: skinny_linear vs hipBLASLt on the decode shapes."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import resilient_llm_amd.ops as ops

SHAPES = [("qkv", 64, 5120, 4096), ("o", 64, 4096, 4096),
          ("gate_up", 64, 28672, 4096), ("down", 64, 4096, 14336),
          ("lm_head", 64, 128256, 4096)]

def timeit(fn, iters=50):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.monotonic() - t0) / iters * 1e6

def probe():
    ops.load_extension(required=True)
    print("— pure nt-stream ceiling at skinny geometry —")
    for name, M, N, K in SHAPES:
        w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
        gb = N * K * 2 / 1e9
        for splitk in (1, 2, 4, 8):
            if (N // 64) * splitk > 2048 and splitk > 1:
                continue
            t = timeit(lambda: torch.ops.rlli.stream_probe(w, splitk))
            print(f"  {name:8s} splitk={splitk}: {t:7.1f} us "
                  f"({gb/t*1e6/1e3:5.2f} TB/s)")

def main():
    probe()
    tot_s = tot_b = 0
    for name, M, N, K in SHAPES:
        x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
        w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
        gb = N * K * 2 / 1e9
        t_b = timeit(lambda: torch.nn.functional.linear(x, w))
        t_s = timeit(lambda: ops.skinny_linear(x, w))
        tot_s += t_s; tot_b += t_b
        print(f"{name:8s} M{M} N{N} K{K}: blaslt {t_b:7.1f} us ({gb/t_b*1e6/1e3:5.2f} TB/s)"
              f"  skinny {t_s:7.1f} us ({gb/t_s*1e6/1e3:5.2f} TB/s)  x{t_b/t_s:.2f}")
    print(f"TOTAL: blaslt {tot_b:.0f} us, skinny {tot_s:.0f} us, x{tot_b/tot_s:.2f}")

if __name__ == "__main__":
    main()
