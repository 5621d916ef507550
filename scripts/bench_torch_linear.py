#!/usr/bin/env python3
"""Time torch F.linear (hipBLASLt via torch's heuristic pick) on the
decode GEMM shapes — the comparison row for scripts/probe_hipblaslt.cpp."""
import sys
import time

import torch
import torch.nn.functional as F

SHAPES = [("qkv   ", 64, 6144, 4096), ("o     ", 64, 4096, 4096),
          ("gateup", 64, 28672, 4096), ("down  ", 64, 4096, 14336),
          ("lmhead", 64, 128256, 4096)]


def main():
    for name, M, N, K in SHAPES:
        x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
        w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
        fn = lambda: F.linear(x, w)
        for _ in range(10):
            fn()
        torch.cuda.synchronize()
        t0 = time.monotonic()
        for _ in range(100):
            fn()
        torch.cuda.synchronize()
        us = (time.monotonic() - t0) / 100 * 1e6
        gb = N * K * 2 / 1e9
        print(f"torch {name} M={M} N={N} K={K}: {us:8.2f} us "
              f"({gb/us*1e3:5.2f} TB/s weight-stream)")


if __name__ == "__main__":
    main()
