#!/usr/bin/env python3
"""Microbench: paged decode attention at serving shapes."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import resilient_llm_amd.ops as ops

def main():
    n_q, n_kv, D, bs = 32, 8, 128, 16
    shapes = [(64, 128), (64, 192), (64, 512), (64, 2048),
              # small batch x long context: split-K (flash-decode) engages
              (8, 2048), (8, 8192), (4, 8192), (1, 8192), (1, 32768)]
    for B, L in shapes:
        nb = (L + bs - 1) // bs
        kc = torch.randn(B * nb + 2, n_kv, bs, D, dtype=torch.bfloat16, device="cuda")
        vc = torch.randn_like(kc)
        q = torch.randn(B, n_q, D, dtype=torch.bfloat16, device="cuda")
        bt = torch.arange(B * nb, dtype=torch.int32, device="cuda").reshape(B, nb).contiguous()
        lens = torch.full((B,), L, dtype=torch.int32, device="cuda")
        fn = lambda: ops.decode_attn(q, kc, vc, bt, lens, 0.088)
        for _ in range(10): fn()
        torch.cuda.synchronize()
        t0 = time.monotonic()
        for _ in range(50): fn()
        torch.cuda.synchronize()
        us = (time.monotonic() - t0) / 50 * 1e6
        gb = B * n_kv * L * D * 2 * 2 / 1e9
        print(f"B={B:3d} L={L:6d}: {us:7.1f} us, "
              f"{gb/us*1e6/1e3:5.2f} TB/s effective KV stream")

if __name__ == "__main__":
    main()
