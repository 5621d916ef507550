#!/usr/bin/env python3
"""Microbench: paged decode attention at serving shapes."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import resilient_llm_amd.ops as ops

def main():
    B, n_q, n_kv, D, bs = 64, 32, 8, 128, 16
    for L in (128, 192, 512, 2048):
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
        print(f"L={L}: {us:7.1f} us, {gb/us*1e6/1e3:5.2f} TB/s effective KV stream")

if __name__ == "__main__":
    main()
