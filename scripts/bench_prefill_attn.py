#!/usr/bin/env python3
"""Microbench: MFMA vs VALU prefill attention at serving shapes."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import resilient_llm_amd.ops as ops

def timeit(fn, iters=30):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.monotonic() - t0) / iters * 1e6

def main():
    n_q, n_kv, D = 32, 8, 128
    for n_seqs, L in ((64, 128), (16, 512), (4, 2048), (1, 8192)):
        T = n_seqs * L
        width = (n_q + 2 * n_kv) * D
        qkv = torch.randn(T, width, dtype=torch.bfloat16, device="cuda")
        cu = torch.arange(0, T + 1, L, dtype=torch.int32, device="cuda")
        q = qkv[:, :n_q * D].reshape(T, n_q, D).contiguous()
        k = qkv[:, n_q * D:(n_q + n_kv) * D].reshape(T, n_kv, D).contiguous()
        v = qkv[:, (n_q + n_kv) * D:].reshape(T, n_kv, D).contiguous()
        scale = 0.088
        t_m = timeit(lambda: ops.prefill_attn_qkv(qkv, cu, scale, n_q, n_kv, D))
        t_v = timeit(lambda: ops.prefill_attn(q, k, v, cu, scale))
        flops = n_seqs * (L * (L + 1) / 2) * D * 2 * 2 * n_q
        print(f"{n_seqs}x{L}: VALU {t_v:8.1f} us ({flops/t_v/1e6:6.1f} TF) "
              f"MFMA {t_m:8.1f} us ({flops/t_m/1e6:6.1f} TF)  x{t_v/t_m:.2f}")

if __name__ == "__main__":
    main()
