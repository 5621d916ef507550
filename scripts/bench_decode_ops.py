"""Isolated decode-shape microbench for the non-GEMM hot kernels.

Times rmsnorm_residual_, silu_mul, decode_attn_rope_qkv and sample at
the headline decode shape (Llama-3-8B, batch 64, seq ~192) both as an
eager back-to-back loop and replayed inside one hipGraph, reporting
us/call and effective TB/s against the op's minimum HBM traffic.

Env knobs (statics in the .so, so A/B variants need fresh processes):
  RLLI_WAVE_NORM=1    wave-per-row rmsnorm variant
  RLLI_ATTN_SPLIT=N   force decode-attn split-K factor

Usage: python scripts/bench_decode_ops.py [--batch 64] [--len 192]
"""
import argparse
import json
import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
from resilient_llm_amd import ops  # noqa: E402


def time_op(fn, iters=200, graph=True):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    s, e = torch.cuda.Event(True), torch.cuda.Event(True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    eager_us = s.elapsed_time(e) * 1e3 / iters
    graph_us = None
    if graph:
        g = torch.cuda.CUDAGraph()
        st = torch.cuda.Stream()
        st.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(st):
            fn()
        torch.cuda.current_stream().wait_stream(st)
        with torch.cuda.graph(g):
            for _ in range(20):
                fn()
        torch.cuda.synchronize()
        s.record()
        for _ in range(iters // 20):
            g.replay()
        e.record()
        torch.cuda.synchronize()
        graph_us = s.elapsed_time(e) * 1e3 / (20 * (iters // 20))
    return eager_us, graph_us


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--len", type=int, default=192, dest="seqlen")
    ap.add_argument("--iters", type=int, default=200)
    ap.add_argument("--only", default="", help="run just one op (attn/norm/silu/sample)")
    ap.add_argument("--kv-fp8", action="store_true",
                    help="fp8-e4m3 KV cache for the attention op")
    args = ap.parse_args()

    dev = "cuda:0"
    torch.manual_seed(0)
    B, L = args.batch, args.seqlen
    dim, inter, vocab = 4096, 14336, 128256
    n_q, n_kv, D = 32, 8, 128
    bs = 16                                   # engine KV block size
    env = {k: os.environ.get(k) for k in ("RLLI_WAVE_NORM", "RLLI_ATTN_SPLIT")}
    print(f"# batch={B} len={L} env={env}", flush=True)

    results = {}

    def report(name, traffic_mb, eager_us, graph_us):
        gu = graph_us if graph_us is not None else eager_us
        results[name] = round(gu, 2)
        print(f"{name:26s} eager {eager_us:7.2f} us  graph {gu:7.2f} us  "
              f"{traffic_mb / gu * 1e-3:5.2f} TB/s eff", flush=True)

    only = args.only
    # --- rmsnorm_residual_ (2/layer + final = 66/step) ---
    if only in ("", "norm"):
        x = torch.randn(B, dim, device=dev, dtype=torch.bfloat16)
        res = torch.randn(B, dim, device=dev, dtype=torch.bfloat16)
        w = torch.randn(dim, device=dev, dtype=torch.bfloat16)
        e, g = time_op(lambda: ops.rmsnorm_residual_(x, res, w), args.iters)
        report("rmsnorm_residual", B * dim * 2 * 3 / 1e6, e, g)

    # --- silu_mul ---
    if only in ("", "silu"):
        gu_t = torch.randn(B, 2 * inter, device=dev, dtype=torch.bfloat16)
        e, g = time_op(lambda: ops.silu_mul(gu_t), args.iters)
        report("silu_mul", B * inter * 2 * 3 / 1e6, e, g)

    # --- decode_attn_rope_qkv (fused rope+append+attn) ---
    if only in ("", "attn"):
        n_blocks = (L + bs - 1) // bs + 1
        total_blocks = B * n_blocks + 8
        kc = torch.randn(total_blocks, n_kv, bs, D, device=dev,
                         dtype=torch.bfloat16)
        vc = torch.randn_like(kc)
        if args.kv_fp8:
            kc = kc.to(torch.float8_e4m3fn)
            vc = vc.to(torch.float8_e4m3fn)
        bt = torch.arange(B * n_blocks, device=dev,
                          dtype=torch.int32).reshape(B, n_blocks)
        seq_lens = torch.full((B,), L, device=dev, dtype=torch.int32)
        positions = seq_lens - 1
        slots = bt.gather(1, ((L - 1) // bs * torch.ones(B, 1, device=dev,
                              dtype=torch.int64))).squeeze(1) * bs + (L - 1) % bs
        slots = slots.int()
        qkv = torch.randn(B, (n_q + 2 * n_kv) * D, device=dev,
                          dtype=torch.bfloat16)
        inv = 1.0 / (10000.0 ** (torch.arange(0, D // 2, device=dev) / (D // 2)))
        ang = torch.outer(torch.arange(4096, device=dev).float(), inv)
        cos_sin = torch.cat([ang.cos(), ang.sin()], -1).contiguous()
        scale = D ** -0.5
        e, g = time_op(lambda: ops.decode_attn_rope_qkv(
            qkv, positions, cos_sin, kc, vc, slots, bt, seq_lens, scale, n_q),
            args.iters)
        ebytes = 1 if args.kv_fp8 else 2
        report("decode_attn_rope_qkv", B * L * n_kv * D * ebytes * 2 / 1e6, e, g)

    # --- sample ---
    if only in ("", "sample"):
        logits = torch.randn(B, vocab, device=dev, dtype=torch.bfloat16)
        temps = torch.zeros(B, device=dev, dtype=torch.float32)
        seeds = torch.arange(B, device=dev, dtype=torch.int64)
        e, g = time_op(lambda: ops.sample(logits, temps, seeds, 0), args.iters)
        report("sample_greedy", B * vocab * 2 / 1e6, e, g)
        temps2 = torch.full((B,), 0.8, device=dev, dtype=torch.float32)
        e, g = time_op(lambda: ops.sample(logits, temps2, seeds, 0), args.iters)
        report("sample_gumbel", B * vocab * 2 / 1e6, e, g)

    print("JSON " + json.dumps({"env": env, "batch": B, "len": L,
                                "us": results}), flush=True)


if __name__ == "__main__":
    main()
