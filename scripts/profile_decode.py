#!/usr/bin/env python3
"""Engine-direct profiling harness (no HTTP): run prefill + decode steps
of the flagship model for rocprofv3 kernel-trace/PMC runs.

Usage: python scripts/profile_decode.py [--model llama-3-8b] [--batch 64]
       [--prompt-tokens 128] [--decode-steps 64] [--graphs]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--prompt-tokens", type=int, default=128)
    ap.add_argument("--decode-steps", type=int, default=64)
    ap.add_argument("--graphs", action="store_true")
    ap.add_argument("--repeat", type=int, default=1)
    args = ap.parse_args()

    import torch
    from resilient_llm_amd.engine import LLMEngine, PagedKVCache, SamplingParams
    from resilient_llm_amd.models import LlamaForCausalLM, get_config

    cfg = get_config(args.model)
    model = LlamaForCausalLM(cfg, device="cuda:0", dtype=torch.bfloat16, seed=0)
    blocks_per_seq = -(-(args.prompt_tokens + args.decode_steps + 8) // 16)
    kv = PagedKVCache.for_model(cfg, args.batch * (blocks_per_seq + 1) + 8,
                                device="cuda:0")
    engine = LLMEngine(model, kv, max_batch_size=args.batch,
                       max_prefill_tokens=args.batch * args.prompt_tokens)
    if args.graphs:
        from resilient_llm_amd.engine.graph import install_graph_runner
        install_graph_runner(engine)

    import random
    rng = random.Random(7)
    for rep in range(args.repeat):
        for i in range(args.batch):
            # DISTINCT prompts: identical prompts hit the prefix cache and
            # skip almost all prefill compute (r01 profiled that by
            # mistake and under-read prefill cost ~20x)
            prompt = [rng.randrange(10, 28000)
                      for _ in range(args.prompt_tokens)]
            engine.add_request(f"p{rep}-{i}", prompt,
                               SamplingParams(max_tokens=args.decode_steps))
        torch.cuda.synchronize()
        t0 = time.monotonic()
        engine.step()                      # prefill
        torch.cuda.synchronize()
        t1 = time.monotonic()
        n = 0
        while engine.has_work():
            engine.step()
            n += 1
        torch.cuda.synchronize()
        t2 = time.monotonic()
        print(f"rep {rep}: prefill {args.batch}x{args.prompt_tokens} tok "
              f"= {(t1 - t0) * 1000:.1f} ms "
              f"({args.batch * args.prompt_tokens / (t1 - t0):.0f} tok/s); "
              f"{n} decode steps = {(t2 - t1) * 1000:.1f} ms "
              f"({(t2 - t1) / n * 1000:.2f} ms/step, "
              f"{args.batch * n / (t2 - t1):.0f} tok/s)", flush=True)


if __name__ == "__main__":
    main()
