"""Worker-process entry point: one engine on one GPU, served over a unix
socket.  Spawned by workers/gpu.py with HIP_VISIBLE_DEVICES pinned."""

from __future__ import annotations

import argparse
import asyncio
import os
import sys


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--device-label", required=True)   # e.g. gpu:3
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--socket", required=True)
    ap.add_argument("--kv-gb", type=float, default=24.0)
    ap.add_argument("--max-batch", type=int, default=64)
    ap.add_argument("--graphs", action="store_true")
    ap.add_argument("--num-blocks", type=int, default=None)
    ap.add_argument("--target-step-ms", type=float, default=None,
                    help="AIMD prefill budget: keep mixed steps near "
                         "this bound (decode-cadence SLO)")
    ap.add_argument("--weights", default=None,
                    help="safetensors dir (HF Llama names); random init "
                         "when absent")
    ap.add_argument("--eos-id", type=int, action="append", default=None,
                    help="override EOS token id(s); repeatable")
    ap.add_argument("--spec-lookup", type=int, default=0,
                    help="prompt-lookup speculative decoding depth "
                         "(greedy requests; 0 = off)")
    ap.add_argument("--quant", default=None, choices=(None, "fp8"),
                    help="W8A8-fp8 weight quantization for gate_up/down/"
                         "lm_head (rowwise scaled_mm)")
    ap.add_argument("--kv-dtype", default="bf16", choices=("bf16", "fp8"),
                    help="paged-KV-cache element type: fp8-e4m3 halves "
                         "bytes/token (2x KV capacity per GiB)")
    ap.add_argument("--device", default=None,
                    help="torch device override (tests: cpu)")
    args = ap.parse_args()

    import torch
    if args.device:
        device = args.device
    else:
        assert torch.cuda.is_available(), \
            "gpu_main needs a GPU (HIP_VISIBLE_DEVICES pinned by spawner)"
        device = "cuda:0"   # pinned by HIP_VISIBLE_DEVICES

    if device.startswith("cuda"):
        # the HIP extension is mandatory on GPU — fail loudly before load
        from .. import ops
        ops.load_extension(required=True)

    from ..utils.logging import log_with_timestamp
    from .engine_worker import EngineWorker
    from .rpc import WorkerRpcServer

    async def run() -> None:
        worker = EngineWorker(device=device, model_name=args.model,
                              device_label=args.device_label,
                              kv_gb=args.kv_gb,
                              max_batch_size=args.max_batch,
                              num_blocks=args.num_blocks,
                              use_graphs=args.graphs,
                              weights=args.weights,
                              eos_id=args.eos_id,
                              kv_dtype=args.kv_dtype,
                              quant=args.quant,
                              spec_lookup=args.spec_lookup,
                              target_step_ms=args.target_step_ms)
        log_with_timestamp(
            f"worker {args.device_label} ready: {args.model} on {device}, "
            f"{worker.engine.kv.num_blocks} KV blocks "
            f"({worker.engine.kv.bytes_used() / 2**30:.1f} GiB), "
            f"params {worker.model.param_bytes() / 2**30:.1f} GiB", "green")
        server = WorkerRpcServer(worker, args.socket,
                                 on_kill=lambda: os._exit(7))
        await server.serve_forever()

    try:
        asyncio.run(run())
    except KeyboardInterrupt:
        sys.exit(0)


if __name__ == "__main__":
    main()
