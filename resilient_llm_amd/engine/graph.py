"""hipGraph capture of the decode step (per batch-size bucket).

A Llama-8B decode step is ~300 kernel launches; eager host-side launch
overhead (~3-4 us each) would put a millisecond-plus floor under every
step.  Replaying a captured hipGraph costs ~10-16 us total (guide:
MI355X_MICROARCH.md graph-replay-floor), so the decode hot loop runs
launch-bound-free.  One graph per batch-size bucket over shared static
buffers; smaller batches pad into the next bucket with a scratch KV
block so padded lanes write garbage nowhere real.
"""

from __future__ import annotations

import torch

from ..ops import autotune as lt_autotune

BUCKETS = [1, 2, 4, 8, 16, 24, 32, 40, 48, 56, 64, 96, 128, 160, 192, 224, 256]


class DecodeGraphRunner:
    def __init__(self, engine, max_blocks_per_seq: int = 64) -> None:
        self.engine = engine
        self.model = engine.model
        self.kv = engine.kv
        self.device = self.model.device
        self.max_batch = engine.max_batch_size
        self.buckets = sorted({b for b in BUCKETS if b <= self.max_batch}
                              | {self.max_batch})
        self.bt_width = max_blocks_per_seq
        # one scratch block absorbs padded lanes' KV appends
        self.scratch_block = self.kv.allocate(1)[0]

        B = self.buckets[-1]
        dev = self.device
        self.in_ids = torch.zeros(B, dtype=torch.int32, device=dev)
        self.positions = torch.zeros(B, dtype=torch.int32, device=dev)
        self.slots = torch.full((B,), self.scratch_block * self.kv.block_size,
                                dtype=torch.int32, device=dev)
        self.block_tables = torch.full((B, self.bt_width), self.scratch_block,
                                       dtype=torch.int32, device=dev)
        self.seq_lens = torch.ones(B, dtype=torch.int32, device=dev)
        self.graphs: dict[int, torch.cuda.CUDAGraph] = {}
        self.logits: dict[int, torch.Tensor] = {}
        self._pool = None
        self._capture_all()

    def _run_eager(self, b: int) -> torch.Tensor:
        return self.model.forward_decode(
            self.in_ids[:b], self.positions[:b], self.kv, self.slots[:b],
            self.block_tables[:b], self.seq_lens[:b])

    def _capture_all(self) -> None:
        # warmup on a side stream (required before capture)
        s = torch.cuda.Stream(device=self.device)
        s.wait_stream(torch.cuda.current_stream(self.device))
        with torch.cuda.stream(s):
            for _ in range(2):
                self._run_eager(self.buckets[-1])
            # visit every bucket eagerly so the measured-GEMM autotuner
            # (ops/autotune.py) pins per-shape algos BEFORE capture —
            # tuning needs syncs and cannot run mid-capture
            for b in self.buckets:
                self._run_eager(b)
            if lt_autotune.enabled():
                # re-decide per shape against the REAL objective: a
                # full decode step at the largest bucket, then copy the
                # per-weight decisions to the other buckets
                bmax = self.buckets[-1]
                keys = [k for k in lt_autotune.tuned_shapes() if k[0] == bmax]
                lt_autotune.tune_in_context(lambda: self._run_eager(bmax),
                                            keys=keys)
                lt_autotune.propagate(keys)
        torch.cuda.current_stream(self.device).wait_stream(s)
        torch.cuda.synchronize(self.device)

        # capture largest-first so the shared pool is sized once
        for b in sorted(self.buckets, reverse=True):
            g = torch.cuda.CUDAGraph()
            if self._pool is None:
                with torch.cuda.graph(g):
                    self.logits[b] = self._run_eager(b)
                self._pool = g.pool()
            else:
                with torch.cuda.graph(g, pool=self._pool):
                    self.logits[b] = self._run_eager(b)
            self.graphs[b] = g
        torch.cuda.synchronize(self.device)

    def _bucket(self, b: int) -> int:
        for x in self.buckets:
            if x >= b:
                return x
        return self.buckets[-1]

    def run(self, input_ids: torch.Tensor, positions: torch.Tensor,
            slots: torch.Tensor, block_tables: torch.Tensor,
            seq_lens: torch.Tensor) -> torch.Tensor:
        B = input_ids.shape[0]
        W = block_tables.shape[1]
        if B > self.max_batch or W > self.bt_width:
            # out-of-envelope (very long seq): eager fallback
            return self.model.forward_decode(input_ids, positions, self.kv,
                                             slots, block_tables, seq_lens)
        nb = self._bucket(B)
        self.in_ids[:B].copy_(input_ids)
        self.positions[:B].copy_(positions)
        self.slots[:B].copy_(slots)
        self.block_tables[:B, :W].copy_(block_tables)
        self.seq_lens[:B].copy_(seq_lens)
        if nb > B:   # neutralize padded lanes
            self.in_ids[B:nb].zero_()
            self.positions[B:nb].zero_()
            self.slots[B:nb].fill_(self.scratch_block * self.kv.block_size)
            self.seq_lens[B:nb].fill_(1)
            self.block_tables[B:nb].fill_(self.scratch_block)
        self.graphs[nb].replay()
        return self.logits[nb][:B]


def install_graph_runner(engine, max_blocks_per_seq: int = 64) -> None:
    engine.graph_runner = DecodeGraphRunner(engine, max_blocks_per_seq)
