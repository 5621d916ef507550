"""hipGraph capture of the WHOLE steady decode step (per batch-size
bucket): slot computation, forward, fused sampling, and the state
advance (in_ids <- sampled tokens, positions += 1, seeds += GOLDEN) all
replay as ONE graph.

A Llama-8B decode step is ~300 kernel launches; the r01 design graphed
only the forward and left ~10 host-dispatched ops per step outside
(slot gather chain, sampler, 5 H2D buffer copies).  The kernel trace
showed the GPU going idle ~100-200 us per step waiting for that host
work (gap before div_floor_kernel in profiles/r02 step trace), and the
engine spent ~450 us of host python per step feeding it.  With the
state advance captured in-graph, a steady step is ONE replay + one
small D2H token copy: the host loop drops to bookkeeping only, and a
batch that stays unchanged never copies anything host->device.

Seed semantics: the sampling kernel premixes `seeds[row] + GOLDEN*step`
(sampling.hip).  The graph passes step=0 and instead advances the seed
buffer by GOLDEN (signed-wrapped) each replay — bit-identical to the
eager path's increasing step offset, and consistent with the
per-(request, absolute output index) seed contract (engine._seq_seed).

Batches with torch-side sampling rows (top-p, penalties) still replay
the graph: the captured logits buffer is returned, the host recomputes
the affected rows, and override_tokens() corrects the fed tokens before
the next replay (the in-graph KV append uses the PREVIOUS token, so a
post-replay correction is race-free).  Only batches whose block tables
exceed the capture envelope (very long sequences) decode eager.
"""

from __future__ import annotations

import torch

from .. import ops
from ..ops import autotune as lt_autotune

BUCKETS = [1, 2, 4, 8, 16, 24, 32, 40, 48, 56, 64, 96, 128, 160, 192, 224, 256]

# 64-bit golden-ratio step premix of sampling.hip, wrapped to signed
# int64 so torch arithmetic reproduces the kernel's mod-2^64 add
GOLDEN_SIGNED = 0x9E3779B97F4A7C15 - (1 << 64)


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
        # one scratch block absorbs padded lanes' KV appends; their
        # block-table rows are ALL scratch, so the in-graph slot
        # computation (clamped block index + pos % block_size) keeps
        # them inside the scratch block no matter how long the batch
        # stays resident
        self.scratch_block = self.kv.allocate(1)[0]

        B = self.buckets[-1]
        dev = self.device
        self.in_ids = torch.zeros(B, dtype=torch.int32, device=dev)
        self.positions = torch.zeros(B, dtype=torch.int32, device=dev)
        self.block_tables = torch.full((B, self.bt_width), self.scratch_block,
                                       dtype=torch.int32, device=dev)
        self.temps = torch.zeros(B, dtype=torch.float32, device=dev)
        self.seeds = torch.zeros(B, dtype=torch.int64, device=dev)
        self.graphs: dict[int, torch.cuda.CUDAGraph] = {}
        self.tokens: dict[int, torch.Tensor] = {}
        self.logits: dict[int, torch.Tensor] = {}
        self._pool = None
        self._nb = 0                      # bucket currently loaded
        self._live = 0                    # live rows of the loaded batch
        self._capture_all()

    # ---- the captured step ----
    def _steady_body(self, b: int) -> torch.Tensor:
        bsz = self.kv.block_size
        pos = self.positions[:b]
        bi = torch.clamp((pos // bsz).long(), max=self.bt_width - 1)
        slots = (self.block_tables[:b].gather(1, bi.unsqueeze(1)).squeeze(1)
                 * bsz + pos % bsz)
        seq_lens = pos + 1
        logits = self.model.forward_decode(self.in_ids[:b], pos, self.kv,
                                           slots, self.block_tables[:b],
                                           seq_lens)
        toks = ops.sample(logits, self.temps[:b], self.seeds[:b], 0)
        self.in_ids[:b].copy_(toks)
        self.positions[:b].add_(1)
        self.seeds[:b].add_(GOLDEN_SIGNED)
        return logits, toks

    def _capture_all(self) -> None:
        # warmup on a side stream (required before capture)
        s = torch.cuda.Stream(device=self.device)
        s.wait_stream(torch.cuda.current_stream(self.device))
        with torch.cuda.stream(s):
            for _ in range(2):
                self._steady_body(self.buckets[-1])
            # visit every bucket eagerly so the measured-GEMM autotuner
            # (ops/autotune.py) pins per-shape algos BEFORE capture —
            # tuning needs syncs and cannot run mid-capture
            for b in self.buckets:
                self._steady_body(b)
            if lt_autotune.enabled():
                bmax = self.buckets[-1]
                keys = [k for k in lt_autotune.tuned_shapes() if k[0] == bmax]
                lt_autotune.tune_in_context(lambda: self._steady_body(bmax),
                                            keys=keys)
                lt_autotune.propagate(keys)
        torch.cuda.current_stream(self.device).wait_stream(s)
        torch.cuda.synchronize(self.device)

        # capture largest-first so the shared pool is sized once
        for b in sorted(self.buckets, reverse=True):
            g = torch.cuda.CUDAGraph()
            if self._pool is None:
                with torch.cuda.graph(g):
                    self.logits[b], self.tokens[b] = self._steady_body(b)
                self._pool = g.pool()
            else:
                with torch.cuda.graph(g, pool=self._pool):
                    self.logits[b], self.tokens[b] = self._steady_body(b)
            self.graphs[b] = g
        torch.cuda.synchronize(self.device)
        # the warmup/captures advanced the state buffers; a load() must
        # precede the first real step()
        self._nb = 0

    def _bucket(self, b: int) -> int:
        for x in self.buckets:
            if x >= b:
                return x
        return self.buckets[-1]

    def steady_ok(self, batch: int, bt_width: int) -> bool:
        return batch <= self.max_batch and bt_width <= self.bt_width

    def load(self, input_ids: torch.Tensor, positions: torch.Tensor,
             block_tables: torch.Tensor, temps: torch.Tensor,
             seeds: torch.Tensor) -> None:
        """Push a (re)built batch into the static buffers.  Called only
        when the batch composition changed — a steady batch replays with
        zero host->device traffic."""
        B = input_ids.shape[0]
        W = block_tables.shape[1]
        nb = self._bucket(B)
        self.in_ids[:B].copy_(input_ids)
        self.positions[:B].copy_(positions)
        self.block_tables[:B, :W].copy_(block_tables)
        self.temps[:B].copy_(temps)
        self.seeds[:B].copy_(seeds)
        if W < self.bt_width:          # stale tail from an earlier batch
            self.block_tables[:B, W:].fill_(self.scratch_block)
        if nb > B:                     # neutralize padded lanes
            self.in_ids[B:nb].zero_()
            self.positions[B:nb].zero_()
            self.block_tables[B:nb].fill_(self.scratch_block)
            self.temps[B:nb].zero_()
            self.seeds[B:nb].zero_()
        self._nb = nb
        self._live = B

    def step(self) -> tuple[torch.Tensor, torch.Tensor]:
        """Replay one whole decode step; returns (tokens, logits)
        device views, valid until the next replay."""
        self.graphs[self._nb].replay()
        return (self.tokens[self._nb][:self._live],
                self.logits[self._nb][:self._live])

    def override_tokens(self, tok_dev: torch.Tensor) -> None:
        """Replace the in-graph sampled tokens before the next replay
        (torch-side sampling rows: top-p / penalties recompute on the
        host from the returned logits and correct the fed tokens)."""
        self.in_ids[:tok_dev.shape[0]].copy_(tok_dev)


def install_graph_runner(engine, max_blocks_per_seq: int = 64) -> None:
    engine.graph_runner = DecodeGraphRunner(engine, max_blocks_per_seq)
