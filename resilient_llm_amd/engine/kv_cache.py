"""Paged KV cache sized for 288 GB of HBM3E per GPU.

Layout [n_layers, num_blocks, n_kv_heads, block_size, head_dim] bf16 for
K and V — each (block, kv-head) row is a contiguous
[block_size, head_dim] tile, which is exactly what one lane-group of the
decode kernel streams (ops/csrc/decode_attn.hip).  Block size 16 tokens.

The allocator is a plain free list: continuous batching allocates a
block when a sequence crosses a 16-token boundary and frees the whole
list when it finishes; there is no copy-on-write/prefix sharing in v1.
"""

from __future__ import annotations

import torch


class OutOfBlocks(RuntimeError):
    """KV budget exhausted — the worker surfaces this as Throttled (X13)."""


class PagedKVCache:
    def __init__(self, n_layers: int, num_blocks: int, n_kv_heads: int,
                 block_size: int, head_dim: int, device="cpu",
                 dtype: torch.dtype = torch.bfloat16) -> None:
        self.n_layers = n_layers
        self.num_blocks = num_blocks
        self.n_kv_heads = n_kv_heads
        self.block_size = block_size
        self.head_dim = head_dim
        shape = (n_layers, num_blocks, n_kv_heads, block_size, head_dim)
        self.k = torch.zeros(shape, dtype=dtype, device=device)
        self.v = torch.zeros(shape, dtype=dtype, device=device)
        self._free: list[int] = list(range(num_blocks - 1, -1, -1))

    @classmethod
    def for_model(cls, config, num_blocks: int, device="cpu",
                  block_size: int = 16, tp_world: int = 1) -> "PagedKVCache":
        return cls(config.n_layers, num_blocks,
                   config.n_kv_heads // tp_world, block_size,
                   config.head_dim, device=device)

    def layer(self, i: int) -> tuple[torch.Tensor, torch.Tensor]:
        return self.k[i], self.v[i]

    @property
    def free_blocks(self) -> int:
        return len(self._free)

    def allocate(self, n: int) -> list[int]:
        if n > len(self._free):
            raise OutOfBlocks(f"need {n} KV blocks, have {len(self._free)}")
        out = [self._free.pop() for _ in range(n)]
        return out

    def free(self, blocks: list[int]) -> None:
        self._free.extend(blocks)

    def bytes_used(self) -> int:
        return self.k.numel() * self.k.element_size() * 2
