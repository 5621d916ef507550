"""Paged KV cache sized for 288 GB of HBM3E per GPU.

Layout [n_layers, num_blocks, n_kv_heads, block_size, head_dim] bf16 for
K and V — each (block, kv-head) row is a contiguous
[block_size, head_dim] tile, which is exactly what one lane-group of the
decode kernel streams (ops/csrc/decode_attn.hip).  Block size 16 tokens.

Allocation is refcounted with PREFIX CACHING: full prompt blocks are
content-addressed (sha1 chain over token ids), so a request whose prompt
shares a cached prefix acquires those blocks instead of recomputing them
— it simply enters the chunked-prefill path at the first uncached token.
Blocks whose refcount drops to zero stay indexed ("cached-free") and are
evicted FIFO only when a fresh allocation needs them.
"""

from __future__ import annotations

import collections
import hashlib
import struct

import torch


class OutOfBlocks(RuntimeError):
    """KV budget exhausted — the worker surfaces this as Throttled (X13)."""


def block_hash_chain(token_ids, block_size: int) -> list[bytes]:
    """Content keys for each FULL block of a prompt: sha1(parent, tokens)."""
    keys = []
    parent = b"root"
    for i in range(len(token_ids) // block_size):
        blk = token_ids[i * block_size:(i + 1) * block_size]
        h = hashlib.sha1(parent + struct.pack(f"<{block_size}i", *blk)).digest()
        keys.append(h)
        parent = h
    return keys


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
        self._ref = [0] * num_blocks
        # prefix cache state
        self._index: dict[bytes, int] = {}          # key -> block id
        self._block_key: dict[int, bytes] = {}      # block id -> key
        self._cached_free: "collections.OrderedDict[int, None]" =             collections.OrderedDict()               # evictable, still indexed
        self.prefix_hits = 0
        self.prefix_lookups = 0

    @classmethod
    def for_model(cls, config, num_blocks: int, device="cpu",
                  block_size: int = 16, tp_world: int = 1,
                  kv_dtype: torch.dtype = torch.bfloat16) -> "PagedKVCache":
        return cls(config.n_layers, num_blocks,
                   config.n_kv_heads // tp_world, block_size,
                   config.head_dim, device=device, dtype=kv_dtype)

    def layer(self, i: int) -> tuple[torch.Tensor, torch.Tensor]:
        return self.k[i], self.v[i]

    @property
    def free_blocks(self) -> int:
        return len(self._free) + len(self._cached_free)

    def _drop_identity(self, b: int) -> None:
        key = self._block_key.pop(b, None)
        if key is not None and self._index.get(key) == b:
            del self._index[key]

    def allocate(self, n: int) -> list[int]:
        if n > self.free_blocks:
            raise OutOfBlocks(f"need {n} KV blocks, have {self.free_blocks}")
        out = []
        for _ in range(n):
            if self._free:
                b = self._free.pop()
            else:
                b, _ = self._cached_free.popitem(last=False)   # FIFO evict
                self._drop_identity(b)
            self._ref[b] = 1
            out.append(b)
        return out

    def free(self, blocks: list[int]) -> None:
        for b in blocks:
            self._ref[b] -= 1
            if self._ref[b] > 0:
                continue
            self._ref[b] = 0
            if b in self._block_key:
                self._cached_free[b] = None     # evictable but reusable
            else:
                self._free.append(b)

    # -------------------------------------------------------- prefix cache
    def lookup_prefix(self, keys: list[bytes]) -> list[int]:
        """Acquire (refcount) the longest cached chain matching ``keys``."""
        out: list[int] = []
        self.prefix_lookups += 1
        for key in keys:
            b = self._index.get(key)
            if b is None:
                break
            self._ref[b] += 1
            if self._ref[b] == 1:
                self._cached_free.pop(b, None)
            out.append(b)
        self.prefix_hits += len(out)
        return out

    def register_block(self, b: int, key: bytes) -> None:
        """Publish a fully-written prompt block under its content key."""
        if key in self._index or b in self._block_key:
            return
        self._index[key] = b
        self._block_key[b] = key

    def bytes_used(self) -> int:
        return self.k.numel() * self.k.element_size() * 2
