"""Paged KV cache sized for MI355X's 288 GB HBM3E.

Layout per layer: [num_pages, PAGE_SIZE=16, Hk, D] bf16 — a token slot is
contiguous over (Hk, D) so prefill append is one ``index_copy_`` per layer
and the decode kernel reads a position's per-head row as one contiguous
256 B stretch (see csrc/decode_attention.hip).
"""

from __future__ import annotations

from typing import List

import torch

PAGE_SIZE = 16


class PagedKVCache:
    def __init__(self, config, num_pages: int, device, dtype=torch.bfloat16,
                 num_kv_heads=None) -> None:
        self.config = config
        self.num_pages = num_pages
        self.device = device
        self.num_kv_heads = num_kv_heads if num_kv_heads is not None else config.num_kv_heads
        shape = (num_pages, PAGE_SIZE, self.num_kv_heads, config.head_dim)
        self.k = [torch.zeros(shape, dtype=dtype, device=device) for _ in range(config.num_layers)]
        self.v = [torch.zeros(shape, dtype=dtype, device=device) for _ in range(config.num_layers)]
        self._free = list(range(num_pages - 1, -1, -1))
        # per-sequence state
        self.block_tables: List[List[int]] = []
        self.seq_lens: List[int] = []

    # --- sequence management ---

    def new_seq(self) -> int:
        self.block_tables.append([])
        self.seq_lens.append(0)
        return len(self.block_tables) - 1

    def _ensure_capacity(self, seq: int, new_len: int) -> None:
        need = (new_len + PAGE_SIZE - 1) // PAGE_SIZE
        bt = self.block_tables[seq]
        while len(bt) < need:
            if not self._free:
                raise RuntimeError(
                f"KV cache out of pages: need {need - len(bt)} more for seq "
                f"{seq} len {new_len} ({self.num_pages} total, "
                f"{len(self._free)} free)")
            bt.append(self._free.pop())

    def free_seq(self, seq: int) -> None:
        self._free.extend(self.block_tables[seq])
        self.block_tables[seq] = []
        self.seq_lens[seq] = 0

    def slot_ids(self, seq: int, start: int, count: int) -> torch.Tensor:
        """Flat slot index (page*16 + off) for positions [start, start+count)."""
        bt = self.block_tables[seq]
        idx = [bt[p // PAGE_SIZE] * PAGE_SIZE + p % PAGE_SIZE for p in range(start, start + count)]
        return torch.tensor(idx, dtype=torch.long, device=self.device)

    # --- appends ---

    def append(self, layer: int, seq: int, k: torch.Tensor, v: torch.Tensor,
               advance_len: bool) -> None:
        """k/v: [T, Hk, D] for T new positions of sequence ``seq``."""
        T = k.shape[0]
        start = self.seq_lens[seq]
        self._ensure_capacity(seq, start + T)
        slots = self.slot_ids(seq, start, T)
        cfg = self.config
        kflat = self.k[layer].view(self.num_pages * PAGE_SIZE, self.num_kv_heads, cfg.head_dim)
        vflat = self.v[layer].view(self.num_pages * PAGE_SIZE, self.num_kv_heads, cfg.head_dim)
        kflat.index_copy_(0, slots, k)
        vflat.index_copy_(0, slots, v)
        if advance_len:  # the last layer advances the logical length
            self.seq_lens[seq] = start + T

    def block_table_tensor(self, seqs: List[int]) -> torch.Tensor:
        maxp = max(len(self.block_tables[s]) for s in seqs)
        bt = torch.zeros(len(seqs), max(maxp, 1), dtype=torch.int32, device=self.device)
        for i, s in enumerate(seqs):
            pages = self.block_tables[s]
            if pages:
                bt[i, : len(pages)] = torch.tensor(pages, dtype=torch.int32, device=self.device)
        return bt

    def ctx_lens_tensor(self, seqs: List[int]) -> torch.Tensor:
        return torch.tensor([self.seq_lens[s] for s in seqs], dtype=torch.int32, device=self.device)
