"""hipGraph-captured decode step.

The decode inner loop is launch-bound (~200 small kernels per token across
32 layers); capturing one full decode step in a hipGraph replaces per-kernel
host launches with a single graph replay (CDNA4 guidance: capture
launch-bound inner loops in hipGraphs).  All graph inputs are
static-address device buffers whose CONTENTS are updated before each
replay: token id, position, KV slot, block table, context length.
"""

from __future__ import annotations

from typing import Optional

import torch

from .kvcache import PAGE_SIZE, PagedKVCache


class DecodeGraph:
    def __init__(self, model, cache: PagedKVCache, max_pages: int, batch: int = 1) -> None:
        self.model = model
        self.cache = cache
        self.B = batch
        dev = model.device
        self.tokens = torch.zeros(batch, dtype=torch.long, device=dev)
        self.pos32 = torch.zeros(batch, dtype=torch.int32, device=dev)
        # int32: rope_kv_append consumes slot as int32 — a long here costs a
        # cast kernel PER LAYER per replayed step (the 4.7 us elementwise in
        # profiles/r01_decode_breakdown2.txt, ~3% of decode)
        self.slot = torch.zeros(batch, dtype=torch.int32, device=dev)
        self.bt = torch.zeros(batch, max_pages, dtype=torch.int32, device=dev)
        self.ctx = torch.zeros(batch, dtype=torch.int32, device=dev)
        self.graph: Optional[torch.cuda.CUDAGraph] = None
        self.out: Optional[torch.Tensor] = None

    def _capture(self) -> None:
        # warmup on a side stream (allocator priming), then capture
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self.model.decode_step_tensors(
                    self.tokens, self.pos32, self.cache.k, self.cache.v,
                    self.slot, self.bt, self.ctx)
        torch.cuda.current_stream().wait_stream(s)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.out = self.model.decode_step_tensors(
                self.tokens, self.pos32, self.cache.k, self.cache.v,
                self.slot, self.bt, self.ctx)

    def step(self, token: int, seq: int) -> torch.Tensor:
        """Decode one token for cache sequence ``seq``; returns hidden [B,H].

        Handles cache bookkeeping host-side (slot allocation, block table,
        logical length) and replays the captured graph.
        """
        cache = self.cache
        pos = cache.seq_lens[seq]
        cache._ensure_capacity(seq, pos + 1)
        page = cache.block_tables[seq][pos // PAGE_SIZE]
        self.tokens.fill_(token)
        self.pos32.fill_(pos)
        self.slot.fill_(page * PAGE_SIZE + pos % PAGE_SIZE)
        pages = cache.block_tables[seq]
        self.bt.zero_()
        self.bt[0, : len(pages)] = torch.tensor(pages, dtype=torch.int32,
                                                device=self.bt.device)
        self.ctx.fill_(pos + 1)
        if self.graph is None:
            self._capture()  # capture records without executing
        self.graph.replay()
        cache.seq_lens[seq] = pos + 1
        return self.out

    def step_batch(self, token_ids, seqs) -> torch.Tensor:
        """Lockstep decode of B sequences (len(seqs) == capture batch).

        Serving-throughput path: one replay advances every sequence one
        token; per-seq cache bookkeeping stays host-side exactly like
        step().  Returns hidden [B, H].
        """
        cache = self.cache
        assert len(seqs) == self.B and len(token_ids) == self.B
        from .kvcache import PAGE_SIZE as _PS
        poss, slots, cts = [], [], []
        for sq in seqs:
            pos = cache.seq_lens[sq]
            cache._ensure_capacity(sq, pos + 1)
            page = cache.block_tables[sq][pos // _PS]
            poss.append(pos)
            slots.append(page * _PS + pos % _PS)
            cts.append(pos + 1)
        dev = self.tokens.device
        self.tokens.copy_(torch.tensor(token_ids, dtype=torch.long))
        self.pos32.copy_(torch.tensor(poss, dtype=torch.int32))
        self.slot.copy_(torch.tensor(slots, dtype=torch.int32))
        self.bt.zero_()
        for i, sq in enumerate(seqs):
            pages = cache.block_tables[sq]
            self.bt[i, : len(pages)] = torch.tensor(pages, dtype=torch.int32,
                                                    device=dev)
        self.ctx.copy_(torch.tensor(cts, dtype=torch.int32))
        if self.graph is None:
            self._capture()
        self.graph.replay()
        for sq in seqs:
            cache.seq_lens[sq] += 1
        return self.out
