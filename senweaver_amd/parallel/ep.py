"""Expert parallelism (EP) — cross-GPU expert sharding with all-to-all
token routing.

At BASELINE scales the experts fit in 288 GB and replicate per rank
(ROADMAP #5), but EP is what larger expert counts need: each rank OWNS
num_experts/world contiguous experts; routed tokens travel to their
expert's owner, run through the owner's grouped GEMM, and travel back.

Wire protocol per MoE layer (tokens already sorted by expert — the
grouped-GEMM order — so the send buffer needs no extra permutation):
  1. all_gather the per-expert counts -> counts_all [world, E]
     (every rank can then derive every segment length; no ids on the wire);
  2. all-to-all of the expert-sorted activations, split by expert OWNER;
  3. receiver permutes [src-major] -> [local-expert-major] (derived from
     counts_all), runs its grouped GEMM, inverts the permutation;
  4. all-to-all back; the caller's existing unsort/combine applies.

Transport: nccl (RCCL) uses all_to_all_single over xGMI (the natural
point-to-point fabric for this); gloo (CPU tests, 1-GPU burn-in) lacks
all_to_all, so an all_gather_object exchange stands in — identical
semantics, test-grade speed.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch
import torch.distributed as dist


class EPContext:
    def __init__(self, rank: int = 0, world: int = 1, group=None) -> None:
        self.rank = rank
        self.world = world
        self.group = group

    @classmethod
    def from_default_group(cls) -> "EPContext":
        if dist.is_initialized():
            return cls(dist.get_rank(), dist.get_world_size(), None)
        return cls()

    # ---- expert ownership (contiguous shards) ----
    def local_experts(self, num_experts: int) -> Tuple[int, int]:
        per = num_experts // self.world
        return self.rank * per, (self.rank + 1) * per

    def owner_of(self, expert: int, num_experts: int) -> int:
        return expert // (num_experts // self.world)

    # ---- transport ----
    def _all_to_all(self, x: torch.Tensor, out_split: List[int],
                    in_split: List[int]) -> torch.Tensor:
        """Exchange: send x's consecutive in_split[r] rows to rank r; receive
        out_split[r] rows from each rank, concatenated in rank order."""
        if self.world == 1 or not dist.is_initialized():
            return x
        if dist.get_backend() == "nccl":
            out = x.new_empty(sum(out_split), *x.shape[1:])
            dist.all_to_all_single(out, x.contiguous(),
                                   output_split_sizes=out_split,
                                   input_split_sizes=in_split,
                                   group=self.group)
            return out
        # gloo fallback: every rank publishes its per-destination pieces
        pieces = list(torch.split(x, in_split, dim=0))
        gathered: List[Optional[list]] = [None] * self.world
        dist.all_gather_object(gathered, [p.cpu() for p in pieces],
                               group=self.group)
        mine = [gathered[src][self.rank].to(x.device)
                for src in range(self.world)]
        return torch.cat(mine, dim=0) if mine else x.new_empty(0, *x.shape[1:])

    # ---- the per-layer routing ----
    def dispatch(self, x_sorted: torch.Tensor, counts: torch.Tensor,
                 num_experts: int):
        """x_sorted: [Tk, H] expert-sorted tokens; counts: [E] per-expert.

        Returns (x_local, local_counts, meta) where x_local is THIS rank's
        experts' tokens in local-expert-major order and local_counts its
        per-local-expert totals; meta drives combine()."""
        W, E = self.world, num_experts
        per = E // W
        counts_cpu = counts.to("cpu", torch.int64)
        counts_all = [torch.empty_like(counts_cpu) for _ in range(W)]
        if W > 1 and dist.is_initialized():
            dist.all_gather(counts_all, counts_cpu, group=self.group)
        else:
            counts_all = [counts_cpu]
        counts_mat = torch.stack(counts_all)             # [W, E]

        in_split = [int(counts_cpu[r * per:(r + 1) * per].sum())
                    for r in range(W)]
        elo, ehi = self.local_experts(E)
        out_split = [int(counts_mat[src, elo:ehi].sum()) for src in range(W)]
        recv = self._all_to_all(x_sorted, out_split, in_split)

        # src-major -> local-expert-major permutation from counts_mat
        seg_of = []          # (src, e) in receive order
        for src in range(W):
            for e in range(elo, ehi):
                seg_of.append((src, e, int(counts_mat[src, e])))
        order = sorted(range(len(seg_of)),
                       key=lambda i: (seg_of[i][1], seg_of[i][0]))
        starts = []
        pos = 0
        for (_s, _e, n) in seg_of:
            starts.append(pos)
            pos += n
        perm_idx: List[int] = []
        for i in order:
            s0 = starts[i]
            perm_idx.extend(range(s0, s0 + seg_of[i][2]))
        perm = torch.tensor(perm_idx, dtype=torch.long, device=recv.device)
        x_local = recv[perm] if len(perm_idx) else recv
        local_counts = counts_mat[:, elo:ehi].sum(dim=0)  # [per]
        meta = (perm, out_split, in_split)
        return x_local, local_counts, meta

    def combine(self, y_local: torch.Tensor, meta) -> torch.Tensor:
        """Inverse of dispatch: local-expert-major results travel back and
        come out aligned with the caller's x_sorted rows."""
        perm, out_split, in_split = meta
        if len(perm):
            inv = torch.empty_like(perm)
            inv[perm] = torch.arange(len(perm), device=perm.device)
            y_src_major = y_local[inv]
        else:
            y_src_major = y_local
        # reverse direction: send out_split back, receive in_split
        return self._all_to_all(y_src_major, in_split, out_split)
