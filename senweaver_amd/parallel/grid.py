"""Parallelism grid: compose TP and EP over one process group.

world = tp * ep, rank = ep_idx * tp + tp_idx (TP ranks contiguous so a
TP group's all_reduces stay on one xGMI neighborhood; EP's all-to-all
strides across groups).  Every rank must call build_tp_ep_grid with the
same (tp, ep): torch.distributed requires new_group to be entered by all
ranks for every subgroup.

Semantics: TP ranks within an EP shard hold identical activations (the
row-parallel all_reduce restores the full hidden), so the EP exchange —
running once per TP index over its own subgroup — routes the same token
set consistently on every TP rank; experts are sharded over ep_idx and
replicated across tp_idx.
"""

from __future__ import annotations

from typing import Tuple

import torch.distributed as dist

from .ep import EPContext
from .tp import TPContext


def build_tp_ep_grid(tp: int, ep: int) -> Tuple[TPContext, EPContext]:
    if not dist.is_initialized():
        assert tp == 1 and ep == 1
        return TPContext(), EPContext()
    world = dist.get_world_size()
    rank = dist.get_rank()
    assert world == tp * ep, f"world {world} != tp {tp} * ep {ep}"
    tp_groups = [dist.new_group(list(range(e * tp, (e + 1) * tp)))
                 for e in range(ep)]
    ep_groups = [dist.new_group(list(range(t, world, tp)))
                 for t in range(tp)]
    ep_idx, tp_idx = divmod(rank, tp)
    return (TPContext(tp_idx, tp, tp_groups[ep_idx]),
            EPContext(ep_idx, ep, ep_groups[tp_idx]))
