"""Tensor parallelism for the 70B scorer (config 5).

Megatron-style sharding sized for one 8-GPU xGMI node:
  - qkv projection: column-parallel (heads sharded — 70B: 64 q / 8 kv heads
    -> 8 q + 1 kv head per rank at TP=8);
  - attention: local on the rank's heads;
  - o_proj and down_proj: row-parallel, ONE bf16 all_reduce each over RCCL
    (2 all_reduces per layer — the standard minimum);
  - gate|up: column-parallel (intermediate sharded);
  - embeddings / norms / lm_head: replicated (fits easily in 288 GB HBM).

The all_reduce is bandwidth-bound at prefill sizes (M x hidden bf16);
with xGMI fully-connected 7x153 GB/s per GPU, RCCL's fully-connected
algorithms apply — no bucketing needed since each call is one tensor.

CPU tests run the same code over gloo (bf16 tensors are reduced in f32 on
gloo, bit-matching is not required there).
"""

from __future__ import annotations


import torch
import torch.distributed as dist


class TPContext:
    def __init__(self, rank: int = 0, world: int = 1, group=None) -> None:
        self.rank = rank
        self.world = world
        self.group = group

    @classmethod
    def from_default_group(cls) -> "TPContext":
        if dist.is_initialized():
            return cls(dist.get_rank(), dist.get_world_size(), None)
        return cls()

    def all_reduce_(self, x: torch.Tensor) -> torch.Tensor:
        if self.world == 1 or not dist.is_initialized():
            return x
        if dist.get_backend() == "gloo" and x.dtype == torch.bfloat16:
            xf = x.float()
            dist.all_reduce(xf, group=self.group)
            x.copy_(xf.to(x.dtype))
            return x
        dist.all_reduce(x, group=self.group)
        return x


def shard_rows(t: torch.Tensor, ctx: TPContext, dim: int = 0) -> torch.Tensor:
    """Slice a replicated tensor along ``dim`` into this rank's shard."""
    n = t.shape[dim]
    assert n % ctx.world == 0, f"cannot shard {n} over {ctx.world}"
    per = n // ctx.world
    return t.narrow(dim, ctx.rank * per, per).contiguous()


def shard_qkv(qkv: torch.Tensor, num_heads: int, num_kv_heads: int, head_dim: int,
              ctx: TPContext) -> torch.Tensor:
    """Shard a fused [Hq*D + 2*Hkv*D, K] qkv weight by heads."""
    q_size = num_heads * head_dim
    kv_size = num_kv_heads * head_dim
    q = shard_rows(qkv[:q_size], ctx)
    k = shard_rows(qkv[q_size: q_size + kv_size], ctx)
    v = shard_rows(qkv[q_size + kv_size:], ctx)
    return torch.cat([q, k, v], dim=0).contiguous()


def shard_gateup(gateup: torch.Tensor, intermediate: int, ctx: TPContext) -> torch.Tensor:
    """Shard a fused [2I, K] gate|up weight so the shard stays [gate|up]."""
    g = shard_rows(gateup[:intermediate], ctx)
    u = shard_rows(gateup[intermediate:], ctx)
    return torch.cat([g, u], dim=0).contiguous()
