from .dist import (
    CandidateParallelScorer,
    broadcast_strings,
    dp_scores_allreduce,
    init_from_env,
    rank_world,
)
from .ep import EPContext
from .grid import build_tp_ep_grid
from .tp import TPContext, shard_gateup, shard_qkv, shard_rows

__all__ = ["CandidateParallelScorer", "broadcast_strings", "dp_scores_allreduce",
           "init_from_env", "rank_world", "TPContext", "EPContext",
           "build_tp_ep_grid", "shard_gateup", "shard_qkv", "shard_rows"]
