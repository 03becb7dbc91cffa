from .tp import TPContext, shard_gateup, shard_qkv, shard_rows
from .dist import (
    CandidateParallelScorer,
    broadcast_strings,
    dp_scores_allreduce,
    init_from_env,
    rank_world,
)

__all__ = ["CandidateParallelScorer", "broadcast_strings", "dp_scores_allreduce",
           "init_from_env", "rank_world", "TPContext", "shard_gateup", "shard_qkv",
           "shard_rows"]
