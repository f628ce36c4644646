from .combine import (
    combine_partials,
    tree_combine,
    tree_combine_allgather,
    tree_combine_allreduce,
)
from .pg import cleanup, is_distributed, local_device, setup
from .tree import TreeAttention, tree_attention, tree_decode

__all__ = [
    "TreeAttention",
    "cleanup",
    "combine_partials",
    "is_distributed",
    "local_device",
    "setup",
    "tree_attention",
    "tree_combine",
    "tree_combine_allgather",
    "tree_combine_allreduce",
    "tree_decode",
]
