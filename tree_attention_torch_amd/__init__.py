"""MI355X-native Tree Attention framework.

A from-scratch implementation of the capability set of
kyegomez/Tree-Attention-Torch (arXiv:2408.04093's tree-topology decode) for
AMD Instinct MI355X (gfx950, CDNA4): per-shard flash-attention partials from
hand-written HIP kernels (MFMA QK^T / PV, LDS-staged K/V, true online-softmax
LSE), combined across the 8 GPUs of a node with a numerically-stable
log-sum-exp tree reduction expressed as RCCL collectives over xGMI.

Public surface (reference parity, SURVEY.md §0 capability set):

    setup / cleanup        process-group lifecycle (model.py:11-33)
    make_data              per-rank synthetic Q/KV shards (model.py:37-56)
    flash_res_lse          local partial (out, true LSE) (model.py:60-83)
    local_attention        same, with GQA/causal/offset controls
    tree_decode            sharded decode (model.py:85-124, fixed semantics)
    tree_attention         general entry: decode + chunked causal prefill
    TreeAttention          nn.Module wrapper
    TreeAttentionConfig    workload/runtime config
"""

from .config import TreeAttentionConfig
from .data import make_data
from .ops.flash import flash_res_lse, hip_available, local_attention
from .ops.reference import attention_reference
from .parallel.combine import (
    combine_partials,
    tree_combine,
    tree_combine_allgather,
    tree_combine_allreduce,
)
from .parallel.pg import cleanup, is_distributed, local_device, setup
from .parallel.tree import TreeAttention, tree_attention, tree_decode
from .session import DecodeSession
from .utils.logging import logger

__version__ = "0.1.0"

__all__ = [
    "DecodeSession",
    "TreeAttention",
    "TreeAttentionConfig",
    "attention_reference",
    "cleanup",
    "combine_partials",
    "flash_res_lse",
    "hip_available",
    "is_distributed",
    "local_attention",
    "local_device",
    "logger",
    "make_data",
    "setup",
    "tree_attention",
    "tree_combine",
    "tree_combine_allgather",
    "tree_combine_allreduce",
    "tree_decode",
]
