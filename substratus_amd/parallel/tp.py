"""Tensor-parallel linear layers over RCCL/xGMI.

Column-parallel: weight rows (output features) sharded; no comm on forward.
Row-parallel: weight cols (input features) sharded; one all-reduce on forward.
Per transformer layer the server does 2 all-reduces (post-attention o_proj and
post-MLP down_proj) — SURVEY.md §2.5 ledger item 9. On 8×MI355X the xGMI
topology is 7 p2p links/GPU, so for the small latency-sensitive decode
all-reduces RCCL's one-shot algorithm is preferred (RCCL chooses by size; we
keep tensors contiguous and per-step message counts minimal).

Reference behavioral contract: multi-GPU Server Pods request amd.com/gpu: N
(reference api/v1/common_types.go:102-107); the parallelism itself is new,
first-party (the reference's data plane is external — SURVEY.md §0).
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from .comm import state, tp_all_reduce_
from .. import ops


def _shard(n: int, world: int, rank: int) -> tuple[int, int]:
    assert n % world == 0, f"cannot shard {n} over {world} ranks"
    per = n // world
    return per * rank, per


class Linear(nn.Module):
    """Plain (replicated) linear, bias optional, bf16 weight."""

    def __init__(self, in_features: int, out_features: int, bias: bool = False,
                 device=None, dtype=None):
        super().__init__()
        self.in_features, self.out_features = in_features, out_features
        self.weight = nn.Parameter(torch.empty(out_features, in_features,
                                               device=device, dtype=dtype))
        self.bias = nn.Parameter(torch.zeros(out_features, device=device,
                                             dtype=dtype)) if bias else None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # ops.linear routes decode-shaped inference GEMMs to the
        # first-party 256x256 glds kernel (mfma_gemm.hip), everything
        # else to hipBLASLt
        return ops.linear(x, self.weight, self.bias)


class ColumnParallelLinear(Linear):
    """Output features sharded across TP ranks; output stays sharded."""

    def __init__(self, in_features: int, out_features: int, bias: bool = False,
                 device=None, dtype=None):
        st = state()
        self.tp_world = st.tp_size
        self.tp_rank = st.tp_rank
        self.full_out = out_features
        _, per = _shard(out_features, self.tp_world, self.tp_rank)
        super().__init__(in_features, per, bias=bias, device=device, dtype=dtype)

    def gather(self, y: torch.Tensor) -> torch.Tensor:
        from .comm import tp_all_gather_cat
        return tp_all_gather_cat(y, dim=-1)


class RowParallelLinear(Linear):
    """Input features sharded across TP ranks; forward all-reduces the output."""

    def __init__(self, in_features: int, out_features: int, bias: bool = False,
                 device=None, dtype=None):
        st = state()
        self.tp_world = st.tp_size
        self.tp_rank = st.tp_rank
        self.full_in = in_features
        _, per = _shard(in_features, self.tp_world, self.tp_rank)
        super().__init__(per, out_features, bias=bias, device=device, dtype=dtype)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = ops.linear(x, self.weight, None)
        tp_all_reduce_(y)
        if self.bias is not None:
            y = y + self.bias
        return y
