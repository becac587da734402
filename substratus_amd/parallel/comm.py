"""Process-group setup for one-process-per-GPU execution over RCCL/xGMI.

torch.distributed backend "nccl" IS RCCL on ROCm; CPU tests use "gloo"
(multi-process paths are covered with gloo world_size>1 on CPU).
The controller guarantees all ranks of a Server/Job land on one node (single
Pod requesting amd.com/gpu: N — reference shape), so every
communicator spans one xGMI-connected node.

Two orthogonal axes:
  * TP (tensor parallel, serving): consecutive ranks form a TP group of size
    tp_size; row-parallel layers all-reduce inside it.
  * DP (data parallel, training or replicated serving): the remaining factor;
    the trainer's gradient buckets all-reduce across DP peers.
"""

from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import Optional

import torch
import torch.distributed as dist


@dataclass
class ParallelState:
    rank: int = 0
    world_size: int = 1
    local_rank: int = 0
    initialized: bool = False
    tp_rank: int = 0
    tp_size: int = 1
    dp_rank: int = 0
    dp_size: int = 1
    tp_group: Optional[object] = None
    dp_group: Optional[object] = None
    # gloo group over the same TP ranks for host-side control messages
    # (per-iteration admission broadcasts): broadcast_object_list over the
    # RCCL group would stage through the GPU and synchronize the device
    # every step — scheduling traffic belongs on the CPU.
    tp_ctrl_group: Optional[object] = None

    @property
    def is_rank0(self) -> bool:
        return self.rank == 0


_STATE = ParallelState()


def state() -> ParallelState:
    return _STATE


class single_process:
    """Context manager: build modules as if world_size==1 (e.g. construct a
    FULL reference model inside a TP process before sharding it)."""

    def __enter__(self):
        global _STATE
        self._saved = _STATE
        _STATE = ParallelState()
        return self

    def __exit__(self, *exc):
        global _STATE
        _STATE = self._saved
        return False


def init_distributed(tp_size: Optional[int] = None,
                     backend: Optional[str] = None) -> ParallelState:
    """Initialize from torchrun env vars. tp_size defaults to world_size
    (pure TP); tp_size=1 gives pure DP. No-op for world=1."""
    global _STATE
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        _STATE = ParallelState()
        return _STATE
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if not dist.is_initialized():
        # The container hostname may not resolve — force loopback rendezvous.
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29531")
        dist.init_process_group(backend=backend)
    rank = dist.get_rank()
    local = int(os.environ.get("LOCAL_RANK", rank))
    if torch.cuda.is_available():
        torch.cuda.set_device(local)
    tp = tp_size if tp_size is not None else world
    assert world % tp == 0, f"world {world} not divisible by tp {tp}"
    dp = world // tp
    tp_group = dp_group = tp_ctrl_group = None
    if tp > 1 and tp < world:
        for i in range(dp):
            ranks = list(range(i * tp, (i + 1) * tp))
            g = dist.new_group(ranks)
            gc = dist.new_group(ranks, backend="gloo") if backend == "nccl" \
                else g
            if rank // tp == i:
                tp_group, tp_ctrl_group = g, gc
    elif tp == world and tp > 1 and backend == "nccl":
        tp_ctrl_group = dist.new_group(backend="gloo")
    if dp > 1 and dp < world:
        for j in range(tp):
            g = dist.new_group(list(range(j, world, tp)))
            if rank % tp == j:
                dp_group = g
    _STATE = ParallelState(
        rank=rank, world_size=world, local_rank=local, initialized=True,
        tp_rank=rank % tp, tp_size=tp, dp_rank=rank // tp, dp_size=dp,
        tp_group=tp_group, dp_group=dp_group, tp_ctrl_group=tp_ctrl_group)
    return _STATE


def barrier() -> None:
    if _STATE.initialized:
        dist.barrier()


def tp_all_reduce_(t: torch.Tensor) -> torch.Tensor:
    if _STATE.initialized and _STATE.tp_size > 1:
        dist.all_reduce(t, group=_STATE.tp_group)
    return t


def tp_all_gather_cat(t: torch.Tensor, dim: int = -1) -> torch.Tensor:
    if not (_STATE.initialized and _STATE.tp_size > 1):
        return t
    parts = [torch.empty_like(t) for _ in range(_STATE.tp_size)]
    dist.all_gather(parts, t.contiguous(), group=_STATE.tp_group)
    return torch.cat(parts, dim=dim)


def dp_all_reduce_(t: torch.Tensor, op=None) -> torch.Tensor:
    if _STATE.initialized and _STATE.dp_size > 1:
        dist.all_reduce(t, group=_STATE.dp_group)
    return t
