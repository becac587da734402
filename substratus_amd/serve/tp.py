"""Tensor-parallel serving coordination.

A multi-GPU Server Pod (amd.com/gpu: N, reference
examples/llama2-70b/server.yaml) runs torchrun with one rank per GPU; rank 0
owns the HTTP endpoint and is the scheduling leader. Every engine iteration
the leader broadcasts the list of newly admitted requests (torch.distributed
broadcast_object_list over the TP group), then ALL ranks execute the
identical engine step — the model's row-parallel layers all-reduce inside the
forward, so schedules must be lockstep. Sampling is deterministic across
ranks: logits are replicated after the final all-reduce and the sampling
draw comes from identically seeded generators.
"""

from __future__ import annotations

from typing import Optional

import torch.distributed as dist

from ..parallel.comm import state


class TPCoordinator:
    def __init__(self):
        self.st = state()

    @property
    def is_leader(self) -> bool:
        return self.st.tp_rank == 0

    @property
    def active(self) -> bool:
        return self.st.initialized and self.st.tp_size > 1

    @property
    def _ctrl(self):
        # host-side control messages ride the gloo group (no GPU sync);
        # tp_group only when running pure-gloo (CPU tests)
        return self.st.tp_ctrl_group or self.st.tp_group

    def publish(self, adds: list[dict], aborts: list[int] | None = None,
                stop: bool = False) -> None:
        """Leader: broadcast this iteration's admissions + aborts to
        followers. Seq ids are assigned by the same monotonic counter on
        every rank (followers replay add_request in publish order), so an
        abort id names the same sequence everywhere."""
        if not self.active:
            return
        obj = [{"adds": adds, "aborts": aborts or [], "stop": stop}]
        dist.broadcast_object_list(obj, src=self.st.rank - self.st.tp_rank,
                                   group=self._ctrl)

    def receive(self) -> dict:
        """Follower: receive the leader's iteration command."""
        obj = [None]
        dist.broadcast_object_list(obj, src=self.st.rank - self.st.tp_rank,
                                   group=self._ctrl)
        return obj[0]

    def broadcast_config(self, payload: Optional[list] = None) -> list:
        """One-shot startup broadcast of the leader's engine configuration
        (EngineConfig wire dict + gguf path) so follower ranks build an
        IDENTICAL engine — quant/GGUF/kv_dtype desync between ranks hangs
        the first collective otherwise."""
        obj = list(payload) if payload is not None else [None, None]
        dist.broadcast_object_list(obj, src=self.st.rank - self.st.tp_rank,
                                   group=self._ctrl)
        return obj


def follower_loop(engine) -> None:
    """Ranks 1..tp-1: mirror the leader's admissions and step in lockstep.
    A step failure is handled exactly like the leader's worker handles it
    (engine.abort_all) — both sides recover to the same empty schedule
    instead of the follower dying and hanging the next collective."""
    coord = TPCoordinator()
    while True:
        cmd = coord.receive()
        if cmd.get("stop"):
            return
        for add in cmd["adds"]:
            engine.add_request(**add)
        for sid in cmd.get("aborts", []):
            engine.abort(sid)
        if engine.has_work():
            try:
                engine.step()
            except Exception:  # noqa: BLE001  (mirror leader recovery)
                engine.abort_all()
