"""First-party DDP gradient buckets over RCCL/xGMI (SURVEY.md §2.5 ledger
item 10) — not torch.nn.parallel.DDP.

Gradients are bucketed (default 50 MiB) and all-reduced asynchronously as
soon as a bucket fills during backward, overlapping communication with the
remaining backward compute. Bucket size targets the xGMI topology: each
MI355X has 7 p2p links × ≈153 GB/s, ring collectives are per-link bound, so
fewer/larger transfers amortize latency while still overlapping.

Works on gloo for world_size>1 CPU tests (SURVEY.md §4 test plan).
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from ..parallel.comm import state


class GradBuckets:
    def __init__(self, params: list[torch.Tensor],
                 bucket_bytes: int = 50 << 20):
        self.st = state()
        self.params = [p for p in params if p.requires_grad]
        self.bucket_bytes = bucket_bytes
        self._pending: list[torch.Tensor] = []
        self._pending_bytes = 0
        self._handles: list = []
        self._hooks = []
        if self.st.dp_size > 1:
            for p in self.params:
                h = p.register_post_accumulate_grad_hook(self._on_grad)
                self._hooks.append(h)

    def _on_grad(self, p: torch.Tensor) -> None:
        g = p.grad
        if g is None:
            return
        self._pending.append(g)
        self._pending_bytes += g.numel() * g.element_size()
        if self._pending_bytes >= self.bucket_bytes:
            self._flush()

    def _flush(self) -> None:
        if not self._pending:
            return
        grads = self._pending
        self._pending = []
        self._pending_bytes = 0
        flat = torch.cat([g.reshape(-1) for g in grads])
        flat /= self.st.dp_size
        work = dist.all_reduce(flat, group=self.st.dp_group, async_op=True)
        self._handles.append((work, flat, grads))

    def finalize(self) -> None:
        """Flush the tail bucket and scatter reduced values back into grads.
        Call after backward(), before the optimizer step."""
        if self.st.dp_size <= 1:
            return
        self._flush()
        for work, flat, grads in self._handles:
            work.wait()
            off = 0
            for g in grads:
                n = g.numel()
                g.copy_(flat[off:off + n].view_as(g))
                off += n
        self._handles = []

    def remove_hooks(self) -> None:
        for h in self._hooks:
            h.remove()
        self._hooks = []
