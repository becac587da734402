"""Controller manager: watch-driven reconcile loop + dependency indexes.

Reference: cmd/controllermanager/main.go:40-241 wires 5 reconcilers + 4
BuildReconcilers into controller-runtime; internal/controller/manager.go:13-72
adds field indexes on spec.model.name / spec.dataset.name so a Model or
Dataset becoming Ready requeues every dependent Model/Server/Notebook.

Here: a single-threaded work queue over MemKube watches. run_until_settled()
drives reconciles to a fixpoint (the test/CLI entry); serve_forever() is the
long-running form.
"""

from __future__ import annotations

import time
from collections import deque
from typing import Optional

from ..api import types as apiv1
from ..cloud.base import Cloud
from ..kube.memapi import MemKube, NotFound
from .build import BuildReconciler
from .dataset import DatasetReconciler
from .model import ModelReconciler
from .notebook import NotebookReconciler
from .server import ServerReconciler


class Manager:
    def __init__(self, kube: MemKube, cloud: Cloud, sci):
        self.kube = kube
        self.cloud = cloud
        self.sci = sci
        build = BuildReconciler(kube, cloud, sci)
        self.reconcilers = {
            "Model": ModelReconciler(kube, cloud, sci, build),
            "Dataset": DatasetReconciler(kube, cloud, sci, build),
            "Server": ServerReconciler(kube, cloud, sci, build),
            "Notebook": NotebookReconciler(kube, cloud, sci, build),
        }
        self.queue: deque = deque()
        self._queued: set = set()
        self._delayed: list = []   # (due_ts, key)
        # observability (reference: controller-runtime prometheus metrics)
        self.metrics = {"reconcile_total": {}, "reconcile_errors_total": {},
                        "requeue_total": 0}
        for kind in list(self.reconcilers) + ["Job", "Pod", "Deployment"]:
            kube.watch(kind, self._on_event)

    # ------------------------------------------------------------ indexing
    def _dependents(self, kind: str, name: str, ns: str) -> list[tuple]:
        """Field-index equivalent (reference manager.go:23-72)."""
        out = []
        if kind == "Model":
            for k in ("Model", "Server", "Notebook"):
                for o in self.kube.list(k, ns):
                    if (o["spec"].get("model") or {}).get("name") == name:
                        out.append((k, o["metadata"]["name"], ns))
        if kind == "Dataset":
            for k in ("Model", "Notebook"):
                for o in self.kube.list(k, ns):
                    if (o["spec"].get("dataset") or {}).get("name") == name:
                        out.append((k, o["metadata"]["name"], ns))
        return out

    def _owners(self, obj: dict) -> list[tuple]:
        ns = obj["metadata"].get("namespace", "default")
        out = []
        for ref in obj["metadata"].get("ownerReferences", []):
            if ref.get("kind") in self.reconcilers:
                out.append((ref["kind"], ref["name"], ns))
        return out

    def _on_event(self, event: str, obj: dict) -> None:
        kind = obj.get("kind")
        ns = obj["metadata"].get("namespace", "default")
        name = obj["metadata"]["name"]
        if kind in self.reconcilers:
            self._enqueue((kind, name, ns))
            for dep in self._dependents(kind, name, ns):
                self._enqueue(dep)
        else:
            for owner in self._owners(obj):
                self._enqueue(owner)

    def _enqueue(self, key: tuple) -> None:
        if key not in self._queued:
            self._queued.add(key)
            self.queue.append(key)

    # ----------------------------------------------------------- reconcile
    def _reconcile_one(self, key: tuple) -> None:
        kind, name, ns = key
        self._queued.discard(key)
        try:
            obj = self.kube.get(kind, name, ns)
        except NotFound:
            return
        errs = apiv1.validate(obj)
        if errs:
            apiv1.set_condition(obj, "Valid", False, "ValidationFailed",
                                "; ".join(errs))
            self._persist_status(kind, name, ns, obj)
            return
        rec = self.reconcilers[kind]
        self.metrics["reconcile_total"][kind] = \
            self.metrics["reconcile_total"].get(kind, 0) + 1
        try:
            result = rec.reconcile(obj)
        except Exception:
            self.metrics["reconcile_errors_total"][kind] = \
                self.metrics["reconcile_errors_total"].get(kind, 0) + 1
            raise
        self._persist(kind, name, ns, obj)
        if not result.success and result.requeue_after > 0:
            self.metrics["requeue_total"] += 1
            self._delayed.append((time.time() + result.requeue_after, key))

    def _persist(self, kind: str, name: str, ns: str, obj: dict) -> None:
        # persist spec mutations (image stamping) + status separately,
        # avoiding event storms for no-op updates
        try:
            cur = self.kube.get(kind, name, ns)
        except NotFound:
            return
        if cur.get("spec") != obj.get("spec") or cur.get("status") != obj.get("status"):
            self.kube.patch(kind, name,
                            {"spec": obj.get("spec", {}),
                             "status": obj.get("status", {})}, ns)

    def _persist_status(self, kind, name, ns, obj) -> None:
        self.kube.patch(kind, name, {"status": obj.get("status", {})}, ns)

    # ------------------------------------------------------------- driving
    def _pump_delayed(self, now: Optional[float] = None) -> None:
        now = now if now is not None else time.time()
        due = [k for (t, k) in self._delayed if t <= now]
        self._delayed = [(t, k) for (t, k) in self._delayed if t > now]
        for k in due:
            self._enqueue(k)

    def run_once(self) -> int:
        """Drain the current queue once; returns number of reconciles."""
        n = 0
        while self.queue:
            self._reconcile_one(self.queue.popleft())
            n += 1
        return n

    def run_until_settled(self, timeout: float = 10.0,
                          force_delayed: bool = True) -> int:
        """Drive to fixpoint: drain queue, fast-forward delayed requeues
        (tests), until nothing is pending or timeout."""
        deadline = time.time() + timeout
        total = 0
        while time.time() < deadline:
            before = self.kube.changes
            total += self.run_once()
            if not self._delayed:
                break
            if force_delayed:
                self._pump_delayed(now=float("inf"))
            else:
                time.sleep(0.05)
                self._pump_delayed()
            if not self.queue:
                break
            if self.kube.changes == before:
                # a full drain + requeue produced no state change: fixpoint
                # with pending waits (e.g. gated on a dependency)
                self.run_once()
                if self.kube.changes == before:
                    break
        return total

    def serve_forever(self, poll: float = 0.2) -> None:
        while True:
            self.run_once()
            self._pump_delayed()
            time.sleep(poll)
