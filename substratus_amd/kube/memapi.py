"""In-memory Kubernetes API server ("envtest-lite").

The reference's controllers talk to a real kube-apiserver (envtest in its
integration tests — SURVEY.md §4); ours talk to this interface. Objects are
plain dicts in unstructured form ({apiVersion, kind, metadata, spec, status})
with resourceVersion bookkeeping, label selectors, watches, and owner-
reference garbage collection. A real-cluster backend can implement the same
five methods against the kubernetes client when one is available; no kubelet
exists either way, so tests fake Job/Pod status transitions exactly like the
reference's fakeJobComplete/fakePodReady (reference
internal/controller/main_test.go:245-265).
"""

from __future__ import annotations

import copy
import itertools
import threading
from typing import Callable, Iterable, Optional


class NotFound(KeyError):
    pass


class Conflict(RuntimeError):
    pass


def _key(kind: str, namespace: str, name: str) -> tuple:
    return (kind, namespace or "default", name)


def _content_equal(a: dict, b: dict) -> bool:
    sa, sb = copy.deepcopy(a), copy.deepcopy(b)
    for o in (sa, sb):
        o.get("metadata", {}).pop("resourceVersion", None)
    return sa == sb


def deep_merge(base: dict, patch: dict) -> dict:
    """Strategic-merge-lite: dicts merge recursively, everything else replaces;
    None deletes a key."""
    out = copy.deepcopy(base)
    for k, v in patch.items():
        if v is None:
            out.pop(k, None)
        elif isinstance(v, dict) and isinstance(out.get(k), dict):
            out[k] = deep_merge(out[k], v)
        else:
            out[k] = copy.deepcopy(v)
    return out


class MemKube:
    def __init__(self):
        self._objs: dict[tuple, dict] = {}
        self.changes = 0   # bumps on every event; managers use it to detect fixpoint
        self._rv = itertools.count(1)
        self._lock = threading.RLock()
        self._watchers: list[tuple[Optional[str], Callable[[str, dict], None]]] = []

    # ------------------------------------------------------------- helpers
    def _notify(self, event: str, obj: dict) -> None:
        self.changes += 1
        for kind, cb in list(self._watchers):
            if kind is None or kind == obj.get("kind"):
                cb(event, copy.deepcopy(obj))

    def watch(self, kind: Optional[str], cb: Callable[[str, dict], None]) -> None:
        """Subscribe to ADDED/MODIFIED/DELETED events for a kind (None=all)."""
        with self._lock:
            self._watchers.append((kind, cb))

    # ----------------------------------------------------------------- CRUD
    def create(self, obj: dict) -> dict:
        with self._lock:
            md = obj.setdefault("metadata", {})
            k = _key(obj["kind"], md.get("namespace", "default"), md["name"])
            if k in self._objs:
                raise Conflict(f"{k} already exists")
            md.setdefault("namespace", "default")
            md["resourceVersion"] = str(next(self._rv))
            md.setdefault("generation", 1)
            md.setdefault("uid", f"uid-{md['name']}-{md['resourceVersion']}")
            stored = copy.deepcopy(obj)
            self._objs[k] = stored
            self._notify("ADDED", stored)
            return copy.deepcopy(stored)

    def get(self, kind: str, name: str, namespace: str = "default") -> dict:
        with self._lock:
            k = _key(kind, namespace, name)
            if k not in self._objs:
                raise NotFound(f"{kind}/{namespace}/{name}")
            return copy.deepcopy(self._objs[k])

    def try_get(self, kind: str, name: str, namespace: str = "default") -> Optional[dict]:
        try:
            return self.get(kind, name, namespace)
        except NotFound:
            return None

    def update(self, obj: dict) -> dict:
        """Full replace; bumps generation when spec changes. A content-
        identical update (ignoring resourceVersion) is a no-op — no RV bump,
        no event — so reconcile→apply cycles settle."""
        with self._lock:
            md = obj["metadata"]
            k = _key(obj["kind"], md.get("namespace", "default"), md["name"])
            if k not in self._objs:
                raise NotFound(str(k))
            old = self._objs[k]
            if _content_equal(old, obj):
                return copy.deepcopy(old)
            new = copy.deepcopy(obj)
            new["metadata"]["resourceVersion"] = str(next(self._rv))
            new["metadata"]["uid"] = old["metadata"]["uid"]
            if new.get("spec") != old.get("spec"):
                new["metadata"]["generation"] = old["metadata"].get("generation", 1) + 1
            self._objs[k] = new
            self._notify("MODIFIED", new)
            return copy.deepcopy(new)

    def patch(self, kind: str, name: str, patch: dict,
              namespace: str = "default") -> dict:
        with self._lock:
            cur = self.get(kind, name, namespace)
            merged = deep_merge(cur, patch)
            merged["kind"] = kind
            merged["metadata"]["name"] = name
            merged["metadata"]["namespace"] = namespace
            return self.update(merged)

    def apply(self, obj: dict) -> dict:
        """Server-side-apply-lite: create or merge-update (the CLI's and the
        controllers' write primitive — reference internal/client/upload.go:110-124)."""
        with self._lock:
            md = obj.setdefault("metadata", {})
            ns = md.setdefault("namespace", "default")
            cur = self.try_get(obj["kind"], md["name"], ns)
            if cur is None:
                return self.create(obj)
            return self.patch(obj["kind"], md["name"], obj, ns)

    def delete(self, kind: str, name: str, namespace: str = "default") -> None:
        with self._lock:
            k = _key(kind, namespace, name)
            if k not in self._objs:
                raise NotFound(str(k))
            obj = self._objs.pop(k)
            self._notify("DELETED", obj)
            # owner-reference cascade (background GC)
            uid = obj["metadata"].get("uid")
            children = [o for o in self._objs.values()
                        if any(ref.get("uid") == uid for ref in
                               o["metadata"].get("ownerReferences", []))]
            for c in children:
                self.delete(c["kind"], c["metadata"]["name"],
                            c["metadata"].get("namespace", "default"))

    def list(self, kind: str, namespace: Optional[str] = "default",
             labels: Optional[dict] = None) -> list[dict]:
        with self._lock:
            out = []
            for (k, ns, _), o in self._objs.items():
                if k != kind:
                    continue
                if namespace is not None and ns != namespace:
                    continue
                if labels:
                    ol = o["metadata"].get("labels", {})
                    if any(ol.get(lk) != lv for lk, lv in labels.items()):
                        continue
                out.append(copy.deepcopy(o))
            return out

    # ------------------------------------------------------ status helpers
    def set_status(self, kind: str, name: str, status: dict,
                   namespace: str = "default") -> dict:
        return self.patch(kind, name, {"status": status}, namespace)
