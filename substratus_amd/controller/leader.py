"""Leader election for the controller manager (reference: controller-runtime
`--leader-elect`, cmd/controllermanager/main.go:51-53).

Two modes behind one interface:
  * Lease objects on the kube API (real cluster / MemKube): acquire-or-renew
    a Lease with holderIdentity + renewTime; stale leases (> lease_seconds)
    are taken over.
  * File lock (the embedded daemon): flock on a pidfile under the data dir —
    prevents two daemons fighting over one bucket/runtime.
"""

from __future__ import annotations

import fcntl
import os
import socket
import time
from typing import Optional


class FileLeaderLock:
    def __init__(self, path: str):
        self.path = path
        self._fh = None

    def acquire(self, block: bool = False) -> bool:
        os.makedirs(os.path.dirname(self.path), exist_ok=True)
        self._fh = open(self.path, "w")
        try:
            flags = fcntl.LOCK_EX | (0 if block else fcntl.LOCK_NB)
            fcntl.flock(self._fh, flags)
        except OSError:
            self._fh.close()
            self._fh = None
            return False
        self._fh.write(f"{os.getpid()}@{socket.gethostname()}\n")
        self._fh.flush()
        return True

    def release(self) -> None:
        if self._fh is not None:
            fcntl.flock(self._fh, fcntl.LOCK_UN)
            self._fh.close()
            self._fh = None


class LeaseLeaderElector:
    """Lease-object election over any kube backend (MemKube or real)."""

    LEASE_NAME = "substratus-controller-manager"

    def __init__(self, kube, identity: Optional[str] = None,
                 lease_seconds: float = 15.0):
        self.kube = kube
        self.identity = identity or f"{socket.gethostname()}-{os.getpid()}"
        self.lease_seconds = lease_seconds

    def try_acquire(self) -> bool:
        now = time.time()
        cur = self.kube.try_get("Lease", self.LEASE_NAME, "kube-system")
        if cur is None:
            self.kube.create({
                "apiVersion": "coordination.k8s.io/v1", "kind": "Lease",
                "metadata": {"name": self.LEASE_NAME,
                             "namespace": "kube-system"},
                "spec": {"holderIdentity": self.identity,
                         "renewTime": now},
            })
            return True
        spec = cur.get("spec", {})
        holder = spec.get("holderIdentity")
        renew = float(spec.get("renewTime", 0))
        if holder == self.identity or now - renew > self.lease_seconds:
            self.kube.patch("Lease", self.LEASE_NAME, {"spec": {
                "holderIdentity": self.identity, "renewTime": now}},
                "kube-system")
            return True
        return False

    def renew(self) -> None:
        self.kube.patch("Lease", self.LEASE_NAME, {"spec": {
            "holderIdentity": self.identity, "renewTime": time.time()}},
            "kube-system")
