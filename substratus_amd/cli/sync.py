"""Two-way notebook file sync (reference internal/client/sync.go:28-293).

The reference ships the nbwatch binary into the pod, execs it over SPDY,
streams fsnotify JSON events, and `kubectl cp`s changed files both ways.
Here the pod side is the notebook shim, which runs nbwatch itself and
exposes the event stream and file bodies over its HTTP surface
(serve/notebook_shim.py: GET /events, GET/PUT/DELETE /files/*), so the
transport is a port-forwardable HTTP endpoint instead of SPDY exec — the
same sync semantics, MI355X-pod-shaped.

Local→remote: nbwatch's scanner runs on the local directory; CREATE/WRITE
push the file, REMOVE deletes it remotely.
Remote→local: the shim's /events stream is polled from a cursor;
CREATE/WRITE pull the file, REMOVE deletes it locally.

Loop prevention mirrors the reference's upload-skip: each side records the
content hash it last wrote for a path and ignores the echo event.
"""

from __future__ import annotations

import hashlib
import os
import threading
import time
from typing import Optional

import httpx

from ..containertools.nbwatch import watched_files


def _hash(data: bytes) -> str:
    return hashlib.md5(data).hexdigest()


class NotebookSync:
    def __init__(self, local_dir: str, base_url: str, token: str = "",
                 client: Optional[httpx.Client] = None):
        self.local_dir = local_dir
        self.base = base_url.rstrip("/")
        self.token = token
        self.http = client or httpx.Client(timeout=30)
        self._cursor = 0
        self._prev_local: Optional[dict] = None
        # path -> md5 of content we last wrote on that side (echo guard)
        self._wrote_remote: dict[str, str] = {}
        self._wrote_local: dict[str, str] = {}
        self._stop = threading.Event()

    # ------------------------------------------------------------- plumbing
    def _params(self) -> dict:
        return {"token": self.token} if self.token else {}

    def _push(self, rel: str) -> None:
        full = os.path.join(self.local_dir, rel)
        try:
            with open(full, "rb") as f:
                data = f.read()
        except OSError:
            return
        if self._wrote_local.get(rel) == _hash(data):
            return  # echo of a remote->local copy
        self.http.put(f"{self.base}/files/{rel}", content=data,
                      params=self._params())
        self._wrote_remote[rel] = _hash(data)

    def _pull(self, rel: str) -> None:
        r = self.http.get(f"{self.base}/files/{rel}", params=self._params())
        if r.status_code != 200:
            return
        data = r.content
        if self._wrote_remote.get(rel) == _hash(data):
            return  # echo of a local->remote copy
        full = os.path.join(self.local_dir, rel)
        os.makedirs(os.path.dirname(full) or self.local_dir, exist_ok=True)
        with open(full, "wb") as f:
            f.write(data)
        self._wrote_local[rel] = _hash(data)

    # ----------------------------------------------------------- sync steps
    def push_all(self) -> int:
        """Initial upload of the local directory (reference: dir upload
        before the watch loop starts)."""
        n = 0
        self._prev_local = watched_files(self.local_dir)
        for path in self._prev_local:
            self._push(os.path.relpath(path, self.local_dir))
            n += 1
        # start the remote cursor AFTER the upload so the pushes' own
        # events don't come back as pulls
        r = self.http.get(f"{self.base}/events",
                          params={**self._params(), "since": 1 << 30})
        self._cursor = r.json()["next"]
        return n

    def sync_local_once(self) -> int:
        """Diff the local dir against the last scan; push changes up."""
        if self._prev_local is None:
            self._prev_local = watched_files(self.local_dir)
            return 0
        cur = watched_files(self.local_dir)
        prev, self._prev_local = self._prev_local, cur
        n = 0
        for path, mtime in cur.items():
            if path not in prev or mtime != prev[path]:
                self._push(os.path.relpath(path, self.local_dir))
                n += 1
        for path in prev:
            if path not in cur:
                rel = os.path.relpath(path, self.local_dir)
                self.http.delete(f"{self.base}/files/{rel}",
                                 params=self._params())
                self._wrote_remote.pop(rel, None)
                n += 1
        return n

    def sync_remote_once(self) -> int:
        """Consume the shim's nbwatch events from the cursor; pull/delete
        locally (reference sync.go:184-273 event loop)."""
        r = self.http.get(f"{self.base}/events",
                          params={**self._params(), "since": self._cursor})
        body = r.json()
        self._cursor = body["next"]
        n = 0
        for ev in body["events"]:
            rel, op = ev["path"], ev["op"]
            if op in ("CREATE", "WRITE"):
                self._pull(rel)
                n += 1
            elif op == "REMOVE":
                if self._wrote_remote.pop(rel, None) is not None and \
                        not os.path.exists(os.path.join(self.local_dir, rel)):
                    continue  # echo of our own local delete
                full = os.path.join(self.local_dir, rel)
                if os.path.isfile(full):
                    os.remove(full)
                    self._wrote_local.pop(rel, None)
                n += 1
        return n

    # ------------------------------------------------------------ run loops
    def run(self, interval: float = 0.5,
            iterations: int = -1) -> None:
        """Blocking two-way loop (Ctrl-C or stop() ends it)."""
        i = 0
        while not self._stop.is_set() and (iterations < 0 or i < iterations):
            self.sync_local_once()
            self.sync_remote_once()
            self._stop.wait(interval)
            i += 1

    def start(self, interval: float = 0.5) -> threading.Thread:
        t = threading.Thread(target=self.run, args=(interval,), daemon=True)
        t.start()
        return t

    def stop(self) -> None:
        self._stop.set()
