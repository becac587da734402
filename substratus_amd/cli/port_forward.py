"""Port-forward with exponential-backoff restart
(reference internal/client/port_forward.go:21-45 +
internal/tui/portforward.go:17-67).

Two transports behind one restart loop:
  * kubectl mode (real cluster): spawn `kubectl port-forward pod/<name>
    local:remote` — the reference's SPDY dial, delegated to kubectl the
    same way its `cp` helper shells out (internal/client/cp/kubectl.go).
  * relay mode (local runtime / tests): an in-process TCP relay
    localhost:local → target host:port. The local runtime's pods are plain
    processes with a `substratus.ai/local-endpoint` annotation, so the
    relay IS the port-forward for them.

The restart loop mirrors the reference: on forwarder exit, retry with
exponential backoff (1s, 2s, 4s, ... capped), reset after a healthy run.
"""

from __future__ import annotations

import socket
import subprocess
import threading
import time
from typing import Callable, Optional


class _TCPRelay:
    """Minimal TCP proxy: accept on local_port, pipe to (host, port)."""

    def __init__(self, local_port: int, host: str, port: int):
        self.local_port = local_port
        self.target = (host, port)
        self._srv: Optional[socket.socket] = None
        self._stop = threading.Event()

    def serve(self) -> None:
        srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        srv.bind(("127.0.0.1", self.local_port))
        srv.listen(16)
        srv.settimeout(0.2)
        self._srv = srv
        try:
            while not self._stop.is_set():
                try:
                    conn, _ = srv.accept()
                except socket.timeout:
                    continue
                threading.Thread(target=self._pipe_pair, args=(conn,),
                                 daemon=True).start()
        finally:
            srv.close()

    def _pipe_pair(self, conn: socket.socket) -> None:
        try:
            up = socket.create_connection(self.target, timeout=5)
        except OSError:
            conn.close()
            return

        def pipe(a, b):
            try:
                while True:
                    data = a.recv(65536)
                    if not data:
                        break
                    b.sendall(data)
            except OSError:
                pass
            finally:
                try:
                    b.shutdown(socket.SHUT_WR)
                except OSError:
                    pass

        t = threading.Thread(target=pipe, args=(up, conn), daemon=True)
        t.start()
        pipe(conn, up)
        t.join(timeout=5)
        conn.close()
        up.close()

    def stop(self) -> None:
        self._stop.set()


class PortForwarder:
    """Runs a forwarder with backoff restarts until stop()."""

    def __init__(self, local_port: int,
                 target_host: str = "", target_port: int = 0,
                 kubectl_target: str = "", namespace: str = "default",
                 max_backoff: float = 16.0):
        self.local_port = local_port
        self.target_host = target_host
        self.target_port = target_port
        self.kubectl_target = kubectl_target   # e.g. "pod/my-notebook"
        self.namespace = namespace
        self.max_backoff = max_backoff
        self._stop = threading.Event()
        self._relay: Optional[_TCPRelay] = None
        self._proc: Optional[subprocess.Popen] = None
        self.restarts = 0

    def _run_once(self) -> None:
        if self.kubectl_target:
            self._proc = subprocess.Popen(
                ["kubectl", "port-forward", "-n", self.namespace,
                 self.kubectl_target,
                 f"{self.local_port}:{self.target_port}"],
                stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
            while not self._stop.is_set() and self._proc.poll() is None:
                time.sleep(0.2)
            if self._proc.poll() is None:
                self._proc.terminate()
        else:
            self._relay = _TCPRelay(self.local_port, self.target_host,
                                    self.target_port)
            self._relay.serve()   # returns on stop or bind error

    def run(self) -> None:
        backoff = 1.0
        while not self._stop.is_set():
            t0 = time.time()
            try:
                self._run_once()
            except OSError:
                pass
            if self._stop.is_set():
                return
            if time.time() - t0 > 10:
                backoff = 1.0   # healthy run: reset (reference behavior)
            self.restarts += 1
            self._stop.wait(backoff)
            backoff = min(backoff * 2, self.max_backoff)

    def start(self) -> threading.Thread:
        t = threading.Thread(target=self.run, daemon=True)
        t.start()
        return t

    def stop(self) -> None:
        self._stop.set()
        if self._relay is not None:
            self._relay.stop()
        if self._proc is not None and self._proc.poll() is None:
            self._proc.terminate()
