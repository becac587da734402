"""HTTP API server over the in-memory Kubernetes API + platform daemon.

This is what makes the platform real without a cluster: `sub up` starts one
process serving the substratus.ai/v1 API over HTTP, with the controller
manager, the local runtime (kubelet-equivalent) and the kind SCI all running
inside — the moral equivalent of the reference's
install/kubernetes/kind/system.yaml stack (namespace + CRDs + controller-
manager Deployment + SCI + registry) collapsed into one binary for
single-node MI355X deployments. The `sub` CLI talks to this API exactly like
the reference CLI talks to a kube-apiserver (server-side apply + status
polling — reference internal/client/client.go:39-152).
"""

from __future__ import annotations

import os
import threading
import time
from typing import Optional

from fastapi import FastAPI, HTTPException, Request
from ..api import types as apiv1
from .memapi import Conflict, MemKube, NotFound


def create_api(kube: MemKube, platform: Optional["Platform"] = None) -> FastAPI:
    app = FastAPI(title="substratus-amd apiserver")
    app.state.kube = kube

    @app.get("/healthz")
    def healthz():
        return {"ok": True}

    @app.get("/readyz")
    def readyz():
        return {"ok": True}

    @app.get("/metrics")
    def metrics():
        # controller metrics in Prometheus exposition format (reference:
        # controller-runtime metrics behind kube-rbac-proxy)
        from fastapi import Response
        lines = []
        if platform is not None:
            m = platform.mgr.metrics
            for kind, n in m["reconcile_total"].items():
                lines.append(
                    f'substratus_reconcile_total{{kind="{kind}"}} {n}')
            for kind, n in m["reconcile_errors_total"].items():
                lines.append(
                    f'substratus_reconcile_errors_total{{kind="{kind}"}} {n}')
            lines.append(f'substratus_requeue_total {m["requeue_total"]}')
        lines.append(f"substratus_objects {len(kube.list('Model', None)) + len(kube.list('Server', None)) + len(kube.list('Dataset', None)) + len(kube.list('Notebook', None))}")
        return Response("\n".join(lines) + "\n", media_type="text/plain")

    @app.post("/api/objects")
    async def create(req: Request):
        obj = await req.json()
        errs = apiv1.validate(obj) if obj.get("kind") in apiv1.KINDS else []
        if errs:
            raise HTTPException(422, "; ".join(errs))
        try:
            return kube.create(obj)
        except Conflict as e:
            raise HTTPException(409, str(e))

    @app.put("/api/objects")
    async def apply(req: Request):
        obj = await req.json()
        errs = apiv1.validate(obj) if obj.get("kind") in apiv1.KINDS else []
        if errs:
            raise HTTPException(422, "; ".join(errs))
        return kube.apply(obj)

    @app.get("/api/objects/{kind}")
    def list_objs(kind: str, namespace: str = "default"):
        return {"items": kube.list(kind.capitalize() if kind[0].islower()
                                   else kind, namespace)}

    @app.get("/api/objects/{kind}/{namespace}/{name}")
    def get_obj(kind: str, namespace: str, name: str):
        try:
            return kube.get(kind, name, namespace)
        except NotFound:
            raise HTTPException(404, f"{kind}/{namespace}/{name} not found")

    @app.patch("/api/objects/{kind}/{namespace}/{name}")
    async def patch_obj(kind: str, namespace: str, name: str, req: Request):
        body = await req.json()
        try:
            return kube.patch(kind, name, body, namespace)
        except NotFound:
            raise HTTPException(404, "not found")

    @app.get("/api/logs/{namespace}/{name}")
    def logs(namespace: str, name: str, tail: int = 200, since: int = -1):
        # pod-log streaming equivalent (the reference TUI streams the
        # default container's logs — internal/tui/pods.go:222-246).
        # since >= 0: return lines from that absolute line index (the
        # follow cursor `sub logs -f` polls with); response carries
        # `next`, the cursor for the following poll.
        if platform is None:
            raise HTTPException(404, "no runtime")
        path = platform.runtime.log_path(namespace, name)
        if not os.path.exists(path):
            raise HTTPException(404, f"no logs for {namespace}/{name}")
        with open(path, "rb") as f:
            lines = f.read().decode(errors="replace").splitlines()
        if since >= 0:
            return {"lines": lines[since:], "next": len(lines)}
        return {"lines": lines[-tail:], "next": len(lines)}

    @app.delete("/api/objects/{kind}/{namespace}/{name}")
    def delete_obj(kind: str, namespace: str, name: str):
        try:
            kube.delete(kind, name, namespace)
        except NotFound:
            raise HTTPException(404, "not found")
        return {"deleted": True}

    return app


class Platform:
    """Embedded control plane: MemKube + controllers + local runtime + SCI."""

    def __init__(self, data_dir: str, sci_address: Optional[str] = None):
        from ..cloud import new_cloud
        from ..controller.manager import Manager
        from ..runtime.local import LocalRuntime
        self.sci_address = sci_address
        os.makedirs(data_dir, exist_ok=True)
        self.bucket_dir = os.path.join(data_dir, "bucket")
        os.makedirs(self.bucket_dir, exist_ok=True)
        self.kube = MemKube()
        self.cloud = new_cloud({"CLOUD": "kind", "CLUSTER_NAME": "local",
                                "BUCKET_DIR": self.bucket_dir})
        self.sci_http_port = int(os.environ.get("SCI_HTTP_PORT", "8071"))
        from ..sci.kind_server import serve as sci_serve
        self._sci_server, self._sci_httpd = sci_serve(
            self.bucket_dir, grpc_port=int(os.environ.get("SCI_GRPC_PORT",
                                                          "10080")),
            http_port=self.sci_http_port,
            url_host=f"http://127.0.0.1:{self.sci_http_port}", block=False)
        from ..sci.client import GrpcSCIClient
        sci_client = GrpcSCIClient(
            self.sci_address or
            f"localhost:{os.environ.get('SCI_GRPC_PORT', '10080')}")
        self.mgr = Manager(self.kube, self.cloud, sci_client)
        self.runtime = LocalRuntime(self.kube, self.bucket_dir)
        self._stop = False
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()

    def _loop(self) -> None:
        while not self._stop:
            try:
                self.mgr.run_once()
                self.mgr._pump_delayed()
                self.runtime.sync()
            except Exception:  # keep the control loop alive
                import traceback
                traceback.print_exc()
            time.sleep(0.25)

    def stop(self) -> None:
        self._stop = True
        # let the reconcile loop observe _stop before tearing the runtime
        # down under it
        if getattr(self, "_thread", None) is not None:
            self._thread.join(timeout=5)
        self.runtime.stop()
        self._sci_server.stop(grace=None)
        self._sci_httpd.shutdown()


def main() -> None:
    import argparse
    import json
    import sys
    import uvicorn
    # flags mirror the reference controller-manager's
    # (cmd/controllermanager/main.go:41-58)
    ap = argparse.ArgumentParser("substratus platform daemon")
    ap.add_argument("--leader-elect", action="store_true",
                    help="refuse to start if another daemon holds the lock")
    ap.add_argument("--sci-address", default=None,
                    help="external SCI gRPC address (default: embedded kind SCI)")
    ap.add_argument("--metrics-bind-address", default=None)
    ap.add_argument("--config-dump-path", default=None,
                    help="dump resolved config as JSON and continue")
    ap.add_argument("--data-dir", default=os.environ.get(
        "SUBSTRATUS_DATA", os.path.expanduser("~/.substratus")))
    args = ap.parse_args()
    if args.leader_elect:
        from ..controller.leader import FileLeaderLock
        lock = FileLeaderLock(os.path.join(args.data_dir, "leader.lock"))
        if not lock.acquire():
            print("another daemon holds the leader lock; exiting",
                  file=sys.stderr)
            raise SystemExit(1)
    platform = Platform(args.data_dir, sci_address=args.sci_address)
    if args.config_dump_path:
        with open(args.config_dump_path, "w") as f:
            json.dump({"cloud": platform.cloud.name(),
                       "cluster_name": platform.cloud.cluster_name,
                       "artifact_bucket_url": platform.cloud.artifact_bucket_url,
                       "registry_url": platform.cloud.registry_url,
                       "data_dir": args.data_dir}, f, indent=2)
    app = create_api(platform.kube, platform)
    port = int(os.environ.get("SUBSTRATUS_API_PORT", "8070"))
    uvicorn.run(app, host="127.0.0.1", port=port, log_level="warning")


if __name__ == "__main__":
    main()
