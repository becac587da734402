"""Notebook shim — the local runtime's stand-in for the Jupyter container.

The production notebook image runs `notebook.sh` → JupyterLab on 8888
(images/Dockerfile.notebook; container contract: readiness GET /api,
NOTEBOOK_TOKEN honored). This offline environment has no jupyterlab, so the
LocalRuntime serves the SAME contract surface with a minimal FastAPI app:
GET /api → version info (the readiness probe), GET/PUT/DELETE /files/* over
the mounted /content tree, and GET /events — the nbwatch change stream
(containertools/nbwatch.py runs INSIDE the shim, the analog of the
reference shipping the nbwatch binary into the pod and exec-ing it,
internal/client/sync.go:28-135) that the CLI's two-way sync consumes.
Token checked like Jupyter's `?token=`.
"""

from __future__ import annotations

import os
import threading

from fastapi import FastAPI, HTTPException, Request
from fastapi.responses import JSONResponse, PlainTextResponse

from ..containertools.nbwatch import watched_files


def create_app(content_dir: str = "/content",
               token: str = "") -> FastAPI:
    app = FastAPI(title="substratus notebook shim")

    def check_token(req: Request) -> None:
        if token and req.query_params.get("token") != token:
            raise HTTPException(403, "bad token")

    @app.get("/api")
    def api():
        # the readiness probe target (container contract)
        return {"version": "substratus-notebook-shim"}

    @app.get("/")
    def root():
        return {"status": "ok", "content": content_dir}

    # ---- nbwatch event stream (consumed by cli/sync.py) ----------------
    events: list[dict] = []
    ev_lock = threading.Lock()
    state = {"prev": None, "stop": False}

    def _scan_once() -> None:
        cur = watched_files(content_dir)
        prev = state["prev"]
        state["prev"] = cur
        if prev is None:
            return
        with ev_lock:
            idx = len(events)
            for path, mtime in cur.items():
                rel = os.path.relpath(path, content_dir)
                if path not in prev:
                    events.append({"index": idx, "path": rel,
                                   "op": "CREATE"}); idx += 1
                elif mtime != prev[path]:
                    events.append({"index": idx, "path": rel,
                                   "op": "WRITE"}); idx += 1
            for path in prev:
                if path not in cur:
                    events.append({"index": idx,
                                   "path": os.path.relpath(path, content_dir),
                                   "op": "REMOVE"}); idx += 1

    def _scanner():
        import time
        while not state["stop"]:
            _scan_once()
            time.sleep(0.3)

    threading.Thread(target=_scanner, daemon=True).start()
    app.state.stop_scanner = lambda: state.update(stop=True)

    @app.get("/events")
    def get_events(req: Request, since: int = 0):
        check_token(req)
        _scan_once()   # make tests deterministic (no poll-interval wait)
        with ev_lock:
            return {"events": [e for e in events if e["index"] >= since],
                    "next": len(events)}

    @app.get("/files")
    def list_files(req: Request):
        check_token(req)
        out = []
        for root, dirs, files in os.walk(content_dir):
            dirs[:] = [d for d in dirs
                       if d not in ("data", "model", "artifacts")
                       and not d.startswith(".")]
            for f in files:
                out.append(os.path.relpath(os.path.join(root, f),
                                           content_dir))
        return {"files": sorted(out)}

    @app.get("/files/{path:path}")
    def get_file(path: str, req: Request):
        check_token(req)
        full = os.path.join(content_dir, path)
        if not os.path.isfile(full):
            raise HTTPException(404, path)
        with open(full) as f:
            return PlainTextResponse(f.read())

    @app.put("/files/{path:path}")
    async def put_file(path: str, req: Request):
        check_token(req)
        full = os.path.join(content_dir, path)
        os.makedirs(os.path.dirname(full) or content_dir, exist_ok=True)
        body = await req.body()
        with open(full, "wb") as f:
            f.write(body)
        return JSONResponse({"written": path})

    @app.delete("/files/{path:path}")
    def delete_file(path: str, req: Request):
        check_token(req)
        full = os.path.join(content_dir, path)
        if os.path.isfile(full):
            os.remove(full)
        return JSONResponse({"deleted": path})

    return app


def main() -> None:
    import uvicorn
    app = create_app(os.environ.get("CONTENT_DIR", "/content"),
                     os.environ.get("NOTEBOOK_TOKEN", ""))
    uvicorn.run(app, host="0.0.0.0",
                port=int(os.environ.get("PORT", 8888)))


if __name__ == "__main__":
    main()
