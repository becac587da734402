"""`sub` — the substratus CLI (reference cmd/sub + internal/cli, cobra).

Commands (reference internal/cli/*.go):
  sub up                      start the local platform daemon
  sub run [dir]               tar dir → Model/Dataset with build.upload →
                              upload → wait ready (-i increment, -r replace)
  sub serve [dir]             apply the Server manifest → wait → print URL
  sub notebook [dir] | nb     convert manifest to Notebook, apply, wait
  sub apply -f FILE           server-side apply any manifest
  sub get [scope]             list kinds with ready state (models, models/x)
  sub delete kind/name        delete
  sub infer                   completion request against a ready Server

Rendering uses rich tables/spinners instead of the reference's bubbletea TUI
(internal/tui; same information architecture: ready checkmarks, conditions,
endpoints).
"""

from __future__ import annotations

import glob
import json
import os
import subprocess
import sys
import time
from typing import Optional

import typer
import yaml
from rich.console import Console
from rich.table import Table

from ..api import types as apiv1
from . import client as libclient

app = typer.Typer(help="substratus-amd: ML platform CLI", no_args_is_help=True)
console = Console()

API_URL = os.environ.get("SUBSTRATUS_API", "http://127.0.0.1:8070")


def _client() -> libclient.HTTPClient:
    c = libclient.HTTPClient(API_URL)
    try:
        c.http.get("/healthz").raise_for_status()
    except Exception:
        console.print("[red]cannot reach the platform API at "
                      f"{API_URL} — run `sub up` first[/red]")
        raise typer.Exit(1)
    return c


def _find_manifests(path: str, kinds: tuple[str, ...]) -> list[dict]:
    """Find *.yaml manifests of the given kinds (reference
    internal/tui/manifests.go:130-263)."""
    out = []
    files = [path] if os.path.isfile(path) else \
        sorted(glob.glob(os.path.join(path, "*.yaml")) +
               glob.glob(os.path.join(path, "*.yml")))
    for f in files:
        with open(f) as fh:
            for doc in yaml.safe_load_all(fh):
                if isinstance(doc, dict) and doc.get("kind") in kinds:
                    out.append(doc)
    return out


@app.command()
def up(foreground: bool = typer.Option(False, "--foreground", "-f")):
    """Start the local platform daemon (apiserver + controllers + runtime +
    SCI — the kind-cluster equivalent, install/kind/up.sh)."""
    if foreground:
        from ..kube.apiserver import main as serve_main
        serve_main()
        return
    log = os.path.expanduser("~/.substratus/daemon.log")
    os.makedirs(os.path.dirname(log), exist_ok=True)
    with open(log, "a") as lf:
        proc = subprocess.Popen(
            [sys.executable, "-m", "substratus_amd.kube.apiserver"],
            stdout=lf, stderr=lf, start_new_session=True)
    with open(os.path.expanduser("~/.substratus/daemon.pid"), "w") as pf:
        pf.write(str(proc.pid))
    for _ in range(50):
        try:
            import httpx
            httpx.get(f"{API_URL}/healthz", timeout=1).raise_for_status()
            console.print(f"[green]platform up at {API_URL}[/green]")
            return
        except Exception:
            time.sleep(0.2)
    console.print(f"[red]daemon did not come up; see {log}[/red]")
    raise typer.Exit(1)


@app.command()
def down():
    """Stop the local platform daemon started by `sub up` (by its recorded
    PID — never by process-name pattern)."""
    import signal
    pidfile = os.path.expanduser("~/.substratus/daemon.pid")
    if not os.path.exists(pidfile):
        console.print("no daemon pidfile; nothing to stop")
        return
    with open(pidfile) as f:
        pid = int(f.read().strip())
    try:
        os.kill(pid, signal.SIGTERM)
        console.print(f"stopped daemon (pid {pid})")
    except ProcessLookupError:
        console.print(f"daemon (pid {pid}) already gone")
    os.remove(pidfile)


@app.command()
def run(directory: str = typer.Argument("."),
        increment: bool = typer.Option(False, "-i"),
        replace: bool = typer.Option(False, "-r"),
        wait: bool = typer.Option(True)):
    """Build-and-run a Model/Dataset from a directory
    (reference internal/cli/run.go:16-104)."""
    c = _client()
    manifests = _find_manifests(directory, ("Model", "Dataset"))
    if not manifests:
        console.print("[red]no Model/Dataset manifest found[/red]")
        raise typer.Exit(1)
    for obj in manifests:
        kind, name = obj["kind"], obj["metadata"]["name"]
        ns = obj["metadata"].setdefault("namespace", "default")
        if replace and c.get(kind, name, ns) is not None:
            c.delete(kind, name, ns)
        elif increment:
            obj["metadata"]["name"] = libclient.increment_name(c, kind, name, ns)
        has_docker = os.path.exists(os.path.join(directory, "Dockerfile"))
        if has_docker and not (obj.get("spec") or {}).get("image"):
            with console.status(f"uploading build context for {kind}/{name}"):
                obj = libclient.upload_build(c, obj, directory)
        else:
            obj = c.apply(obj)
        if wait:
            _wait_and_report(c, obj)


def _wait_and_report(c, obj: dict) -> None:
    kind, name = obj["kind"], obj["metadata"]["name"]
    ns = obj["metadata"].get("namespace", "default")
    with console.status(f"waiting for {kind}/{name} to become ready") as st:
        def prog(o):
            conds = o.get("status", {}).get("conditions", []) or []
            if conds:
                last = conds[-1]
                st.update(f"{kind}/{name}: {last.get('type')}="
                          f"{last.get('status')} ({last.get('reason', '')})")
        libclient.wait_ready(c, kind, name, ns, progress=prog)
    console.print(f"[green]✓ {kind}/{name} ready[/green]")


@app.command()
def serve(directory: str = typer.Argument(".")):
    """Apply the Server manifest in the directory and wait for it
    (reference internal/cli/serve.go:17-96)."""
    c = _client()
    manifests = _find_manifests(directory, ("Server",))
    if not manifests:
        console.print("[red]no Server manifest found[/red]")
        raise typer.Exit(1)
    for obj in manifests:
        obj = c.apply(obj)
        _wait_and_report(c, obj)
        name = obj["metadata"]["name"]
        ns = obj["metadata"].get("namespace", "default")
        svc = c.get("Service", f"{name}-server", ns)
        ep = (svc or {}).get("metadata", {}).get("annotations", {}).get(
            "substratus.ai/local-endpoint", "")
        if ep:
            console.print(f"serving at [bold]{ep}[/bold] "
                          f"(POST {ep}/v1/completions)")


@app.command()
def notebook(directory: str = typer.Argument("."),
             sync: bool = typer.Option(False, "--sync",
                                       help="two-way file sync with the "
                                            "notebook until Ctrl-C"),
             port: int = typer.Option(0, "--port",
                                      help="local port-forward port "
                                           "(0 = direct endpoint)")):
    """Convert the directory's manifest to a Notebook, open it, and
    optionally sync files both ways (reference internal/cli/notebook.go +
    internal/client/sync.go: nbwatch events -> copy both directions)."""
    c = _client()
    manifests = _find_manifests(
        directory, ("Model", "Dataset", "Server", "Notebook"))
    if not manifests:
        console.print("[red]no manifest found[/red]")
        raise typer.Exit(1)
    nb = manifests[0] if manifests[0]["kind"] == "Notebook" else \
        libclient.notebook_for_object(manifests[0])
    nb = c.apply(nb)
    _wait_and_report(c, nb)
    name = nb["metadata"]["name"]
    ns = nb["metadata"].get("namespace", "default")
    pod = c.get("Pod", f"{name}-notebook", ns) or {}
    ep = pod.get("metadata", {}).get("annotations", {}).get(
        "substratus.ai/local-endpoint", "http://localhost:8888")
    forwarder = None
    if port:
        from urllib.parse import urlparse
        from .port_forward import PortForwarder
        u = urlparse(ep)
        forwarder = PortForwarder(port, target_host=u.hostname or "127.0.0.1",
                                  target_port=u.port or 8888)
        forwarder.start()
        ep = f"http://localhost:{port}"
    console.print(f"notebook [bold]{name}[/bold] ready at "
                  f"{ep}/?token=default")
    if sync:
        from .sync import NotebookSync
        s = NotebookSync(directory, ep, token="default")
        n = s.push_all()
        console.print(f"synced {n} files up; watching both ways "
                      f"(Ctrl-C to stop)")
        try:
            s.run()
        except KeyboardInterrupt:
            pass
        finally:
            s.stop()
            if forwarder:
                forwarder.stop()


app.command(name="nb")(notebook)


@app.command()
def apply(file: str = typer.Option(..., "-f", "--filename")):
    """Server-side apply any manifest (reference internal/cli/apply.go)."""
    c = _client()
    with open(file) as fh:
        for doc in yaml.safe_load_all(fh):
            if not isinstance(doc, dict):
                continue
            out = c.apply(doc)
            console.print(f"{out['kind'].lower()}/"
                          f"{out['metadata']['name']} applied")


@app.command()
def get(scope: str = typer.Argument("all"),
        watch: bool = typer.Option(False, "--watch", "-w",
                                   help="refresh until interrupted"),
        logs: bool = typer.Option(False, "--logs",
                                  help="with --watch: stream the workload "
                                       "logs of not-ready objects under "
                                       "the table")):
    """List objects: `sub get models`, `sub get models/name`, `sub get all`
    (reference internal/cli/get.go; scope grammar tui/get.go:219-287;
    --watch mirrors the live watch-driven listing; --watch --logs adds the
    reference TUI's pod-log viewports, internal/tui/pods.go:222-246)."""
    c = _client()
    if watch:
        from rich.live import Live
        with Live(_watch_view(c, scope, logs), refresh_per_second=2) as live:
            while True:
                time.sleep(1)
                live.update(_watch_view(c, scope, logs))
    console.print(_get_table(c, scope))


_WORKLOAD_SUFFIX = {"Model": "-modeller", "Dataset": "-data-loader",
                    "Server": "-server", "Notebook": "-notebook"}


def _watch_view(c, scope: str, show_logs: bool):
    table = _get_table(c, scope)
    if not show_logs:
        return table
    from rich.console import Group
    from rich.panel import Panel
    panels = [table]
    for kind in apiv1.KINDS:
        for obj in c.list(kind):
            if apiv1.get_status_ready(obj):
                continue
            name = obj["metadata"]["name"]
            ns = obj["metadata"].get("namespace", "default")
            wl = name + _WORKLOAD_SUFFIX[kind]
            try:
                r = c.http.get(f"/api/logs/{ns}/{wl}", params={"tail": 8})
                if r.status_code == 200 and r.json()["lines"]:
                    panels.append(Panel("\n".join(r.json()["lines"]),
                                        title=f"{kind.lower()}/{name} "
                                              f"({wl})", height=10))
            except Exception:
                pass
    return Group(*panels)


def _get_table(c, scope: str):
    kind_filter, name_filter = None, None
    if scope != "all":
        parts = scope.split("/")
        kind_filter = parts[0].rstrip("s").capitalize()
        if len(parts) > 1:
            name_filter = parts[1]
    table = Table()
    for col in ("KIND", "NAME", "READY", "CONDITION"):
        table.add_column(col)
    for kind in apiv1.KINDS:
        if kind_filter and kind != kind_filter:
            continue
        for obj in c.list(kind):
            name = obj["metadata"]["name"]
            if name_filter and name != name_filter:
                continue
            ready = "✓" if apiv1.get_status_ready(obj) else "…"
            conds = obj.get("status", {}).get("conditions", []) or []
            last = (f"{conds[-1].get('type')}={conds[-1].get('status')}"
                    f" ({conds[-1].get('reason', '')})") if conds else ""
            table.add_row(kind, name, ready, last)
    return table


@app.command()
def delete(target: str):
    """Delete kind/name (reference internal/cli/delete.go)."""
    c = _client()
    kind, name = target.split("/", 1)
    kind = kind.rstrip("s").capitalize()
    c.delete(kind, name)
    console.print(f"{kind.lower()}/{name} deleted")


@app.command()
def logs(name: str, tail: int = typer.Option(100, "--tail"),
         namespace: str = typer.Option("default", "-n"),
         follow: bool = typer.Option(False, "--follow", "-f",
                                     help="stream new lines until Ctrl-C"),
         poll: float = typer.Option(0.5, "--poll", hidden=True),
         max_polls: int = typer.Option(-1, "--max-polls", hidden=True)):
    """Logs of an object's workload: `sub logs myserver-server` /
    `sub logs -f mymodel-modeller` (reference: pod log streaming in the
    TUI, internal/tui/pods.go:222-246)."""
    c = _client()
    r = c.http.get(f"/api/logs/{namespace}/{name}", params={"tail": tail})
    if r.status_code == 404:
        console.print(f"[red]no logs for {name}[/red]")
        raise typer.Exit(1)
    body = r.json()
    for line in body["lines"]:
        console.print(line, highlight=False)
    if not follow:
        return
    cursor = body.get("next", 0)
    n = 0
    try:
        while max_polls < 0 or n < max_polls:
            time.sleep(poll)
            n += 1
            r = c.http.get(f"/api/logs/{namespace}/{name}",
                           params={"since": cursor})
            if r.status_code != 200:
                continue
            body = r.json()
            for line in body["lines"]:
                console.print(line, highlight=False)
            cursor = body.get("next", cursor)
    except KeyboardInterrupt:
        pass


@app.command()
def suspend(name: str, namespace: str = typer.Option("default", "-n")):
    """Suspend a Notebook (reaps its pod; the reference TUI's 'q' menu
    suspend action, internal/tui/notebook.go)."""
    c = _client()
    if c.get("Notebook", name, namespace) is None:
        console.print(f"[red]notebook {name} not found[/red]")
        raise typer.Exit(1)
    c.patch("Notebook", name, {"spec": {"suspend": True}}, namespace)
    console.print(f"notebook [bold]{name}[/bold] suspended")


@app.command()
def resume(name: str, namespace: str = typer.Option("default", "-n")):
    """Resume a suspended Notebook."""
    c = _client()
    if c.get("Notebook", name, namespace) is None:
        console.print(f"[red]notebook {name} not found[/red]")
        raise typer.Exit(1)
    c.patch("Notebook", name, {"spec": {"suspend": False}}, namespace)
    console.print(f"notebook [bold]{name}[/bold] resumed")


@app.command()
def infer(server: str = typer.Option(..., "--server", "-s"),
          prompt: str = typer.Argument(...),
          max_tokens: int = typer.Option(32),
          stream: bool = typer.Option(False, "--stream",
                                      help="print tokens as they arrive "
                                           "(SSE; ctrl-c aborts the "
                                           "request server-side)")):
    """One completion request against a ready Server
    (reference internal/cli/infer.go — the chat TUI stub, made functional)."""
    import json as _json

    import httpx
    c = _client()
    svc = c.get("Service", f"{server}-server")
    ep = (svc or {}).get("metadata", {}).get("annotations", {}).get(
        "substratus.ai/local-endpoint")
    if not ep:
        console.print(f"[red]server {server} has no local endpoint[/red]")
        raise typer.Exit(1)
    if stream:
        import sys
        with httpx.stream("POST", f"{ep}/v1/completions",
                          json={"prompt": prompt, "max_tokens": max_tokens,
                                "stream": True}, timeout=300) as r:
            r.raise_for_status()
            for line in r.iter_lines():
                if not line.startswith("data:"):
                    continue
                payload = line[5:].strip()
                if payload == "[DONE]":
                    break
                obj = _json.loads(payload)
                if "error" in obj:
                    console.print(f"[red]{obj['error']['message']}[/red]")
                    raise typer.Exit(1)
                sys.stdout.write(obj["choices"][0]["text"])
                sys.stdout.flush()
        sys.stdout.write("\n")
        return
    r = httpx.post(f"{ep}/v1/completions",
                   json={"prompt": prompt, "max_tokens": max_tokens},
                   timeout=300)
    r.raise_for_status()
    console.print(r.json()["choices"][0]["text"])


def main() -> None:
    app()


if __name__ == "__main__":
    main()
