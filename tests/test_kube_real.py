"""RealKube (real-cluster backend) against a MOCKED kubernetes client
(the backend was 157 untested lines whose watch
ignored core kinds). The `kubernetes` package is not installed in this
image, so the mock doubles as the only way to exercise the code paths:
routing, CRUD arg shapes, and the reconnecting multi-kind watch."""

import sys
import threading
import time
import types

import pytest


class _Recorder:
    def __init__(self):
        self.calls = []

    def _rec(self, name, *a, **k):
        self.calls.append((name, a, k))
        return {"metadata": {"name": a[-2] if len(a) >= 2 else "?"}}


def _install_fake_kubernetes(stream_script):
    """Build a fake `kubernetes` package. stream_script: list of callables,
    one per Watch.stream() call — each returns an iterable of events or
    raises."""
    rec = _Recorder()

    class CustomObjectsApi:
        def create_namespaced_custom_object(self, *a, **k):
            return rec._rec("crd.create", *a, **k)

        def get_namespaced_custom_object(self, *a, **k):
            return rec._rec("crd.get", *a, **k)

        def patch_namespaced_custom_object(self, *a, **k):
            return rec._rec("crd.patch", *a, **k)

        def delete_namespaced_custom_object(self, *a, **k):
            return rec._rec("crd.delete", *a, **k)

        def list_namespaced_custom_object(self, *a, **k):
            rec._rec("crd.list", *a, **k)
            return {"items": []}

    class CoreV1Api:
        def __getattr__(self, name):
            if name.startswith("list_namespaced_"):
                def fn(*a, **k):
                    return rec._rec(f"core.{name}", *a, **k)
                fn.__name__ = name
                return fn
            raise AttributeError(name)

    class BatchV1Api:
        def list_namespaced_job(self, *a, **k):
            return rec._rec("batch.list_job", *a, **k)

    class AppsV1Api:
        def list_namespaced_deployment(self, *a, **k):
            return rec._rec("apps.list_dep", *a, **k)

    class ApiClient:
        def sanitize_for_serialization(self, obj):
            return obj if isinstance(obj, dict) else {"metadata": {}}

    class Watch:
        streams = list(stream_script)
        calls = []

        def stream(self, fn, *a, **k):
            Watch.calls.append((fn, a, k))
            script = Watch.streams.pop(0) if Watch.streams else None
            if script is None:
                time.sleep(0.05)
                return iter(())
            return script(k)

    client_mod = types.ModuleType("kubernetes.client")
    client_mod.CustomObjectsApi = CustomObjectsApi
    client_mod.CoreV1Api = CoreV1Api
    client_mod.BatchV1Api = BatchV1Api
    client_mod.AppsV1Api = AppsV1Api
    client_mod.ApiClient = ApiClient
    config_mod = types.ModuleType("kubernetes.config")
    config_mod.load_kube_config = lambda **k: None
    config_mod.load_incluster_config = lambda: (_ for _ in ()).throw(
        RuntimeError("not in cluster"))
    watch_mod = types.ModuleType("kubernetes.watch")
    watch_mod.Watch = Watch
    pkg = types.ModuleType("kubernetes")
    pkg.client = client_mod
    pkg.config = config_mod
    pkg.watch = watch_mod
    sys.modules["kubernetes"] = pkg
    sys.modules["kubernetes.client"] = client_mod
    sys.modules["kubernetes.config"] = config_mod
    sys.modules["kubernetes.watch"] = watch_mod
    return rec, Watch


@pytest.fixture()
def fake_kube():
    saved = {k: sys.modules.get(k) for k in
             ("kubernetes", "kubernetes.client", "kubernetes.config",
              "kubernetes.watch", "kubernetes.dynamic")}
    yield
    for k, v in saved.items():
        if v is None:
            sys.modules.pop(k, None)
        else:
            sys.modules[k] = v


def test_crud_routing(fake_kube):
    rec, _ = _install_fake_kubernetes([])
    from substratus_amd.kube.real import RealKube
    rk = RealKube(kubeconfig="x")
    rk.create({"kind": "Model", "apiVersion": "substratus.ai/v1",
               "metadata": {"name": "m1", "namespace": "ns1"}})
    rk.get("Server", "s1", "ns2")
    rk.patch("Dataset", "d1", {"status": {"ready": True}})
    rk.delete("Notebook", "n1")
    names = [c[0] for c in rec.calls]
    assert names == ["crd.create", "crd.get", "crd.patch", "crd.delete"]
    # CRD group/version/plural arg shape
    assert rec.calls[0][1][:3] == ("substratus.ai", "v1", "ns1")
    assert rec.calls[0][1][3] == "models"
    assert rec.calls[1][1][3] == "servers"
    assert rk.try_get("Model", "whatever") is not None


def test_watch_covers_core_batch_apps_kinds(fake_kube):
    """Jobs/Pods/Deployments are watchable (r01: silently ignored)."""
    def one_event(kw):
        yield {"type": "ADDED", "object": {
            "metadata": {"name": "j1", "resourceVersion": "5"}}}
        while True:
            time.sleep(0.05)
            yield {"type": "BOOKMARK", "object": {"metadata": {}}}

    rec, Watch = _install_fake_kubernetes([one_event, one_event, one_event])
    from substratus_amd.kube.real import RealKube
    rk = RealKube(kubeconfig="x")
    got = []
    stops = []
    for kind in ("Job", "Pod", "Deployment"):
        ev = threading.Event()
        stops.append(rk.watch(
            kind, lambda t, o, ev=ev: (got.append((t, o)), ev.set())))
        assert ev.wait(5), f"no event for {kind}"
    for s in stops:
        s()
    assert len(got) >= 3
    streamed_fns = [getattr(c[0], "__name__", "") + str(c[0])
                    for c in Watch.calls]
    assert any("job" in s for s in streamed_fns)
    assert any("pod" in s for s in streamed_fns)
    assert any("deployment" in s for s in streamed_fns)


def test_watch_reconnects_and_resumes_rv(fake_kube):
    """Stream dies -> reconnect with the last resourceVersion; a 410
    expiry clears it (fresh relist)."""
    def first(kw):
        yield {"type": "ADDED", "object": {
            "metadata": {"name": "a", "resourceVersion": "41"}}}
        raise RuntimeError("connection reset")

    def second(kw):
        # must resume from rv 41
        assert kw.get("resource_version") == "41"
        yield {"type": "MODIFIED", "object": {
            "metadata": {"name": "a", "resourceVersion": "42"}}}
        raise RuntimeError("410 Gone: too old resource version")

    def third(kw):
        # 410 cleared the rv
        assert "resource_version" not in kw
        yield {"type": "ADDED", "object": {
            "metadata": {"name": "a", "resourceVersion": "43"}}}
        while True:
            time.sleep(0.05)
            yield {"type": "BOOKMARK", "object": {"metadata": {}}}

    rec, Watch = _install_fake_kubernetes([first, second, third])
    from substratus_amd.kube.real import RealKube
    rk = RealKube(kubeconfig="x")
    seen = []
    done = threading.Event()

    def cb(t, o):
        seen.append((t, o["metadata"].get("resourceVersion")))
        if len(seen) >= 3:
            done.set()

    stop = rk.watch("Model", cb)
    assert done.wait(30), seen
    stop()
    assert [s[1] for s in seen[:3]] == ["41", "42", "43"]


def test_unroutable_kind_raises(fake_kube):
    _install_fake_kubernetes([])
    from substratus_amd.kube.real import RealKube
    rk = RealKube(kubeconfig="x")
    with pytest.raises(KeyError):
        rk.watch("Gizmo", lambda t, o: None)
