"""Controllers driven through RealKube against a STATEFUL fake kubernetes
client — the offline analog of the reference's envtest tier
(internal/controller/main_test.go:46-191: real kube-apiserver, fake cloud +
SCI, Job/Pod statuses patched by the test). Here the kubernetes client API
is faked on top of MemKube storage, so every controller write/read goes
through RealKube's actual marshaling (CRD group/version/plural routing,
dynamic server-side apply, merge patches) instead of short-circuiting to
MemKube."""

import sys
import time
import types

import pytest

from substratus_amd.api import types as apiv1
from substratus_amd.api import conditions as cond
from substratus_amd.cloud import new_cloud
from substratus_amd.kube.memapi import MemKube
from substratus_amd.sci.base import FakeSCIClient

PLURAL_TO_KIND = {"models": "Model", "datasets": "Dataset",
                  "servers": "Server", "notebooks": "Notebook"}


def _install_stateful_fake(store: MemKube):
    class CustomObjectsApi:
        def create_namespaced_custom_object(self, group, ver, ns, plural,
                                            body):
            assert group == "substratus.ai" and ver == "v1"
            return store.create(body)

        def get_namespaced_custom_object(self, group, ver, ns, plural,
                                         name):
            return store.get(PLURAL_TO_KIND[plural], name, ns)

        def patch_namespaced_custom_object(self, group, ver, ns, plural,
                                           name, body):
            return store.patch(PLURAL_TO_KIND[plural], name, body, ns)

        def delete_namespaced_custom_object(self, group, ver, ns, plural,
                                            name):
            store.delete(PLURAL_TO_KIND[plural], name, ns)

        def list_namespaced_custom_object(self, group, ver, ns, plural,
                                          label_selector=None):
            labels = dict(p.split("=", 1) for p in
                          label_selector.split(",")) if label_selector \
                else None
            return {"items": store.list(PLURAL_TO_KIND[plural], ns,
                                        labels)}

    class _Obj:
        def __init__(self, d):
            self._d = d

        def to_dict(self):
            return self._d

        @property
        def items(self):
            return [_Obj(i) for i in self._d["items"]]

    class _Resource:
        def __init__(self, kind):
            self.kind = kind

        def server_side_apply(self, body=None, field_manager=None,
                              namespace="default"):
            return _Obj(store.apply(body))

        def get(self, name=None, namespace="default", label_selector=None):
            if name is not None:
                return _Obj(store.get(self.kind, name, namespace))
            labels = dict(p.split("=", 1) for p in
                          label_selector.split(",")) if label_selector \
                else None
            return _Obj({"items": store.list(self.kind, namespace, labels)})

        def patch(self, name=None, namespace="default", body=None,
                  content_type=None):
            return _Obj(store.patch(self.kind, name, body, namespace))

        def delete(self, name=None, namespace="default"):
            store.delete(self.kind, name, namespace)

    class _Resources:
        def get(self, api_version=None, kind=None):
            return _Resource(kind)

    class DynamicClient:
        def __init__(self, api_client):
            self.resources = _Resources()

    class ApiClient:
        def sanitize_for_serialization(self, obj):
            return obj

    class Watch:
        def stream(self, fn, *a, **k):
            time.sleep(0.2)
            return iter(())

    def _mk(name):
        return types.ModuleType(name)

    client_mod = _mk("kubernetes.client")
    client_mod.CustomObjectsApi = CustomObjectsApi
    client_mod.CoreV1Api = type("CoreV1Api", (), {"__getattr__":
        lambda s, n: (lambda *a, **k: {"items": []})})
    client_mod.BatchV1Api = type("BatchV1Api", (), {
        "list_namespaced_job": lambda s, *a, **k: {"items": []}})
    client_mod.AppsV1Api = type("AppsV1Api", (), {
        "list_namespaced_deployment": lambda s, *a, **k: {"items": []}})
    client_mod.ApiClient = ApiClient
    config_mod = _mk("kubernetes.config")
    config_mod.load_kube_config = lambda **k: None
    config_mod.load_incluster_config = lambda: None
    watch_mod = _mk("kubernetes.watch")
    watch_mod.Watch = Watch
    dynamic_mod = _mk("kubernetes.dynamic")
    dynamic_mod.DynamicClient = DynamicClient
    pkg = _mk("kubernetes")
    pkg.client, pkg.config, pkg.watch = client_mod, config_mod, watch_mod
    for name, mod in (("kubernetes", pkg), ("kubernetes.client", client_mod),
                      ("kubernetes.config", config_mod),
                      ("kubernetes.watch", watch_mod),
                      ("kubernetes.dynamic", dynamic_mod)):
        sys.modules[name] = mod


@pytest.fixture()
def realkube():
    saved = {k: sys.modules.get(k) for k in
             ("kubernetes", "kubernetes.client", "kubernetes.config",
              "kubernetes.watch", "kubernetes.dynamic")}
    store = MemKube()
    _install_stateful_fake(store)
    from substratus_amd.kube.real import RealKube
    yield RealKube(kubeconfig="fake"), store
    for k, v in saved.items():
        if v is None:
            sys.modules.pop(k, None)
        else:
            sys.modules[k] = v


@pytest.mark.timeout(120)
def test_model_reconcile_through_realkube(realkube, tmp_path):
    """Full Model flow — create -> params CM -> SA -> modeller Job ->
    fake Job completion -> Ready — with every API call crossing the
    kubernetes-client surface."""
    rk, store = realkube
    cloud = new_cloud({"CLOUD": "kind", "CLUSTER_NAME": "rt",
                       "BUCKET_DIR": str(tmp_path / "bucket")})
    from substratus_amd.controller.manager import Manager
    mgr = Manager(rk, cloud, FakeSCIClient())

    rk.create(apiv1.new_object("Model", "m1", {"image": "loader:v1",
                                               "params": {"name": "x"}}))
    for _ in range(5):
        mgr._enqueue(("Model", "m1", "default"))
        mgr.run_once()
    job = rk.try_get("Job", "m1-modeller")
    assert job is not None, [o["metadata"]["name"]
                             for o in store.list("Job", "default")]
    assert job["spec"]["template"]["spec"]["containers"][0]["image"] == \
        "loader:v1"
    # params ConfigMap + SA went through the dynamic-apply path
    assert rk.try_get("ConfigMap", "m1-model-params") is not None
    assert rk.try_get("ServiceAccount", "modeller") is not None
    # fake the kubelet: Job completes (reference fakeJobComplete)
    rk.patch("Job", "m1-modeller",
             {"status": {"conditions": [{"type": "Complete",
                                         "status": "True"}]}})
    for _ in range(5):
        mgr._enqueue(("Model", "m1", "default"))
        mgr.run_once()
    m = rk.get("Model", "m1")
    assert apiv1.get_status_ready(m), m.get("status")


@pytest.mark.timeout(120)
def test_server_reconcile_through_realkube(realkube, tmp_path):
    rk, store = realkube
    cloud = new_cloud({"CLOUD": "kind", "CLUSTER_NAME": "rt",
                       "BUCKET_DIR": str(tmp_path / "bucket")})
    from substratus_amd.controller.manager import Manager
    mgr = Manager(rk, cloud, FakeSCIClient())
    rk.create(apiv1.new_object("Model", "base", {"image": "x"}))
    for _ in range(3):
        mgr._enqueue(("Model", "base", "default"))
        mgr.run_once()
    rk.patch("Job", "base-modeller",
             {"status": {"conditions": [{"type": "Complete",
                                         "status": "True"}]}})
    for _ in range(3):
        mgr._enqueue(("Model", "base", "default"))
        mgr.run_once()
    rk.create(apiv1.new_object("Server", "s1", {
        "image": "srv:v1", "model": {"name": "base"}}))
    for _ in range(5):
        mgr._enqueue(("Server", "s1", "default"))
        mgr.run_once()
    dep = rk.try_get("Deployment", "s1-server")
    svc = rk.try_get("Service", "s1-server")
    assert dep is not None and svc is not None
    assert svc["spec"]["ports"][0]["port"] == 8080
    # fake readiness
    rk.patch("Deployment", "s1-server", {"status": {"readyReplicas": 1}})
    for _ in range(3):
        mgr._enqueue(("Server", "s1", "default"))
        mgr.run_once()
    assert apiv1.get_status_ready(rk.get("Server", "s1"))


@pytest.mark.timeout(120)
def test_dataset_and_notebook_through_realkube(realkube, tmp_path):
    """Dataset (loader Job -> Ready) and Notebook (pod, suspend reap)
    through the kubernetes-client surface."""
    rk, store = realkube
    cloud = new_cloud({"CLOUD": "kind", "CLUSTER_NAME": "rt",
                       "BUCKET_DIR": str(tmp_path / "bucket")})
    from substratus_amd.controller.manager import Manager
    mgr = Manager(rk, cloud, FakeSCIClient())

    rk.create(apiv1.new_object("Dataset", "d1", {
        "image": "loader:v1", "params": {"urls": "file:///x.jsonl"}}))
    for _ in range(3):
        mgr._enqueue(("Dataset", "d1", "default"))
        mgr.run_once()
    job = rk.try_get("Job", "d1-data-loader")
    assert job is not None and job["spec"]["backoffLimit"] == 2
    rk.patch("Job", "d1-data-loader",
             {"status": {"conditions": [{"type": "Complete",
                                         "status": "True"}]}})
    for _ in range(3):
        mgr._enqueue(("Dataset", "d1", "default"))
        mgr.run_once()
    assert apiv1.get_status_ready(rk.get("Dataset", "d1"))

    rk.create(apiv1.new_object("Notebook", "nb1", {"image": "nb:v1"}))
    for _ in range(3):
        mgr._enqueue(("Notebook", "nb1", "default"))
        mgr.run_once()
    pod = rk.try_get("Pod", "nb1-notebook")
    assert pod is not None
    # suspend reaps the pod (reference notebook_controller suspend path)
    rk.patch("Notebook", "nb1", {"spec": {"suspend": True}})
    for _ in range(3):
        mgr._enqueue(("Notebook", "nb1", "default"))
        mgr.run_once()
    assert rk.try_get("Pod", "nb1-notebook") is None
