"""Real-cluster backend: the MemKube interface over the Kubernetes API.

Used when the `kubernetes` Python client is installed and a kubeconfig is
reachable (this offline image has neither — construction raises with a clear
message; the embedded platform or MemKube serve everywhere else). The
controller manager and CLI program against the shared five-method surface
(create/get/update/patch/delete + list/watch), so swapping backends is a
constructor change.
"""

from __future__ import annotations

from typing import Callable, Optional

from ..api import types as apiv1

GROUP_KINDS = {k: f"{k.lower()}s" for k in apiv1.KINDS}
CORE_KINDS = {"Pod": "pods", "Service": "services", "ConfigMap": "configmaps",
              "ServiceAccount": "serviceaccounts", "Secret": "secrets"}
BATCH_KINDS = {"Job": "jobs"}
APPS_KINDS = {"Deployment": "deployments"}


class RealKube:
    def __init__(self, kubeconfig: Optional[str] = None):
        try:
            from kubernetes import client, config, watch  # type: ignore
        except ImportError as e:
            raise RuntimeError(
                "real-cluster mode needs the `kubernetes` Python client "
                "(not installed in this offline image); use the embedded "
                "platform (`sub up`) instead") from e
        if kubeconfig:
            config.load_kube_config(config_file=kubeconfig)
        else:
            try:
                config.load_incluster_config()
            except Exception:
                config.load_kube_config()
        self._client = client
        self._dyn = client.CustomObjectsApi()
        self._core = client.CoreV1Api()
        self._batch = client.BatchV1Api()
        self._apps = client.AppsV1Api()
        self._watch = watch
        self.changes = 0

    # -- routing -----------------------------------------------------------
    def _route(self, kind: str):
        if kind in GROUP_KINDS:
            return ("crd", GROUP_KINDS[kind])
        if kind in CORE_KINDS:
            return ("core", CORE_KINDS[kind])
        if kind in BATCH_KINDS:
            return ("batch", BATCH_KINDS[kind])
        if kind in APPS_KINDS:
            return ("apps", APPS_KINDS[kind])
        raise KeyError(f"unroutable kind {kind}")

    def create(self, obj: dict) -> dict:
        api, plural = self._route(obj["kind"])
        ns = obj["metadata"].get("namespace", "default")
        if api == "crd":
            return self._dyn.create_namespaced_custom_object(
                "substratus.ai", "v1", ns, plural, obj)
        return self._apply_dynamic(obj)

    def apply(self, obj: dict) -> dict:
        return self._apply_dynamic(obj)

    def _apply_dynamic(self, obj: dict) -> dict:
        # server-side apply via the dynamic client (field manager "sub")
        from kubernetes.dynamic import DynamicClient  # type: ignore
        dyn = DynamicClient(self._client.ApiClient())
        resource = dyn.resources.get(api_version=obj["apiVersion"],
                                     kind=obj["kind"])
        return resource.server_side_apply(
            body=obj, field_manager="substratus",
            namespace=obj["metadata"].get("namespace", "default")).to_dict()

    def get(self, kind: str, name: str, namespace: str = "default") -> dict:
        api, plural = self._route(kind)
        if api == "crd":
            return self._dyn.get_namespaced_custom_object(
                "substratus.ai", "v1", namespace, plural, name)
        from kubernetes.dynamic import DynamicClient  # type: ignore
        dyn = DynamicClient(self._client.ApiClient())
        res = dyn.resources.get(kind=kind, api_version=_api_version(kind))
        return res.get(name=name, namespace=namespace).to_dict()

    def try_get(self, kind, name, namespace="default"):
        try:
            return self.get(kind, name, namespace)
        except Exception:
            return None

    def patch(self, kind: str, name: str, patch: dict,
              namespace: str = "default") -> dict:
        api, plural = self._route(kind)
        if api == "crd":
            return self._dyn.patch_namespaced_custom_object(
                "substratus.ai", "v1", namespace, plural, name, patch)
        from kubernetes.dynamic import DynamicClient  # type: ignore
        dyn = DynamicClient(self._client.ApiClient())
        res = dyn.resources.get(kind=kind, api_version=_api_version(kind))
        return res.patch(name=name, namespace=namespace, body=patch,
                         content_type="application/merge-patch+json").to_dict()

    def delete(self, kind: str, name: str, namespace: str = "default") -> None:
        api, plural = self._route(kind)
        if api == "crd":
            self._dyn.delete_namespaced_custom_object(
                "substratus.ai", "v1", namespace, plural, name)
            return
        from kubernetes.dynamic import DynamicClient  # type: ignore
        dyn = DynamicClient(self._client.ApiClient())
        res = dyn.resources.get(kind=kind, api_version=_api_version(kind))
        res.delete(name=name, namespace=namespace)

    def list(self, kind: str, namespace: Optional[str] = "default",
             labels: Optional[dict] = None) -> list[dict]:
        api, plural = self._route(kind)
        sel = ",".join(f"{k}={v}" for k, v in (labels or {}).items()) or None
        if api == "crd":
            out = self._dyn.list_namespaced_custom_object(
                "substratus.ai", "v1", namespace or "default", plural,
                label_selector=sel)
            return out.get("items", [])
        from kubernetes.dynamic import DynamicClient  # type: ignore
        dyn = DynamicClient(self._client.ApiClient())
        res = dyn.resources.get(kind=kind, api_version=_api_version(kind))
        return [i.to_dict() for i in res.get(
            namespace=namespace, label_selector=sel).items]

    # -- watch -------------------------------------------------------------
    def _list_func(self, kind: str, namespace: str):
        """The list callable + positional args the watch stream wraps."""
        api, plural = self._route(kind)
        if api == "crd":
            return (self._dyn.list_namespaced_custom_object,
                    ("substratus.ai", "v1", namespace, plural))
        if api == "core":
            fn = {"Pod": self._core.list_namespaced_pod,
                  "Service": self._core.list_namespaced_service,
                  "ConfigMap": self._core.list_namespaced_config_map,
                  "ServiceAccount":
                      self._core.list_namespaced_service_account,
                  "Secret": self._core.list_namespaced_secret}[kind]
            return (fn, (namespace,))
        if api == "batch":
            return (self._batch.list_namespaced_job, (namespace,))
        return (self._apps.list_namespaced_deployment, (namespace,))

    def watch(self, kind: str, cb: Callable,
              namespace: str = "default") -> Callable[[], None]:
        """Watch EVERY routable kind (CRDs, Pods, Jobs, Deployments, core) —
        the dependent-requeue signal the controllers need (reference
        internal/client/client.go:137-152). Reconnects with backoff,
        resuming from the last resourceVersion; a 410 Gone (expired RV)
        falls back to a fresh list. Returns a stop() callable."""
        import threading
        stop = threading.Event()
        fn, args = self._list_func(kind, namespace)  # route errors raise NOW

        def run():
            backoff = 1.0
            rv = None
            while not stop.is_set():
                try:
                    w = self._watch.Watch()
                    kwargs = {"timeout_seconds": 60}
                    if rv:
                        kwargs["resource_version"] = rv
                    for event in w.stream(fn, *args, **kwargs):
                        if stop.is_set():
                            return
                        obj = event["object"]
                        if not isinstance(obj, dict):  # typed model -> dict
                            obj = self._client.ApiClient()\
                                .sanitize_for_serialization(obj)
                        rv = (obj.get("metadata") or {})\
                            .get("resourceVersion") or rv
                        self.changes += 1
                        cb(event["type"], obj)
                    backoff = 1.0  # clean stream timeout: reconnect at once
                except Exception as e:  # noqa: BLE001
                    if "410" in str(e) or "Expired" in str(e):
                        rv = None  # relist from scratch
                    stop.wait(backoff)
                    backoff = min(backoff * 2, 30.0)

        threading.Thread(target=run, daemon=True).start()
        return stop.set


def _api_version(kind: str) -> str:
    if kind in BATCH_KINDS:
        return "batch/v1"
    if kind in APPS_KINDS:
        return "apps/v1"
    return "v1"
