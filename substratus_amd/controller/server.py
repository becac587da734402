"""ServerReconciler (reference internal/controller/server_controller.go:
50-341): gate on model Ready → model-server SA → Service (8080→http-serve) +
single-replica Deployment (readiness GET /, model bucket RO at /content/model)
→ Ready when ReadyReplicas>0. MI355X addition: HBM admission — a Server whose
model won't fit gpu.count × 288 GB is rejected with a condition instead of a
Pod (SURVEY.md §2.3 port note)."""

from __future__ import annotations

from ..api import conditions as cond
from ..api import types as apiv1
from ..cloud.base import BucketMount, Cloud
from ..cloud.resources import ResourceError, apply_resources, check_hbm_fit
from ..kube.memapi import MemKube
from .build import BuildReconciler
from .params import mount_params_configmap, reconcile_params_configmap
from .service_accounts import MODEL_SERVER_SA, reconcile_service_account
from .utils import Result, owned, param_env, resolve_env


def server_name(obj: dict) -> str:
    return f"{obj['metadata']['name']}-server"


class ServerReconciler:
    kind = "Server"

    def __init__(self, kube: MemKube, cloud: Cloud, sci,
                 build: BuildReconciler):
        self.kube = kube
        self.cloud = cloud
        self.sci = sci
        self.build = build

    def reconcile(self, obj: dict) -> Result:
        ns = obj["metadata"].get("namespace", "default")
        res = self.build.reconcile(obj)
        if not res.success:
            return res
        mref = (obj["spec"].get("model") or {}).get("name")
        if not mref:
            apiv1.set_condition(obj, cond.DEPLOYED, False, "NoModelRef",
                                "spec.model.name is required")
            return Result.wait(30)
        model = self.kube.try_get("Model", mref, ns)
        if model is None:
            apiv1.set_condition(obj, cond.DEPLOYED, False,
                                cond.REASON_DEPENDENCY_NOT_FOUND,
                                f"model {mref} not found")
            return Result.wait(5)
        if not apiv1.get_status_ready(model):
            apiv1.set_condition(obj, cond.DEPLOYED, False,
                                cond.REASON_MODEL_NOT_READY)
            return Result.wait(5)
        # MI355X HBM sizing admission
        gpu = apiv1.get_resources(obj).get("gpu") or {}
        model_bytes = int((model.get("status", {}).get("artifacts") or {})
                          .get("sizeBytes", 0))
        try:
            check_hbm_fit(gpu.get("type", ""), int(gpu.get("count", 0)),
                          model_bytes)
        except ResourceError as e:
            apiv1.set_condition(obj, cond.DEPLOYED, False, "InsufficientHBM",
                                str(e))
            apiv1.set_status_ready(obj, False)
            return Result.wait(60)

        reconcile_params_configmap(self.kube, obj)
        reconcile_service_account(self.kube, self.cloud, self.sci,
                                  MODEL_SERVER_SA, ns)
        self.kube.apply(self._service(obj))
        dep = self.kube.apply(self._deployment(obj, model))
        ready = (dep.get("status", {}).get("readyReplicas", 0) or 0) > 0
        apiv1.set_condition(obj, cond.DEPLOYED, ready,
                            cond.REASON_DEPLOYMENT_READY if ready
                            else cond.REASON_DEPLOYMENT_NOT_READY)
        apiv1.set_status_ready(obj, ready)
        return Result() if ready else Result.wait(3)

    def _service(self, obj: dict) -> dict:
        # 8080 → http-serve (reference server_controller.go:307-335)
        svc = {
            "apiVersion": "v1",
            "kind": "Service",
            "metadata": {
                "name": server_name(obj),
                "namespace": obj["metadata"].get("namespace", "default"),
            },
            "spec": {
                "selector": {"app": server_name(obj)},
                "ports": [{"name": "http-serve", "port": 8080,
                           "targetPort": "http-serve"}],
            },
        }
        return owned(svc, obj)

    def _deployment(self, obj: dict, model: dict) -> dict:
        spec = obj["spec"]
        container = {
            "name": "serve",
            "image": apiv1.get_image(obj),
            "env": resolve_env(spec.get("env")) + param_env(apiv1.get_params(obj)),
            "ports": [{"name": "http-serve", "containerPort": 8080}],
            # container contract: ready == GET / → 200 on 8080
            # (reference docs/container-contract.md:50-55)
            "readinessProbe": {
                "httpGet": {"path": "/", "port": "http-serve"},
                "periodSeconds": 3,
            },
        }
        if spec.get("command"):
            container["command"] = list(spec["command"])
        dep = {
            "apiVersion": "apps/v1",
            "kind": "Deployment",
            "metadata": {
                "name": server_name(obj),
                "namespace": obj["metadata"].get("namespace", "default"),
            },
            "spec": {
                "replicas": 1,  # reference server_controller.go:115
                "selector": {"matchLabels": {"app": server_name(obj)}},
                "template": {
                    "metadata": {
                        "labels": {"app": server_name(obj)},
                        "annotations": {
                            "kubectl.kubernetes.io/default-container": "serve"},
                    },
                    "spec": {
                        "serviceAccountName": MODEL_SERVER_SA,
                        "containers": [container],
                    },
                },
            },
        }
        pod_md = dep["spec"]["template"]["metadata"]
        pod_spec = dep["spec"]["template"]["spec"]
        apply_resources(pod_spec, apiv1.get_resources(obj), self.cloud.name())
        mount_params_configmap(pod_spec, obj)
        self.cloud.mount_bucket(pod_md, pod_spec, model,
                                BucketMount("model", "/content/model",
                                            read_only=True))
        return owned(dep, obj)
