"""NotebookReconciler (reference internal/controller/notebook_controller.go:
38-454): suspend → delete Pod; else build gate → notebook SA → gate on
model/dataset refs → {name}-notebook Pod (jupyter on 8888, readiness GET
/api), delete+recreate on immutable-field conflicts; Ready when the Pod is."""

from __future__ import annotations

import copy

from ..api import conditions as cond
from ..api import types as apiv1
from ..cloud.base import BucketMount, Cloud
from ..cloud.resources import apply_resources
from ..kube.memapi import MemKube
from .build import BuildReconciler
from .params import mount_params_configmap, reconcile_params_configmap
from .service_accounts import NOTEBOOK_SA, reconcile_service_account
from .utils import Result, is_pod_ready, owned, param_env, resolve_env

DEFAULT_COMMAND = ["bash", "-c",
                   "notebook.sh || jupyter lab --allow-root --ip=0.0.0.0 "
                   "--NotebookApp.token=$(NOTEBOOK_TOKEN)"]


def nb_pod_name(obj: dict) -> str:
    return f"{obj['metadata']['name']}-notebook"


def is_suspended(obj: dict) -> bool:
    return bool(obj.get("spec", {}).get("suspend", False))


class NotebookReconciler:
    kind = "Notebook"

    def __init__(self, kube: MemKube, cloud: Cloud, sci,
                 build: BuildReconciler):
        self.kube = kube
        self.cloud = cloud
        self.sci = sci
        self.build = build

    def reconcile(self, obj: dict) -> Result:
        ns = obj["metadata"].get("namespace", "default")
        name = nb_pod_name(obj)
        if is_suspended(obj):
            if self.kube.try_get("Pod", name, ns) is not None:
                self.kube.delete("Pod", name, ns)
            apiv1.set_condition(obj, cond.DEPLOYED, False,
                                cond.REASON_SUSPENDED)
            apiv1.set_status_ready(obj, False)
            return Result()
        res = self.build.reconcile(obj)
        if not res.success:
            return res
        reconcile_params_configmap(self.kube, obj)
        reconcile_service_account(self.kube, self.cloud, self.sci,
                                  NOTEBOOK_SA, ns)
        model = dataset = None
        mref = (obj["spec"].get("model") or {}).get("name")
        if mref:
            model = self.kube.try_get("Model", mref, ns)
            if model is None or not apiv1.get_status_ready(model):
                apiv1.set_condition(obj, cond.DEPLOYED, False,
                                    cond.REASON_MODEL_NOT_READY)
                return Result.wait(5)
        dref = (obj["spec"].get("dataset") or {}).get("name")
        if dref:
            dataset = self.kube.try_get("Dataset", dref, ns)
            if dataset is None or not apiv1.get_status_ready(dataset):
                apiv1.set_condition(obj, cond.DEPLOYED, False,
                                    cond.REASON_DATASET_NOT_READY)
                return Result.wait(5)
        pod = self._notebook_pod(obj, model, dataset)
        live = self.kube.try_get("Pod", name, ns)
        if live is None:
            live = self.kube.create(pod)
        elif _immutable_conflict(live, pod):
            # delete + recreate on immutable-field changes
            # (reference notebook_controller.go:131-310)
            self.kube.delete("Pod", name, ns)
            live = self.kube.create(pod)
        ready = is_pod_ready(live)
        apiv1.set_condition(obj, cond.DEPLOYED, ready,
                            cond.REASON_POD_READY if ready
                            else cond.REASON_POD_NOT_READY)
        apiv1.set_status_ready(obj, ready)
        return Result() if ready else Result.wait(3)

    def _notebook_pod(self, obj: dict, model, dataset) -> dict:
        spec = obj["spec"]
        container = {
            "name": "notebook",
            "image": apiv1.get_image(obj),
            "command": list(spec.get("command") or DEFAULT_COMMAND),
            "env": ([{"name": "NOTEBOOK_TOKEN", "value": "default"}] +
                    resolve_env(spec.get("env")) +
                    param_env(apiv1.get_params(obj))),
            "ports": [{"name": "http", "containerPort": 8888}],
            # readiness GET /api (reference notebook_controller.go:320-329)
            "readinessProbe": {"httpGet": {"path": "/api", "port": 8888}},
        }
        pod = {
            "apiVersion": "v1",
            "kind": "Pod",
            "metadata": {
                "name": nb_pod_name(obj),
                "namespace": obj["metadata"].get("namespace", "default"),
                "annotations": {
                    "kubectl.kubernetes.io/default-container": "notebook"},
            },
            "spec": {
                "serviceAccountName": NOTEBOOK_SA,
                "securityContext": {"fsGroup": 3003},
                "containers": [container],
            },
        }
        apply_resources(pod["spec"], apiv1.get_resources(obj),
                        self.cloud.name())
        mount_params_configmap(pod["spec"], obj)
        self.cloud.mount_bucket(pod["metadata"], pod["spec"], obj,
                                BucketMount("artifacts", "/content/artifacts"))
        if model is not None:
            self.cloud.mount_bucket(pod["metadata"], pod["spec"], model,
                                    BucketMount("model", "/content/model",
                                                read_only=True))
        if dataset is not None:
            self.cloud.mount_bucket(pod["metadata"], pod["spec"], dataset,
                                    BucketMount("data", "/content/data",
                                                read_only=True))
        return owned(pod, obj)


def _immutable_conflict(live: dict, want: dict) -> bool:
    a = copy.deepcopy(live.get("spec", {}))
    b = copy.deepcopy(want.get("spec", {}))
    return a != b
