"""DatasetReconciler (reference internal/controller/dataset_controller.go:
35-217): build gate → params CM → data-loader SA → {name}-data-loader Job
(backoffLimit 2, artifacts bucket RW) → Ready on completion."""

from __future__ import annotations

from ..api import conditions as cond
from ..api import types as apiv1
from ..cloud.base import BucketMount, Cloud
from ..cloud.resources import apply_resources
from ..kube.memapi import MemKube
from .build import BuildReconciler
from .params import mount_params_configmap, reconcile_params_configmap
from .service_accounts import DATA_LOADER_SA, reconcile_service_account
from .utils import Result, job_result, owned, param_env, reconcile_job, resolve_env


def loader_job_name(obj: dict) -> str:
    return f"{obj['metadata']['name']}-data-loader"


class DatasetReconciler:
    kind = "Dataset"

    def __init__(self, kube: MemKube, cloud: Cloud, sci,
                 build: BuildReconciler):
        self.kube = kube
        self.cloud = cloud
        self.sci = sci
        self.build = build

    def reconcile(self, obj: dict) -> Result:
        ns = obj["metadata"].get("namespace", "default")
        if apiv1.get_status_ready(obj):
            return Result()
        res = self.build.reconcile(obj)
        if not res.success:
            return res
        reconcile_params_configmap(self.kube, obj)
        reconcile_service_account(self.kube, self.cloud, self.sci,
                                  DATA_LOADER_SA, ns)
        job = self._load_job(obj)
        live = reconcile_job(self.kube, job)
        complete, failed = job_result(live)
        if failed:
            apiv1.set_condition(obj, cond.LOADED, False, cond.REASON_JOB_FAILED)
            return Result.wait(30)
        if not complete:
            apiv1.set_condition(obj, cond.LOADED, False,
                                cond.REASON_JOB_NOT_COMPLETE)
            return Result.wait(3)
        apiv1.set_condition(obj, cond.LOADED, True, cond.REASON_JOB_COMPLETE)
        apiv1.set_status_ready(obj, True)
        obj.setdefault("status", {})["artifacts"] = {
            "url": self.cloud.object_artifact_url(obj)}
        return Result()

    def _load_job(self, obj: dict) -> dict:
        spec = obj["spec"]
        container = {
            "name": "loader",
            "image": apiv1.get_image(obj),
            "env": resolve_env(spec.get("env")) + param_env(apiv1.get_params(obj)),
        }
        if spec.get("command"):
            container["command"] = list(spec["command"])
        job = {
            "apiVersion": "batch/v1",
            "kind": "Job",
            "metadata": {
                "name": loader_job_name(obj),
                "namespace": obj["metadata"].get("namespace", "default"),
            },
            "spec": {
                "backoffLimit": 2,  # reference dataset_controller.go:162
                "template": {
                    "metadata": {"annotations": {
                        "kubectl.kubernetes.io/default-container": "loader"}},
                    "spec": {
                        "serviceAccountName": DATA_LOADER_SA,
                        "restartPolicy": "Never",
                        "securityContext": {"fsGroup": 3003},
                        "containers": [container],
                    },
                },
            },
        }
        pod_md = job["spec"]["template"]["metadata"]
        pod_spec = job["spec"]["template"]["spec"]
        apply_resources(pod_spec, apiv1.get_resources(obj), self.cloud.name())
        mount_params_configmap(pod_spec, obj)
        self.cloud.mount_bucket(pod_md, pod_spec, obj,
                                BucketMount("artifacts", "/content/artifacts"))
        return owned(job, obj)
