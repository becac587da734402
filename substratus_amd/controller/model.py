"""ModelReconciler (reference internal/controller/model_controller.go:43-395).

Flow: status.ready short-circuit → image built (BuildReconciler) → params
ConfigMap → modeller ServiceAccount (+BindIdentity) → gate on base-model /
dataset refs → {name}-modeller Job (params CM + artifacts RW + data RO +
model RO mounts, resources applied, FSGroup 3003, backoffLimit 0 for GPU
jobs) → Ready on Job completion.
"""

from __future__ import annotations

from ..api import conditions as cond
from ..api import types as apiv1
from ..cloud.base import BucketMount, Cloud
from ..cloud.resources import apply_resources
from ..kube.memapi import MemKube
from .build import BuildReconciler
from .params import mount_params_configmap, reconcile_params_configmap
from .service_accounts import MODELLER_SA, reconcile_service_account
from .utils import Result, job_result, owned, param_env, reconcile_job, resolve_env


def modeller_job_name(obj: dict) -> str:
    return f"{obj['metadata']['name']}-modeller"  # suffix <=13 (docs/design.md)


class ModelReconciler:
    kind = "Model"

    def __init__(self, kube: MemKube, cloud: Cloud, sci,
                 build: BuildReconciler):
        self.kube = kube
        self.cloud = cloud
        self.sci = sci
        self.build = build

    def reconcile(self, obj: dict) -> Result:
        ns = obj["metadata"].get("namespace", "default")
        if apiv1.get_status_ready(obj):
            # deterministic artifact naming makes Ready durable across
            # recreates (reference model_controller.go:73-75)
            return Result()
        res = self.build.reconcile(obj)
        if not res.success:
            return res
        reconcile_params_configmap(self.kube, obj)
        reconcile_service_account(self.kube, self.cloud, self.sci, MODELLER_SA, ns)

        base_model = None
        ref = (obj["spec"].get("model") or {}).get("name")
        if ref:
            base_model = self.kube.try_get("Model", ref, ns)
            if base_model is None:
                apiv1.set_condition(obj, cond.MODELLED, False,
                                    cond.REASON_DEPENDENCY_NOT_FOUND,
                                    f"base model {ref} not found")
                return Result.wait(5)
            if not apiv1.get_status_ready(base_model):
                apiv1.set_condition(obj, cond.MODELLED, False,
                                    cond.REASON_MODEL_NOT_READY,
                                    f"base model {ref} not ready")
                return Result.wait(5)
        dataset = None
        dref = (obj["spec"].get("dataset") or {}).get("name")
        if dref:
            dataset = self.kube.try_get("Dataset", dref, ns)
            if dataset is None:
                apiv1.set_condition(obj, cond.MODELLED, False,
                                    cond.REASON_DEPENDENCY_NOT_FOUND,
                                    f"dataset {dref} not found")
                return Result.wait(5)
            if not apiv1.get_status_ready(dataset):
                apiv1.set_condition(obj, cond.MODELLED, False,
                                    cond.REASON_DATASET_NOT_READY,
                                    f"dataset {dref} not ready")
                return Result.wait(5)

        job = self._modeller_job(obj, base_model, dataset)
        live = reconcile_job(self.kube, job)
        complete, failed = job_result(live)
        if failed:
            apiv1.set_condition(obj, cond.MODELLED, False,
                                cond.REASON_JOB_FAILED)
            return Result.wait(30)
        if not complete:
            apiv1.set_condition(obj, cond.MODELLED, False,
                                cond.REASON_JOB_NOT_COMPLETE)
            return Result.wait(3)
        apiv1.set_condition(obj, cond.MODELLED, True, cond.REASON_JOB_COMPLETE)
        apiv1.set_status_ready(obj, True)
        artifacts = {"url": self.cloud.object_artifact_url(obj)}
        # loader writes sizeBytes into completed.json; surface it so the
        # Server controller's HBM admission has a real number to check
        completed = self.cloud.read_artifact_json(obj, "completed.json")
        if completed and completed.get("sizeBytes") is not None:
            artifacts["sizeBytes"] = int(completed["sizeBytes"])
        obj.setdefault("status", {})["artifacts"] = artifacts
        return Result()

    def _modeller_job(self, obj: dict, base_model, dataset) -> dict:
        spec = obj["spec"]
        container = {
            "name": "model",
            "image": apiv1.get_image(obj),
            "env": resolve_env(spec.get("env")) + param_env(apiv1.get_params(obj)),
        }
        if spec.get("command"):
            container["command"] = list(spec["command"])
        gpu = apiv1.gpu_count(obj)
        job = {
            "apiVersion": "batch/v1",
            "kind": "Job",
            "metadata": {
                "name": modeller_job_name(obj),
                "namespace": obj["metadata"].get("namespace", "default"),
            },
            "spec": {
                # GPU jobs are not retried (backoffLimit 0); cheap imports
                # (cpu<=3, no gpu) get 2 (reference model_controller.go:294-303)
                "backoffLimit": 0 if gpu > 0 or
                                     apiv1.get_resources(obj)["cpu"] > 3 else 2,
                "template": {
                    "metadata": {"annotations": {
                        "kubectl.kubernetes.io/default-container": "model"}},
                    "spec": {
                        "serviceAccountName": MODELLER_SA,
                        "restartPolicy": "Never",
                        # FSGroup 3003 for bucket-mount write access
                        # (reference model_controller.go:326-328)
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
        if base_model is not None:
            self.cloud.mount_bucket(pod_md, pod_spec, base_model,
                                    BucketMount("model", "/content/model",
                                                read_only=True))
        if dataset is not None:
            self.cloud.mount_bucket(pod_md, pod_spec, dataset,
                                    BucketMount("data", "/content/data",
                                                read_only=True))
        return owned(job, obj)
