"""Params ConfigMap reconciler (reference internal/controller/
params_reconciler.go:28-104): {name}-{kind}-params ConfigMap carrying
params.json, mounted via SubPath at /content/params.json."""

from __future__ import annotations

import json

from ..api import types as apiv1
from ..kube.memapi import MemKube
from .utils import owned


def params_configmap_name(obj: dict) -> str:
    return f"{obj['metadata']['name']}-{obj['kind'].lower()}-params"


def reconcile_params_configmap(kube: MemKube, obj: dict) -> dict:
    params = apiv1.get_params(obj)
    cm = {
        "apiVersion": "v1",
        "kind": "ConfigMap",
        "metadata": {
            "name": params_configmap_name(obj),
            "namespace": obj["metadata"].get("namespace", "default"),
        },
        "data": {"params.json": json.dumps(params, sort_keys=True)},
    }
    owned(cm, obj)
    return kube.apply(cm)


def mount_params_configmap(pod_spec: dict, obj: dict) -> None:
    """SubPath mount at /content/params.json (reference
    params_reconciler.go:78-104)."""
    pod_spec.setdefault("volumes", []).append({
        "name": "params",
        "configMap": {"name": params_configmap_name(obj)},
    })
    for c in pod_spec.get("containers", []):
        c.setdefault("volumeMounts", []).append({
            "name": "params",
            "mountPath": "/content/params.json",
            "subPath": "params.json",
            "readOnly": True,
        })
