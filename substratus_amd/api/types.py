"""The substratus.ai/v1 CRD API: Model, Dataset, Server, Notebook.

Mirrors the reference API surface (api/v1/model_types.go:10-36, dataset_types
.go:10-28, server_types.go:10-31, notebook_types.go:10-38, common_types.go:
8-111) over unstructured dict objects, plus the generic accessor layer every
reconciler uses (reference api/v1/model_types.go:38-83).

GPU types: the reference enumerated nvidia-a100/t4/l4 (common_types.go:94-100);
this platform adds amd-mi355x as the first-class type and keeps the enum open.
"""

from __future__ import annotations

import copy
import time
from typing import Any, Optional

GROUP = "substratus.ai"
VERSION = "v1"
API_VERSION = f"{GROUP}/{VERSION}"

KINDS = ("Model", "Dataset", "Server", "Notebook")

# GPU type enum (open): reference common_types.go:94-100 + MI355X
GPU_AMD_MI355X = "amd-mi355x"
GPU_NVIDIA_A100 = "nvidia-a100"
GPU_NVIDIA_T4 = "nvidia-t4"
GPU_NVIDIA_L4 = "nvidia-l4"

DEFAULT_CPU = 2        # reference common_types.go Resources defaults
DEFAULT_DISK = 10
DEFAULT_MEMORY = 10


def new_object(kind: str, name: str, spec: Optional[dict] = None,
               namespace: str = "default") -> dict:
    assert kind in KINDS, kind
    return {
        "apiVersion": API_VERSION,
        "kind": kind,
        "metadata": {"name": name, "namespace": namespace},
        "spec": spec or {},
    }


# ----------------------------------------------------------------- accessors
# The generic interface every kind implements in the reference
# (GetBuild/SetImage/GetImage/GetConditions/GetStatusReady/SetStatusReady/
#  GetStatusUpload/SetStatusUpload/GetParams).

def get_build(obj: dict) -> Optional[dict]:
    return obj.get("spec", {}).get("build")


def get_image(obj: dict) -> str:
    return obj.get("spec", {}).get("image", "")


def set_image(obj: dict, image: str) -> None:
    obj.setdefault("spec", {})["image"] = image


def get_params(obj: dict) -> dict:
    return obj.get("spec", {}).get("params", {}) or {}


def get_status_ready(obj: dict) -> bool:
    return bool(obj.get("status", {}).get("ready", False))


def set_status_ready(obj: dict, ready: bool) -> None:
    obj.setdefault("status", {})["ready"] = ready


def get_status_upload(obj: dict) -> dict:
    return obj.get("status", {}).get("buildUpload", {}) or {}


def set_status_upload(obj: dict, upload: dict) -> None:
    obj.setdefault("status", {})["buildUpload"] = upload


def get_conditions(obj: dict) -> list[dict]:
    return obj.setdefault("status", {}).setdefault("conditions", [])


def get_condition(obj: dict, ctype: str) -> Optional[dict]:
    for c in obj.get("status", {}).get("conditions", []) or []:
        if c.get("type") == ctype:
            return c
    return None


def set_condition(obj: dict, ctype: str, status: bool, reason: str = "",
                  message: str = "") -> bool:
    """meta.SetStatusCondition semantics; returns True when changed."""
    conds = get_conditions(obj)
    new_status = "True" if status else "False"
    for c in conds:
        if c.get("type") == ctype:
            changed = (c.get("status") != new_status or
                       c.get("reason") != reason)
            if c.get("status") != new_status:
                c["lastTransitionTime"] = _now()
            c["status"] = new_status
            c["reason"] = reason
            c["message"] = message
            c["observedGeneration"] = obj.get("metadata", {}).get("generation")
            return changed
    conds.append({
        "type": ctype, "status": new_status, "reason": reason,
        "message": message, "lastTransitionTime": _now(),
        "observedGeneration": obj.get("metadata", {}).get("generation"),
    })
    return True


def is_condition_true(obj: dict, ctype: str) -> bool:
    c = get_condition(obj, ctype)
    return bool(c and c.get("status") == "True")


def _now() -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())


# ------------------------------------------------------------------ resources
def get_resources(obj: dict) -> dict:
    """Resources with reference defaults applied (common_types.go:77-92):
    cpu=2, disk=10Gi, memory=10Gi, gpu optional {type, count}."""
    r = copy.deepcopy(obj.get("spec", {}).get("resources") or {})
    r.setdefault("cpu", DEFAULT_CPU)
    r.setdefault("disk", DEFAULT_DISK)
    r.setdefault("memory", DEFAULT_MEMORY)
    return r


def gpu_count(obj: dict) -> int:
    gpu = get_resources(obj).get("gpu") or {}
    return int(gpu.get("count", 0))


# ----------------------------------------------------------------- validation
def validate(obj: dict) -> list[str]:
    """Structural validation mirroring the CRD schemas' required fields."""
    errs: list[str] = []
    kind = obj.get("kind")
    if kind not in KINDS:
        errs.append(f"unknown kind {kind!r}")
        return errs
    if obj.get("apiVersion") != API_VERSION:
        errs.append(f"apiVersion must be {API_VERSION}")
    name = obj.get("metadata", {}).get("name", "")
    if not name:
        errs.append("metadata.name required")
    if len(name) > 50:
        # naming budget: suffixes <=13 chars, parent names <=50
        # (reference docs/design.md "Naming")
        errs.append("metadata.name must be <= 50 characters")
    spec = obj.get("spec", {})
    if kind == "Server" and not (spec.get("model") or {}).get("name"):
        errs.append("Server spec.model.name required")
    if not spec.get("image") and not spec.get("build"):
        if kind != "Notebook":
            errs.append(f"{kind} needs spec.image or spec.build")
    build = spec.get("build") or {}
    if build:
        if bool(build.get("git")) == bool(build.get("upload")):
            errs.append("spec.build needs exactly one of git or upload")
        if build.get("git") and not build["git"].get("url"):
            errs.append("spec.build.git.url required")
    gpu = (spec.get("resources") or {}).get("gpu") or {}
    if gpu and not gpu.get("type"):
        errs.append("spec.resources.gpu.type required when gpu set")
    return errs
