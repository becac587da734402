"""Shared reconciler utilities (reference internal/controller/utils.go:17-93)."""

from __future__ import annotations

import re
from dataclasses import dataclass, field
from typing import Optional

from ..kube.memapi import MemKube


@dataclass
class Result:
    """reconcile outcome (reference utils.go:17-21): complete when success,
    otherwise requeue (optionally after a delay)."""
    success: bool = True
    requeue_after: float = 0.0

    @staticmethod
    def wait(after: float = 3.0) -> "Result":
        return Result(success=False, requeue_after=after)


def owned(child: dict, owner: dict) -> dict:
    child.setdefault("metadata", {}).setdefault("ownerReferences", []).append({
        "apiVersion": owner["apiVersion"],
        "kind": owner["kind"],
        "name": owner["metadata"]["name"],
        "uid": owner["metadata"].get("uid", ""),
        "controller": True,
    })
    return child


def reconcile_job(kube: MemKube, job: dict) -> dict:
    """Create-if-absent, then return the live Job (Jobs are immutable —
    reference utils.go:23-35)."""
    ns = job["metadata"].get("namespace", "default")
    cur = kube.try_get("Job", job["metadata"]["name"], ns)
    if cur is None:
        return kube.create(job)
    return cur


def job_result(job: dict) -> tuple[bool, bool]:
    """(complete, failed) from Job conditions (reference utils.go:37-49)."""
    complete = failed = False
    for c in job.get("status", {}).get("conditions", []) or []:
        if c.get("type") == "Complete" and c.get("status") == "True":
            complete = True
        if c.get("type") == "Failed" and c.get("status") == "True":
            failed = True
    return complete, failed


def is_pod_ready(pod: dict) -> bool:
    """reference utils.go:51-65."""
    for c in pod.get("status", {}).get("conditions", []) or []:
        if c.get("type") == "Ready":
            return c.get("status") == "True"
    return False


_SECRET_RE = re.compile(r"^\$\{\{\s*secrets\.([^.\s]+)\.([^.\s}]+)\s*\}\}$")


def resolve_env(env: Optional[dict]) -> list[dict]:
    """spec.env map → container env list, resolving GitHub-Actions-style
    secret refs `${{ secrets.name.key }}` → SecretKeyRef
    (reference internal/controller/utils.go:67-93)."""
    out = []
    for k, v in sorted((env or {}).items()):
        m = _SECRET_RE.match(str(v))
        if m:
            out.append({"name": k, "valueFrom": {"secretKeyRef": {
                "name": m.group(1), "key": m.group(2)}}})
        else:
            out.append({"name": k, "value": str(v)})
    return out


def param_env(params: dict) -> list[dict]:
    """params map → PARAM_{UPPER(key)} env vars
    (reference docs/container-contract.md:34-48)."""
    return [{"name": f"PARAM_{re.sub(r'[^A-Z0-9]', '_', str(k).upper())}",
             "value": str(v)} for k, v in sorted(params.items())]
