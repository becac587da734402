"""Kind (local) cloud: hostPath bucket + local registry, no identity.

Reference internal/cloud/kind.go:23-94 — bucket = hostPath /bucket with URL
scheme tar:///bucket (kaniko-compatible), registry discovered from the
REGISTRY_PORT_5000_TCP_ADDR service env, BindIdentity is a no-op. This is the
fake-backend path every local/system test runs on.
"""

from __future__ import annotations

import os

from .base import BucketMount, Cloud


class KindCloud(Cloud):
    def __init__(self, env: dict):
        self.env = env
        self.cluster_name = env.get("CLUSTER_NAME", "substratus")
        self.artifact_bucket_url = env.get("ARTIFACT_BUCKET_URL", "")
        self.registry_url = env.get("REGISTRY_URL", "")
        self.principal = ""
        self.bucket_dir = env.get("BUCKET_DIR", "/bucket")

    def name(self) -> str:
        return "kind"

    def auto_configure(self) -> None:
        if not self.artifact_bucket_url:
            self.artifact_bucket_url = f"tar://{self.bucket_dir}"
        if not self.registry_url:
            addr = self.env.get("REGISTRY_PORT_5000_TCP_ADDR", "registry")
            self.registry_url = f"{addr}:5000"

    def associate_principal(self, sa: dict) -> None:
        pass  # no identity on kind (reference kind.go:92-94)

    def get_principal(self, sa: dict) -> tuple[str, bool]:
        return "", True

    def read_artifact_json(self, obj: dict, relpath: str):
        """hostPath bucket is directly readable by the controller."""
        import json
        url = self.object_artifact_url(obj)
        digest = url.rsplit("/", 1)[-1]
        path = os.path.join(self.bucket_dir, digest, "artifacts", relpath)
        try:
            with open(path) as f:
                return json.load(f)
        except (OSError, ValueError):
            return None

    def mount_bucket(self, pod_metadata: dict, pod_spec: dict, obj: dict,
                    mount: BucketMount) -> None:
        """hostPath volume at {bucket_dir}/{hash}/artifacts
        (reference internal/cloud/kind.go:50-90)."""
        url = self.object_artifact_url(obj)
        digest = url.rsplit("/", 1)[-1]
        vol = {
            "name": mount.name,
            "hostPath": {
                "path": f"{self.bucket_dir}/{digest}/artifacts",
                "type": "DirectoryOrCreate",
            },
        }
        pod_spec.setdefault("volumes", []).append(vol)
        for c in pod_spec.get("containers", []):
            c.setdefault("volumeMounts", []).append({
                "name": mount.name,
                "mountPath": mount.mount_path,
                "readOnly": mount.read_only,
            })
