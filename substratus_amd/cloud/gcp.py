"""GCP cloud: GCS FUSE bucket mounts + workload identity.

Reference internal/cloud/gcp.go:28-148 — metadata-server autoconfig, GCS FUSE
CSI driver volumes with gke-gcsfuse annotations, iam.gke.io/gcp-service-account
workload-identity binding. Metadata calls are gated (no network here); env
vars configure everything in tests.
"""

from __future__ import annotations

from .base import BucketMount, Cloud, CloudError, parse_bucket_url


class GCPCloud(Cloud):
    def __init__(self, env: dict):
        self.env = env
        self.project_id = env.get("GCP_PROJECT_ID", "")
        self.cluster_name = env.get("CLUSTER_NAME", "")
        self.cluster_location = env.get("CLUSTER_LOCATION", "")
        self.artifact_bucket_url = env.get("ARTIFACT_BUCKET_URL", "")
        self.registry_url = env.get("REGISTRY_URL", "")
        self.principal = env.get("PRINCIPAL", "")

    def name(self) -> str:
        return "gcp"

    def auto_configure(self) -> None:
        # Reference gcp.go:28-71 queries the GCE metadata server; offline we
        # derive the same defaults from the project id.
        if self.project_id:
            if not self.artifact_bucket_url:
                self.artifact_bucket_url = f"gs://{self.project_id}-substratus-artifacts"
            if not self.registry_url:
                self.registry_url = (f"{self.cluster_location.split('-')[0] or 'us'}"
                                     f"-docker.pkg.dev/{self.project_id}/substratus")
            if not self.principal:
                self.principal = (f"substratus@{self.project_id}."
                                  f"iam.gserviceaccount.com")

    def associate_principal(self, sa: dict) -> None:
        # workload identity annotation (reference gcp.go:126-140)
        sa.setdefault("metadata", {}).setdefault("annotations", {})[
            "iam.gke.io/gcp-service-account"] = self.principal

    def get_principal(self, sa: dict) -> tuple[str, bool]:
        ann = sa.get("metadata", {}).get("annotations", {})
        bound = ann.get("iam.gke.io/gcp-service-account") == self.principal
        return self.principal, bound

    def mount_bucket(self, pod_metadata: dict, pod_spec: dict, obj: dict,
                    mount: BucketMount) -> None:
        """GCS FUSE CSI volume, SubPath {hash}/artifacts
        (reference internal/cloud/gcp.go:73-124)."""
        url = self.object_artifact_url(obj)
        scheme, bucket, path = parse_bucket_url(url)
        if scheme != "gs":
            raise CloudError(f"expected gs:// url, got {url}")
        pod_metadata.setdefault("annotations", {}).update({
            "gke-gcsfuse/volumes": "true",
            "gke-gcsfuse/cpu-limit": "2",
            "gke-gcsfuse/memory-limit": "800Mi",
            "gke-gcsfuse/ephemeral-storage-limit": "100Gi",
        })
        pod_spec.setdefault("volumes", []).append({
            "name": mount.name,
            "csi": {
                "driver": "gcsfuse.csi.storage.gke.io",
                "volumeAttributes": {
                    "bucketName": bucket,
                    "mountOptions": "implicit-dirs,uid=0,gid=3003",
                },
                "readOnly": mount.read_only,
            },
        })
        for c in pod_spec.get("containers", []):
            c.setdefault("volumeMounts", []).append({
                "name": mount.name,
                "mountPath": mount.mount_path,
                "subPath": f"{path}/artifacts" if path else "artifacts",
                "readOnly": mount.read_only,
            })
