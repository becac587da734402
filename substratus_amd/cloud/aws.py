"""AWS cloud: S3 mountpoint bucket mounts + IRSA identity.

Reference parity: the reference ships an AWS SCI (internal/sci/aws) and eksctl
/karpenter install templates (install/kubernetes/aws/*.tpl); its cloud
abstraction for AWS follows the same Common URL schemes. IRSA =
eks.amazonaws.com/role-arn annotation.
"""

from __future__ import annotations

from .base import BucketMount, Cloud, CloudError, parse_bucket_url


class AWSCloud(Cloud):
    def __init__(self, env: dict):
        self.env = env
        self.account_id = env.get("AWS_ACCOUNT_ID", "")
        self.region = env.get("AWS_REGION", "us-west-2")
        self.cluster_name = env.get("CLUSTER_NAME", "")
        self.artifact_bucket_url = env.get("ARTIFACT_BUCKET_URL", "")
        self.registry_url = env.get("REGISTRY_URL", "")
        self.principal = env.get("PRINCIPAL", "")

    def name(self) -> str:
        return "aws"

    def auto_configure(self) -> None:
        if self.account_id:
            if not self.artifact_bucket_url:
                self.artifact_bucket_url = (
                    f"s3://{self.account_id}-substratus-artifacts")
            if not self.registry_url:
                self.registry_url = (f"{self.account_id}.dkr.ecr."
                                     f"{self.region}.amazonaws.com/substratus")
            if not self.principal:
                self.principal = (f"arn:aws:iam::{self.account_id}:role/"
                                  f"substratus")

    def associate_principal(self, sa: dict) -> None:
        sa.setdefault("metadata", {}).setdefault("annotations", {})[
            "eks.amazonaws.com/role-arn"] = self.principal

    def get_principal(self, sa: dict) -> tuple[str, bool]:
        ann = sa.get("metadata", {}).get("annotations", {})
        return self.principal, ann.get("eks.amazonaws.com/role-arn") == self.principal

    def mount_bucket(self, pod_metadata: dict, pod_spec: dict, obj: dict,
                    mount: BucketMount) -> None:
        url = self.object_artifact_url(obj)
        scheme, bucket, path = parse_bucket_url(url)
        if scheme != "s3":
            raise CloudError(f"expected s3:// url, got {url}")
        pod_spec.setdefault("volumes", []).append({
            "name": mount.name,
            "csi": {
                "driver": "s3.csi.aws.com",
                "volumeAttributes": {
                    "bucketName": bucket,
                    "mountOptions": f"--prefix {path}/artifacts/" if path
                                    else "--prefix artifacts/",
                },
                "readOnly": mount.read_only,
            },
        })
        for c in pod_spec.get("containers", []):
            c.setdefault("volumeMounts", []).append({
                "name": mount.name,
                "mountPath": mount.mount_path,
                "readOnly": mount.read_only,
            })
