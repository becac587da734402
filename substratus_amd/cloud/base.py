"""Cloud abstraction (reference internal/cloud/cloud.go:20-85 + common.go).

The Cloud interface the controllers program against: image URLs, artifact
bucket URLs, identity binding, and bucket mounting. Selected at startup from
the CLOUD env var (reference cloud.go:48-85; the GCE-metadata autodetect is
kept behind the same factory).
"""

from __future__ import annotations

import hashlib
import os
from abc import ABC, abstractmethod
from typing import Optional


class CloudError(RuntimeError):
    pass


class Cloud(ABC):
    """7-method interface (reference internal/cloud/cloud.go:20-46)."""

    # Common env-config (reference internal/cloud/common.go:11-16)
    cluster_name: str = ""
    artifact_bucket_url: str = ""   # e.g. gs://bucket/path, tar:///bucket
    registry_url: str = ""
    principal: str = ""

    @abstractmethod
    def name(self) -> str: ...

    @abstractmethod
    def auto_configure(self) -> None: ...

    def object_built_image_url(self, obj: dict) -> str:
        """Deterministic image URL:
        {registry}/{cluster}-{kind}-{ns}-{name}:{tag} where tag is the git
        tag/branch or the upload MD5, else "latest"
        (reference internal/cloud/common.go:18-43)."""
        md = obj["metadata"]
        kind = obj["kind"].lower()
        tag = "latest"
        build = obj.get("spec", {}).get("build") or {}
        git = build.get("git") or {}
        if git.get("tag"):
            tag = git["tag"]
        elif git.get("branch"):
            tag = git["branch"]
        elif build.get("upload"):
            md5 = (build.get("upload") or {}).get("md5checksum", "")
            if md5:
                tag = md5
        return (f"{self.registry_url}/{self.cluster_name}-{kind}-"
                f"{md['namespace']}-{md['name']}:{tag}")

    def object_artifact_url(self, obj: dict) -> str:
        """{bucket}/{md5(clusters/{cluster}/namespaces/{ns}/{kind}s/{name})}
        — deterministic, name-keyed so delete/recreate resumes from the bucket
        (reference internal/cloud/common.go:45-67; docs/design.md 'Buckets')."""
        md = obj["metadata"]
        path = (f"clusters/{self.cluster_name}/namespaces/{md['namespace']}/"
                f"{obj['kind'].lower()}s/{md['name']}")
        digest = hashlib.md5(path.encode()).hexdigest()
        return f"{self.artifact_bucket_url.rstrip('/')}/{digest}"

    @abstractmethod
    def associate_principal(self, sa: dict) -> None:
        """Annotate a ServiceAccount with the cloud principal."""

    @abstractmethod
    def get_principal(self, sa: dict) -> tuple[str, bool]:
        """Return (principal, bound?) for a ServiceAccount."""

    @abstractmethod
    def mount_bucket(self, pod_metadata: dict, pod_spec: dict, obj: dict,
                     mount: "BucketMount") -> None:
        """Wire the object's artifact bucket into a Pod spec."""

    def read_artifact_json(self, obj: dict, relpath: str) -> Optional[dict]:
        """Read a small JSON artifact (e.g. completed.json) from the
        object's bucket, or None when the cloud can't reach it from the
        controller (GCS/S3 without mounted credentials). The Model
        controller uses this to surface loader-reported metadata
        (sizeBytes) into status."""
        return None


class BucketMount:
    def __init__(self, name: str, mount_path: str, read_only: bool = False):
        self.name = name                # volume name, e.g. "model" | "data"
        self.mount_path = mount_path    # e.g. /content/model
        self.read_only = read_only


def parse_bucket_url(url: str) -> tuple[str, str, str]:
    """scheme://bucket/path → (scheme, bucket, path)
    (reference internal/cloud/utils.go:9-48)."""
    if "://" not in url:
        raise CloudError(f"bucket URL missing scheme: {url!r}")
    scheme, rest = url.split("://", 1)
    if "/" in rest:
        bucket, path = rest.split("/", 1)
    else:
        bucket, path = rest, ""
    return scheme, bucket, path


def new_cloud(env: Optional[dict] = None):
    """Factory (reference internal/cloud/cloud.go:48-85): CLOUD env var
    selects the implementation; validates after auto_configure."""
    from .kind import KindCloud
    from .gcp import GCPCloud
    from .aws import AWSCloud
    env = env if env is not None else dict(os.environ)
    name = env.get("CLOUD", "").lower()
    impls = {"kind": KindCloud, "gcp": GCPCloud, "aws": AWSCloud}
    if name not in impls:
        raise CloudError(
            f"CLOUD env var must be one of {sorted(impls)}, got {name!r}")
    cloud = impls[name](env)
    cloud.auto_configure()
    for field in ("cluster_name", "artifact_bucket_url", "registry_url"):
        if not getattr(cloud, field):
            raise CloudError(f"cloud {name}: {field} not configured")
    return cloud
