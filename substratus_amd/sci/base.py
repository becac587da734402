"""SCI — Substratus Cloud Interface.

The reference defines a gRPC `Controller` service with three RPCs
(internal/sci/sci.proto:6-38): CreateSignedURL(bucket, object, expiry, md5) →
url; GetObjectMd5(bucket, object) → md5; BindIdentity(ksa, ns, principal).
One binary per cloud (cmd/sci-gcp | sci-aws | sci-kind).

This module defines the client interface the controllers consume and the fake
used by controller tests (reference internal/sci/fake_sci_client.go:9-22).
The kind implementation (gRPC server + HTTP PUT endpoint over a local bucket
directory) lives in substratus_amd/sci/kind_server.py; GCP/AWS
implementations are credential-gated like the reference's
(internal/sci/gcp/manager_test.go:20-27).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional, Protocol


class SCIClient(Protocol):
    def create_signed_url(self, bucket: str, object_name: str,
                          expiration_seconds: int,
                          md5_checksum: str = "") -> str: ...

    def get_object_md5(self, bucket: str, object_name: str) -> Optional[str]: ...

    def bind_identity(self, ksa: str, namespace: str, principal: str) -> None: ...


@dataclass
class FakeSCIClient:
    """Returns empty responses; records calls (reference fake_sci_client.go)."""
    signed_urls: list = field(default_factory=list)
    md5s: dict = field(default_factory=dict)
    bound: list = field(default_factory=list)
    signed_url_value: str = "http://fake-signed-url"

    def create_signed_url(self, bucket, object_name, expiration_seconds,
                          md5_checksum=""):
        self.signed_urls.append((bucket, object_name, md5_checksum))
        return self.signed_url_value

    def get_object_md5(self, bucket, object_name):
        return self.md5s.get((bucket, object_name))

    def bind_identity(self, ksa, namespace, principal):
        self.bound.append((ksa, namespace, principal))
