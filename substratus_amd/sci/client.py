"""gRPC SCI client — what the controller-manager dials
(reference cmd/controllermanager/main.go:104-114: insecure channel to
sci.substratus.svc.cluster.local:10080)."""

from __future__ import annotations

from typing import Optional

import grpc

from . import proto


class GrpcSCIClient:
    def __init__(self, address: str = "localhost:10080",
                 timeout: float = 10.0):
        self.channel = grpc.insecure_channel(address)
        self.timeout = timeout
        self._stubs = {}
        for name, (req_cls, resp_cls) in proto.RPCS.items():
            self._stubs[name] = self.channel.unary_unary(
                f"/{proto.SERVICE_NAME}/{name}",
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=resp_cls.FromString)

    def create_signed_url(self, bucket: str, object_name: str,
                          expiration_seconds: int,
                          md5_checksum: str = "") -> str:
        resp = self._stubs["CreateSignedURL"](
            proto.CreateSignedURLRequest(
                bucketName=bucket, objectName=object_name,
                expirationSeconds=expiration_seconds,
                md5Checksum=md5_checksum), timeout=self.timeout)
        return resp.url

    def get_object_md5(self, bucket: str, object_name: str) -> Optional[str]:
        try:
            resp = self._stubs["GetObjectMd5"](
                proto.GetObjectMd5Request(bucketName=bucket,
                                          objectName=object_name),
                timeout=self.timeout)
            return resp.md5Checksum or None
        except grpc.RpcError as e:
            if e.code() == grpc.StatusCode.NOT_FOUND:
                return None
            raise

    def bind_identity(self, ksa: str, namespace: str, principal: str) -> None:
        self._stubs["BindIdentity"](
            proto.BindIdentityRequest(principal=principal,
                                      kubernetesServiceAccount=ksa,
                                      kubernetesNamespace=namespace),
            timeout=self.timeout)
