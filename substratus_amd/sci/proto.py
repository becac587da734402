"""The SCI protobuf schema, built programmatically (no protoc in this image).

Mirrors reference internal/sci/sci.proto:6-38: service Controller with
CreateSignedURL / GetObjectMd5 / BindIdentity. Message classes are generated
at import time from a FileDescriptorProto via google.protobuf's
message_factory — wire-compatible with a protoc-generated sci_pb2.
"""

from __future__ import annotations

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_FD = descriptor_pb2.FileDescriptorProto()
_FD.name = "sci.proto"
_FD.package = "sci"
_FD.syntax = "proto3"

_T_STR = descriptor_pb2.FieldDescriptorProto.TYPE_STRING
_T_U64 = descriptor_pb2.FieldDescriptorProto.TYPE_UINT64


def _msg(name: str, fields: list[tuple[str, int, int]]):
    m = _FD.message_type.add()
    m.name = name
    for fname, num, ftype in fields:
        f = m.field.add()
        f.name = fname
        f.number = num
        f.type = ftype
        f.label = descriptor_pb2.FieldDescriptorProto.LABEL_OPTIONAL
    return m


_msg("CreateSignedURLRequest", [
    ("bucketName", 1, _T_STR),
    ("objectName", 2, _T_STR),
    ("expirationSeconds", 3, _T_U64),
    ("md5Checksum", 4, _T_STR),
])
_msg("CreateSignedURLResponse", [("url", 1, _T_STR)])
_msg("GetObjectMd5Request", [
    ("bucketName", 1, _T_STR),
    ("objectName", 2, _T_STR),
])
_msg("GetObjectMd5Response", [("md5Checksum", 1, _T_STR)])
_msg("BindIdentityRequest", [
    ("principal", 1, _T_STR),
    ("kubernetesServiceAccount", 2, _T_STR),
    ("kubernetesNamespace", 3, _T_STR),
])
_msg("BindIdentityResponse", [])

_pool = descriptor_pool.DescriptorPool()
_file = _pool.Add(_FD)


def _cls(name: str):
    return message_factory.GetMessageClass(_pool.FindMessageTypeByName(f"sci.{name}"))


CreateSignedURLRequest = _cls("CreateSignedURLRequest")
CreateSignedURLResponse = _cls("CreateSignedURLResponse")
GetObjectMd5Request = _cls("GetObjectMd5Request")
GetObjectMd5Response = _cls("GetObjectMd5Response")
BindIdentityRequest = _cls("BindIdentityRequest")
BindIdentityResponse = _cls("BindIdentityResponse")

SERVICE_NAME = "sci.Controller"

RPCS = {
    "CreateSignedURL": (CreateSignedURLRequest, CreateSignedURLResponse),
    "GetObjectMd5": (GetObjectMd5Request, GetObjectMd5Response),
    "BindIdentity": (BindIdentityRequest, BindIdentityResponse),
}
