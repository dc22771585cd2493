"""sci.v1 protobuf messages, built dynamically (no protoc in this image).

Constructs the same FileDescriptorProto that `protoc` would emit for
sci.proto, so the messages are wire-compatible with the reference's
generated Go stubs (reference internal/sci/sci.pb.go).
"""
from __future__ import annotations

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_FIELD_STRING = descriptor_pb2.FieldDescriptorProto.TYPE_STRING
_FIELD_INT64 = descriptor_pb2.FieldDescriptorProto.TYPE_INT64
_LABEL_OPTIONAL = descriptor_pb2.FieldDescriptorProto.LABEL_OPTIONAL

_MESSAGES = {
    "BindIdentityRequest": [
        ("kubernetes_service_account", 1, _FIELD_STRING),
        ("kubernetes_namespace", 2, _FIELD_STRING),
        ("principal", 3, _FIELD_STRING),
    ],
    "BindIdentityResponse": [],
    "CreateSignedURLRequest": [
        ("bucket_name", 1, _FIELD_STRING),
        ("object_name", 2, _FIELD_STRING),
        ("expiration_seconds", 3, _FIELD_INT64),
        ("md5_checksum", 4, _FIELD_STRING),
    ],
    "CreateSignedURLResponse": [("url", 1, _FIELD_STRING)],
    "GetObjectMd5Request": [
        ("bucket_name", 1, _FIELD_STRING),
        ("object_name", 2, _FIELD_STRING),
    ],
    "GetObjectMd5Response": [("md5_checksum", 1, _FIELD_STRING)],
}


def _build():
    fd = descriptor_pb2.FileDescriptorProto()
    fd.name = "runbooks_amd/sci/sci.proto"
    fd.package = "sci.v1"
    fd.syntax = "proto3"
    for mname, fields in _MESSAGES.items():
        m = fd.message_type.add()
        m.name = mname
        for fname, num, ftype in fields:
            f = m.field.add()
            f.name = fname
            f.number = num
            f.type = ftype
            f.label = _LABEL_OPTIONAL
    pool = descriptor_pool.Default()
    file_desc = pool.Add(fd)
    out = {}
    for mname in _MESSAGES:
        out[mname] = message_factory.GetMessageClass(
            file_desc.message_types_by_name[mname])
    return out


_classes = _build()

BindIdentityRequest = _classes["BindIdentityRequest"]
BindIdentityResponse = _classes["BindIdentityResponse"]
CreateSignedURLRequest = _classes["CreateSignedURLRequest"]
CreateSignedURLResponse = _classes["CreateSignedURLResponse"]
GetObjectMd5Request = _classes["GetObjectMd5Request"]
GetObjectMd5Response = _classes["GetObjectMd5Response"]

SERVICE = "sci.v1.Controller"

# method name → (request class, response class)
METHODS = {
    "CreateSignedURL": (CreateSignedURLRequest, CreateSignedURLResponse),
    "GetObjectMd5": (GetObjectMd5Request, GetObjectMd5Response),
    "BindIdentity": (BindIdentityRequest, BindIdentityResponse),
}
