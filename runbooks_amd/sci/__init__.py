"""SCI — Substratus Cloud Interface (storage/identity gRPC shim).

Parity: reference internal/sci/ (proto + gcp/aws/kind servers + fake client).
The controller-manager talks to exactly one SCI server over gRPC
(reference cmd/controllermanager/main.go:104-114).
"""
from .client import ControllerClient, FakeSCIClient
from .proto import (
    BindIdentityRequest,
    BindIdentityResponse,
    CreateSignedURLRequest,
    CreateSignedURLResponse,
    GetObjectMd5Request,
    GetObjectMd5Response,
)

__all__ = [
    "ControllerClient", "FakeSCIClient",
    "CreateSignedURLRequest", "CreateSignedURLResponse",
    "GetObjectMd5Request", "GetObjectMd5Response",
    "BindIdentityRequest", "BindIdentityResponse",
]
