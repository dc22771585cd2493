"""gRPC client for sci.v1.Controller + the test fake.

Parity: the generated client the reference dials insecurely from the
controller-manager (reference cmd/controllermanager/main.go:104-114) and
FakeSCIControllerClient (reference internal/sci/fake_sci_client.go:9-21).
"""
from __future__ import annotations

import grpc

from . import proto


class ControllerClient:
    def __init__(self, address: str, channel: grpc.Channel | None = None):
        self._channel = channel or grpc.insecure_channel(address)
        self._stubs = {}
        for method, (req_cls, resp_cls) in proto.METHODS.items():
            self._stubs[method] = self._channel.unary_unary(
                f"/{proto.SERVICE}/{method}",
                request_serializer=req_cls.SerializeToString,
                response_deserializer=resp_cls.FromString,
            )

    def create_signed_url(self, bucket_name: str, object_name: str,
                          expiration_seconds: int = 300,
                          md5_checksum: str = "") -> proto.CreateSignedURLResponse:
        return self._stubs["CreateSignedURL"](proto.CreateSignedURLRequest(
            bucket_name=bucket_name, object_name=object_name,
            expiration_seconds=expiration_seconds,
            md5_checksum=md5_checksum))

    def get_object_md5(self, bucket_name: str,
                       object_name: str) -> proto.GetObjectMd5Response:
        return self._stubs["GetObjectMd5"](proto.GetObjectMd5Request(
            bucket_name=bucket_name, object_name=object_name))

    def bind_identity(self, kubernetes_service_account: str,
                      kubernetes_namespace: str,
                      principal: str = "") -> proto.BindIdentityResponse:
        return self._stubs["BindIdentity"](proto.BindIdentityRequest(
            kubernetes_service_account=kubernetes_service_account,
            kubernetes_namespace=kubernetes_namespace, principal=principal))

    def close(self):
        self._channel.close()


class FakeSCIClient:
    """Returns empty/canned responses (reference fake_sci_client.go:9-21)."""

    def __init__(self, signed_url: str = "", object_md5: str = ""):
        self.signed_url = signed_url
        self.object_md5 = object_md5
        self.calls: list[tuple] = []

    def create_signed_url(self, bucket_name, object_name,
                          expiration_seconds=300, md5_checksum=""):
        self.calls.append(("CreateSignedURL", bucket_name, object_name,
                           md5_checksum))
        return proto.CreateSignedURLResponse(url=self.signed_url)

    def get_object_md5(self, bucket_name, object_name):
        self.calls.append(("GetObjectMd5", bucket_name, object_name))
        return proto.GetObjectMd5Response(md5_checksum=self.object_md5)

    def bind_identity(self, kubernetes_service_account, kubernetes_namespace,
                      principal=""):
        self.calls.append(("BindIdentity", kubernetes_service_account,
                           kubernetes_namespace, principal))
        return proto.BindIdentityResponse()
