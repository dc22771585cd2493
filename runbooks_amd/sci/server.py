"""Generic gRPC service scaffolding shared by the SCI servers.

Registers the three sci.v1.Controller methods on a grpc.Server with the
dynamically-built message classes, plus the standard gRPC health service
the reference exposes on every SCI (reference cmd/sci-kind/main.go:44-47).
"""
from __future__ import annotations

from concurrent import futures

import grpc

from . import proto


class ControllerServicer:
    """Implementations subclass and override the three methods."""

    def CreateSignedURL(self, request, context):
        raise NotImplementedError

    def GetObjectMd5(self, request, context):
        raise NotImplementedError

    def BindIdentity(self, request, context):
        return proto.BindIdentityResponse()


def _handler(servicer, method, req_cls, resp_cls):
    fn = getattr(servicer, method)

    def unary(request, context):
        try:
            return fn(request, context)
        except NotImplementedError:
            context.abort(grpc.StatusCode.UNIMPLEMENTED, method)
        except Exception as e:  # surface as INTERNAL like the Go servers
            context.abort(grpc.StatusCode.INTERNAL, str(e))

    return grpc.unary_unary_rpc_method_handler(
        unary,
        request_deserializer=req_cls.FromString,
        response_serializer=resp_cls.SerializeToString)


def serve(servicer: ControllerServicer, address: str = "0.0.0.0:10080",
          max_workers: int = 8) -> grpc.Server:
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=max_workers))
    handlers = {
        m: _handler(servicer, m, req, resp)
        for m, (req, resp) in proto.METHODS.items()
    }
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(proto.SERVICE, handlers),))
    try:
        from grpc_health.v1 import health, health_pb2_grpc
        health_pb2_grpc.add_HealthServicer_to_server(health.HealthServicer(),
                                                     server)
    except ImportError:
        pass  # health service optional when grpcio-health-checking is absent
    port = server.add_insecure_port(address)
    server.start()
    server.bound_port = port  # ephemeral binds (":0") report the real port
    return server
