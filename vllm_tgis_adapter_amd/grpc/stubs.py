"""Hand-written gRPC service plumbing for fmaas.GenerationService and health.

The reference generates these with grpcio-tools (generation_pb2_grpc.py);
grpc_tools is unavailable here, so the method handlers / client stubs are
written directly against the grpcio generic API.  Wire behavior is identical:
method paths, serializers and streaming arity match the reference service
definition (reference: grpc/pb/generation.proto:9-18).
"""

from __future__ import annotations

import grpc

from . import proto


def _unary(handler, request_cls):
    return grpc.unary_unary_rpc_method_handler(
        handler,
        request_deserializer=request_cls.FromString,
        response_serializer=lambda m: m.SerializeToString(),
    )


def _server_stream(handler, request_cls):
    return grpc.unary_stream_rpc_method_handler(
        handler,
        request_deserializer=request_cls.FromString,
        response_serializer=lambda m: m.SerializeToString(),
    )


def add_generation_service(server: grpc.aio.Server | grpc.Server, servicer) -> None:
    handlers = {
        "Generate": _unary(servicer.Generate, proto.BatchedGenerationRequest),
        "GenerateStream": _server_stream(servicer.GenerateStream, proto.SingleGenerationRequest),
        "Tokenize": _unary(servicer.Tokenize, proto.BatchedTokenizeRequest),
        "ModelInfo": _unary(servicer.ModelInfo, proto.ModelInfoRequest),
    }
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(proto.SERVICE_NAME, handlers),)
    )


def add_health_service(server, servicer) -> None:
    handlers = {
        "Check": _unary(servicer.Check, proto.HealthCheckRequest),
        "Watch": _server_stream(servicer.Watch, proto.HealthCheckRequest),
    }
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(proto.HEALTH_SERVICE_NAME, handlers),)
    )


class GenerationStub:
    """Client stub for fmaas.GenerationService (sync or aio channel)."""

    def __init__(self, channel):
        p = "/" + proto.SERVICE_NAME + "/"
        self.Generate = channel.unary_unary(
            p + "Generate",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=proto.BatchedGenerationResponse.FromString,
        )
        self.GenerateStream = channel.unary_stream(
            p + "GenerateStream",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=proto.GenerationResponse.FromString,
        )
        self.Tokenize = channel.unary_unary(
            p + "Tokenize",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=proto.BatchedTokenizeResponse.FromString,
        )
        self.ModelInfo = channel.unary_unary(
            p + "ModelInfo",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=proto.ModelInfoResponse.FromString,
        )


class HealthStub:
    def __init__(self, channel):
        p = "/" + proto.HEALTH_SERVICE_NAME + "/"
        self.Check = channel.unary_unary(
            p + "Check",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=proto.HealthCheckResponse.FromString,
        )


class HealthServicer:
    """Minimal async health servicer (grpc.health.v1 semantics)."""

    def __init__(self):
        self._status: dict[str, int] = {"": 1}  # SERVING by default for ""

    def set(self, service: str, status: int) -> None:
        self._status[service] = status

    async def Check(self, request, context):
        st = self._status.get(request.service)
        if st is None:
            await context.abort(grpc.StatusCode.NOT_FOUND, "unknown service")
        return proto.HealthCheckResponse(status=st)

    async def Watch(self, request, context):
        # Single snapshot then hold; TGIS probes use Check, Watch is best-effort.
        st = self._status.get(request.service)
        if st is None:
            st = proto.HealthCheckResponse.ServingStatus.SERVICE_UNKNOWN
        yield proto.HealthCheckResponse(status=st)
