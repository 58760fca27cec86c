"""Minimal gRPC server-reflection (v1alpha) service.

grpcio-reflection is not installed in this environment, so the reflection
wire schema is declared with the same runtime-descriptor DSL as the fmaas
schema and the servicer implements the subset clients actually use
(list_services, file_by_filename, file_containing_symbol).
"""

from __future__ import annotations

import grpc

from .proto import F, Msg, build_file, message_classes, serialized_file_descriptors

_PKG = "grpc.reflection.v1alpha"

_MESSAGES = [
    Msg("ServerReflectionRequest", [
        F("host", 1, "string"),
        F("file_by_filename", 3, "string", oneof="message_request"),
        F("file_containing_symbol", 4, "string", oneof="message_request"),
        F("file_containing_extension", 5, f".{_PKG}.ExtensionRequest", oneof="message_request"),
        F("all_extension_numbers_of_type", 6, "string", oneof="message_request"),
        F("list_services", 7, "string", oneof="message_request"),
    ]),
    Msg("ExtensionRequest", [
        F("containing_type", 1, "string"),
        F("extension_number", 2, "int32"),
    ]),
    Msg("ServerReflectionResponse", [
        F("valid_host", 1, "string"),
        F("original_request", 2, f".{_PKG}.ServerReflectionRequest"),
        F("file_descriptor_response", 4, f".{_PKG}.FileDescriptorResponse", oneof="message_response"),
        F("all_extension_numbers_response", 5, f".{_PKG}.ExtensionNumberResponse", oneof="message_response"),
        F("list_services_response", 6, f".{_PKG}.ListServiceResponse", oneof="message_response"),
        F("error_response", 7, f".{_PKG}.ErrorResponse", oneof="message_response"),
    ]),
    Msg("FileDescriptorResponse", [
        F("file_descriptor_proto", 1, "bytes", "repeated"),
    ]),
    Msg("ExtensionNumberResponse", [
        F("base_type_name", 1, "string"),
        F("extension_number", 2, "int32", "repeated"),
    ]),
    Msg("ListServiceResponse", [
        F("service", 1, f".{_PKG}.ServiceResponse", "repeated"),
    ]),
    Msg("ServiceResponse", [
        F("name", 1, "string"),
    ]),
    Msg("ErrorResponse", [
        F("error_code", 1, "int32"),
        F("error_message", 2, "string"),
    ]),
]

_file = build_file("grpc/reflection/v1alpha/reflection.proto", _PKG, _MESSAGES)
_cls = message_classes(_file)

ServerReflectionRequest = _cls["ServerReflectionRequest"]
ServerReflectionResponse = _cls["ServerReflectionResponse"]

SERVICE_NAME = f"{_PKG}.ServerReflection"


class ReflectionServicer:
    def __init__(self, service_names: list[str]):
        self.service_names = list(service_names)
        self._files = serialized_file_descriptors()
        # symbol -> file name (coarse: prefix match on package/service)
        self._symbol_file = {}
        for name in self._files:
            if name.startswith("fmaas"):
                self._symbol_file["fmaas"] = name
            if "health" in name:
                self._symbol_file["grpc.health.v1"] = name

    def _respond(self, request):
        resp = ServerReflectionResponse(valid_host=request.host)
        resp.original_request.CopyFrom(request)
        which = request.WhichOneof("message_request")
        if which == "list_services":
            for s in self.service_names:
                resp.list_services_response.service.add().name = s
        elif which == "file_by_filename":
            data = self._files.get(request.file_by_filename)
            if data is None:
                resp.error_response.error_code = 5  # NOT_FOUND
                resp.error_response.error_message = "file not found"
            else:
                resp.file_descriptor_response.file_descriptor_proto.append(data)
        elif which == "file_containing_symbol":
            sym = request.file_containing_symbol
            fn = None
            for prefix, name in self._symbol_file.items():
                if sym.startswith(prefix):
                    fn = name
                    break
            if fn is None:
                resp.error_response.error_code = 5
                resp.error_response.error_message = f"symbol not found: {sym}"
            else:
                resp.file_descriptor_response.file_descriptor_proto.append(self._files[fn])
        else:
            resp.error_response.error_code = 12  # UNIMPLEMENTED
            resp.error_response.error_message = "not implemented"
        return resp

    async def ServerReflectionInfo(self, request_iterator, context):
        async for request in request_iterator:
            yield self._respond(request)


def enable_server_reflection(service_names: list[str], server) -> None:
    servicer = ReflectionServicer(service_names)
    handler = grpc.stream_stream_rpc_method_handler(
        servicer.ServerReflectionInfo,
        request_deserializer=ServerReflectionRequest.FromString,
        response_serializer=lambda m: m.SerializeToString(),
    )
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(SERVICE_NAME, {"ServerReflectionInfo": handler}),)
    )
