"""Runtime-constructed protobuf messages for the TGIS ``fmaas`` wire API.

The reference project compiles ``generation.proto`` with protoc at build time
(reference: setup.py:10-40, grpc/pb/generation.proto).  This environment has no
``grpc_tools``/protoc, so we build the identical wire schema at import time
from hand-written :class:`descriptor_pb2.FileDescriptorProto` definitions and
materialise message classes with ``google.protobuf.message_factory``.  The
message names, field names, field numbers, types and oneofs below ARE the wire
contract (reference: grpc/pb/generation.proto:1-279) — they must not drift.

Also defines the standard ``grpc.health.v1`` schema (the reference pulls it in
via grpcio-health-checking, absent here) so the health service and the
``grpc_healthcheck`` CLI work identically.
"""

from __future__ import annotations

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

# ---------------------------------------------------------------------------
# Tiny declarative layer over descriptor_pb2
# ---------------------------------------------------------------------------

_T = descriptor_pb2.FieldDescriptorProto

_SCALAR = {
    "string": _T.TYPE_STRING,
    "uint32": _T.TYPE_UINT32,
    "uint64": _T.TYPE_UINT64,
    "int32": _T.TYPE_INT32,
    "int64": _T.TYPE_INT64,
    "float": _T.TYPE_FLOAT,
    "double": _T.TYPE_DOUBLE,
    "bool": _T.TYPE_BOOL,
    "bytes": _T.TYPE_BYTES,
}


class F:
    """One message field.

    kind: scalar name from _SCALAR, or ".pkg.Message" / ".pkg.Enum" type name.
    flags: "repeated", "optional" (proto3 explicit presence), or "" (singular).
    oneof: name of a real oneof group this field belongs to.
    """

    def __init__(self, name: str, number: int, kind: str, flags: str = "", oneof: str | None = None):
        self.name = name
        self.number = number
        self.kind = kind
        self.flags = flags
        self.oneof = oneof


class Enum:
    def __init__(self, name: str, values: dict[str, int]):
        self.name = name
        self.values = values


class Msg:
    def __init__(self, name: str, fields: list[F], nested: list["Msg | Enum"] | None = None):
        self.name = name
        self.fields = fields
        self.nested = nested or []


def _fill_enum(ed: descriptor_pb2.EnumDescriptorProto, e: Enum) -> None:
    ed.name = e.name
    for vname, vnum in e.values.items():
        v = ed.value.add()
        v.name = vname
        v.number = vnum


def _fill_msg(md: descriptor_pb2.DescriptorProto, m: Msg) -> None:
    md.name = m.name
    real_oneofs: list[str] = []
    for f in m.fields:
        if f.oneof and f.oneof not in real_oneofs:
            real_oneofs.append(f.oneof)
    for name in real_oneofs:
        md.oneof_decl.add().name = name
    synth_start = len(real_oneofs)
    synth: list[str] = []
    for f in m.fields:
        fd = md.field.add()
        fd.name = f.name
        fd.number = f.number
        if f.kind in _SCALAR:
            fd.type = _SCALAR[f.kind]
        else:
            fd.type_name = f.kind
            # Enum vs message resolved by the pool from type_name; we must set
            # type explicitly, so the convention is: enum kinds end in "!".
            if f.kind.endswith("!"):
                fd.type_name = f.kind[:-1]
                fd.type = _T.TYPE_ENUM
            else:
                fd.type = _T.TYPE_MESSAGE
        if f.flags == "repeated":
            fd.label = _T.LABEL_REPEATED
        else:
            fd.label = _T.LABEL_OPTIONAL
        if f.oneof:
            fd.oneof_index = real_oneofs.index(f.oneof)
        elif f.flags == "optional":
            fd.proto3_optional = True
            fd.oneof_index = synth_start + len(synth)
            synth.append("_" + f.name)
    for name in synth:
        md.oneof_decl.add().name = name
    for n in m.nested:
        if isinstance(n, Enum):
            _fill_enum(md.enum_type.add(), n)
        else:
            _fill_msg(md.nested_type.add(), n)


def build_file(
    name: str,
    package: str,
    messages: list[Msg],
    enums: list[Enum] | None = None,
    pool: descriptor_pool.DescriptorPool | None = None,
):
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = name
    fdp.package = package
    fdp.syntax = "proto3"
    for e in enums or []:
        _fill_enum(fdp.enum_type.add(), e)
    for m in messages:
        _fill_msg(fdp.message_type.add(), m)
    pool = pool or descriptor_pool.Default()
    return pool.Add(fdp)


def message_classes(file_desc) -> dict[str, type]:
    out = {}
    for name, md in file_desc.message_types_by_name.items():
        out[name] = message_factory.GetMessageClass(md)
    return out


# ---------------------------------------------------------------------------
# fmaas/generation wire schema (reference: grpc/pb/generation.proto)
# ---------------------------------------------------------------------------

_GENERATION_MESSAGES = [
    Msg("BatchedGenerationRequest", [
        F("model_id", 1, "string"),
        F("prefix_id", 2, "string", "optional"),
        F("requests", 3, ".fmaas.GenerationRequest", "repeated"),
        F("adapter_id", 4, "string", "optional"),
        F("params", 10, ".fmaas.Parameters"),
    ]),
    Msg("SingleGenerationRequest", [
        F("model_id", 1, "string"),
        F("prefix_id", 2, "string", "optional"),
        F("request", 3, ".fmaas.GenerationRequest"),
        F("adapter_id", 4, "string", "optional"),
        F("params", 10, ".fmaas.Parameters"),
    ]),
    Msg("BatchedGenerationResponse", [
        F("responses", 1, ".fmaas.GenerationResponse", "repeated"),
    ]),
    Msg("GenerationRequest", [
        F("text", 2, "string"),
    ]),
    Msg("GenerationResponse", [
        F("generated_token_count", 2, "uint32"),
        F("text", 4, "string"),
        F("input_token_count", 6, "uint32"),
        F("stop_reason", 7, ".fmaas.StopReason!"),
        F("tokens", 8, ".fmaas.TokenInfo", "repeated"),
        F("input_tokens", 9, ".fmaas.TokenInfo", "repeated"),
        F("seed", 10, "uint64"),
        F("stop_sequence", 11, "string"),
    ]),
    Msg("Parameters", [
        F("method", 1, ".fmaas.DecodingMethod!"),
        F("sampling", 2, ".fmaas.SamplingParameters"),
        F("stopping", 3, ".fmaas.StoppingCriteria"),
        F("response", 4, ".fmaas.ResponseOptions"),
        F("decoding", 5, ".fmaas.DecodingParameters"),
        F("truncate_input_tokens", 6, "uint32"),
    ]),
    Msg("DecodingParameters", [
        F("repetition_penalty", 1, "float"),
        F("length_penalty", 2, ".fmaas.DecodingParameters.LengthPenalty", "optional"),
        F("format", 3, ".fmaas.DecodingParameters.ResponseFormat!", oneof="guided"),
        F("json_schema", 4, "string", oneof="guided"),
        F("regex", 5, "string", oneof="guided"),
        F("choice", 6, ".fmaas.DecodingParameters.StringChoices", oneof="guided"),
        F("grammar", 7, "string", oneof="guided"),
    ], nested=[
        Msg("LengthPenalty", [
            F("start_index", 1, "uint32"),
            F("decay_factor", 2, "float"),
        ]),
        Msg("StringChoices", [
            F("choices", 1, "string", "repeated"),
        ]),
        Enum("ResponseFormat", {"TEXT": 0, "JSON": 1}),
    ]),
    Msg("SamplingParameters", [
        F("temperature", 1, "float", "optional"),
        F("top_k", 2, "uint32"),
        F("top_p", 3, "float"),
        F("typical_p", 4, "float"),
        F("seed", 5, "uint64", "optional"),
    ]),
    Msg("StoppingCriteria", [
        F("max_new_tokens", 1, "uint32"),
        F("min_new_tokens", 2, "uint32"),
        F("time_limit_millis", 3, "uint32"),
        F("stop_sequences", 4, "string", "repeated"),
        F("include_stop_sequence", 5, "bool", "optional"),
    ]),
    Msg("ResponseOptions", [
        F("input_text", 1, "bool"),
        F("generated_tokens", 2, "bool"),
        F("input_tokens", 3, "bool"),
        F("token_logprobs", 4, "bool"),
        F("token_ranks", 5, "bool"),
        F("top_n_tokens", 6, "uint32"),
    ]),
    Msg("TokenInfo", [
        F("text", 2, "string"),
        F("logprob", 3, "float"),
        F("rank", 4, "uint32"),
        F("top_tokens", 5, ".fmaas.TokenInfo.TopToken", "repeated"),
    ], nested=[
        Msg("TopToken", [
            F("text", 2, "string"),
            F("logprob", 3, "float"),
        ]),
    ]),
    Msg("BatchedTokenizeRequest", [
        F("model_id", 1, "string"),
        F("requests", 2, ".fmaas.TokenizeRequest", "repeated"),
        F("return_tokens", 3, "bool"),
        F("return_offsets", 4, "bool"),
        F("truncate_input_tokens", 5, "uint32"),
        F("prefix_id", 6, "string", "optional"),
        F("adapter_id", 7, "string", "optional"),
    ]),
    Msg("BatchedTokenizeResponse", [
        F("responses", 1, ".fmaas.TokenizeResponse", "repeated"),
    ]),
    Msg("TokenizeRequest", [
        F("text", 1, "string"),
    ]),
    Msg("TokenizeResponse", [
        F("token_count", 1, "uint32"),
        F("tokens", 2, "string", "repeated"),
        F("offsets", 3, ".fmaas.TokenizeResponse.Offset", "repeated"),
    ], nested=[
        Msg("Offset", [
            F("start", 1, "uint32"),
            F("end", 2, "uint32"),
        ]),
    ]),
    Msg("ModelInfoRequest", [
        F("model_id", 1, "string"),
    ]),
    Msg("ModelInfoResponse", [
        F("model_kind", 1, ".fmaas.ModelInfoResponse.ModelKind!"),
        F("max_sequence_length", 2, "uint32"),
        F("max_new_tokens", 3, "uint32"),
    ], nested=[
        Enum("ModelKind", {"DECODER_ONLY": 0, "ENCODER_DECODER": 1}),
    ]),
]

_GENERATION_ENUMS = [
    Enum("DecodingMethod", {"GREEDY": 0, "SAMPLE": 1}),
    Enum("StopReason", {
        "NOT_FINISHED": 0,
        "MAX_TOKENS": 1,
        "EOS_TOKEN": 2,
        "CANCELLED": 3,
        "TIME_LIMIT": 4,
        "STOP_SEQUENCE": 5,
        "TOKEN_LIMIT": 6,
        "ERROR": 7,
    }),
]

_gen_file = build_file("fmaas/generation.proto", "fmaas", _GENERATION_MESSAGES, _GENERATION_ENUMS)
_gen_cls = message_classes(_gen_file)

BatchedGenerationRequest = _gen_cls["BatchedGenerationRequest"]
SingleGenerationRequest = _gen_cls["SingleGenerationRequest"]
BatchedGenerationResponse = _gen_cls["BatchedGenerationResponse"]
GenerationRequest = _gen_cls["GenerationRequest"]
GenerationResponse = _gen_cls["GenerationResponse"]
Parameters = _gen_cls["Parameters"]
DecodingParameters = _gen_cls["DecodingParameters"]
SamplingParameters = _gen_cls["SamplingParameters"]
StoppingCriteria = _gen_cls["StoppingCriteria"]
ResponseOptions = _gen_cls["ResponseOptions"]
TokenInfo = _gen_cls["TokenInfo"]
BatchedTokenizeRequest = _gen_cls["BatchedTokenizeRequest"]
BatchedTokenizeResponse = _gen_cls["BatchedTokenizeResponse"]
TokenizeRequest = _gen_cls["TokenizeRequest"]
TokenizeResponse = _gen_cls["TokenizeResponse"]
ModelInfoRequest = _gen_cls["ModelInfoRequest"]
ModelInfoResponse = _gen_cls["ModelInfoResponse"]

DecodingMethod = _gen_file.enum_types_by_name["DecodingMethod"]
StopReason = _gen_file.enum_types_by_name["StopReason"]

GREEDY = DecodingMethod.values_by_name["GREEDY"].number
SAMPLE = DecodingMethod.values_by_name["SAMPLE"].number


class _StopReasonNS:
    """Attribute access for StopReason values (pb2-module style)."""

    NOT_FINISHED = 0
    MAX_TOKENS = 1
    EOS_TOKEN = 2
    CANCELLED = 3
    TIME_LIMIT = 4
    STOP_SEQUENCE = 5
    TOKEN_LIMIT = 6
    ERROR = 7

    @staticmethod
    def Name(number: int) -> str:
        return StopReason.values_by_number[number].name


StopReasonValue = _StopReasonNS

SERVICE_NAME = "fmaas.GenerationService"

# ---------------------------------------------------------------------------
# grpc.health.v1 schema (standard health-checking protocol)
# ---------------------------------------------------------------------------

_HEALTH_MESSAGES = [
    Msg("HealthCheckRequest", [
        F("service", 1, "string"),
    ]),
    Msg("HealthCheckResponse", [
        F("status", 1, ".grpc.health.v1.HealthCheckResponse.ServingStatus!"),
    ], nested=[
        Enum("ServingStatus", {
            "UNKNOWN": 0,
            "SERVING": 1,
            "NOT_SERVING": 2,
            "SERVICE_UNKNOWN": 3,
        }),
    ]),
]

_health_file = build_file("grpc/health/v1/health.proto", "grpc.health.v1", _HEALTH_MESSAGES)
_health_cls = message_classes(_health_file)

HealthCheckRequest = _health_cls["HealthCheckRequest"]
HealthCheckResponse = _health_cls["HealthCheckResponse"]

HEALTH_SERVICE_NAME = "grpc.health.v1.Health"


def serialized_file_descriptors() -> dict[str, bytes]:
    """name -> serialized FileDescriptorProto, for server reflection."""
    out = {}
    for fd in (_gen_file, _health_file):
        fdp = descriptor_pb2.FileDescriptorProto()
        fd.CopyToProto(fdp)
        out[fd.name] = fdp.SerializeToString()
    return out
