"""Envoy ext-proc v3 wire messages, built at runtime from descriptors.

The reference's L1 is the Envoy external-processing protocol
(envoy.service.ext_proc.v3.ExternalProcessor/Process, FULL_DUPLEX_STREAMED
— /root/reference/pkg/epp/handlers/server.go:168, README.md:63-66). This
module reconstructs the message subset that protocol exchanges with the
EXACT field numbers and wire types of Envoy's public ext_proc.proto /
config.core.v3 headers / type.v3 HttpStatus, so byte streams produced here
are wire-compatible with an Envoy peer. There is no protoc in this image;
classes are generated from a hand-built FileDescriptorProto via
google.protobuf.message_factory (full protobuf semantics: oneofs,
unknown-field tolerance, deterministic serialization).

Field-number provenance (Envoy API, public):
  ProcessingRequest:  request_headers=2 response_headers=3 request_body=4
                      response_body=5 request_trailers=6 response_trailers=7
                      metadata_context=8 attributes=9 observability_mode=10
  ProcessingResponse: request_headers=1 response_headers=2 request_body=3
                      response_body=4 request_trailers=5 response_trailers=6
                      immediate_response=7 dynamic_metadata=8
  HttpHeaders{headers=1,end_of_stream=3} HttpBody{body=1,end_of_stream=2}
  HttpTrailers{trailers=1} HeadersResponse/BodyResponse{response=1}
  TrailersResponse{header_mutation=1}
  CommonResponse{status=1,header_mutation=2,body_mutation=3,trailers=4,
                 clear_route_cache=5}
  HeaderMutation{set_headers=1,remove_headers=2}
  BodyMutation{body=1,clear_body=2}
  ImmediateResponse{status=1,headers=2,body=3,details=5}
  HttpStatus{code=1}  HeaderMap{headers=1}
  HeaderValue{key=1,value=2,raw_value=3}
  HeaderValueOption{header=1,append_action=3}
"""
from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_PKG = "ldsext"
_FILE = "lds_extproc.proto"

# protobuf wire types
_STR, _BYTES, _BOOL, _INT32, _MSG = 9, 12, 8, 5, 11
_OPT, _REP = 1, 3

# CommonResponse.status values (ext_proc.proto CommonResponse.ResponseStatus)
CONTINUE = 0
CONTINUE_AND_REPLACE = 1


def _build_file() -> descriptor_pb2.FileDescriptorProto:
    f = descriptor_pb2.FileDescriptorProto()
    f.name = _FILE
    f.package = _PKG
    f.syntax = "proto3"
    f.dependency.append("google/protobuf/struct.proto")

    def msg(name):
        m = f.message_type.add()
        m.name = name
        return m

    def fld(m, name, number, ftype, label=_OPT, tname=None, oneof=None):
        fd = m.field.add()
        fd.name = name
        fd.number = number
        fd.type = ftype
        fd.label = label
        if tname:
            fd.type_name = tname
        if oneof is not None:
            fd.oneof_index = oneof

    def t(name):
        return f".{_PKG}.{name}"

    m = msg("HeaderValue")
    fld(m, "key", 1, _STR)
    fld(m, "value", 2, _STR)
    fld(m, "raw_value", 3, _BYTES)

    m = msg("HeaderMap")
    fld(m, "headers", 1, _MSG, _REP, t("HeaderValue"))

    m = msg("HeaderValueOption")
    fld(m, "header", 1, _MSG, tname=t("HeaderValue"))
    fld(m, "append_action", 3, _INT32)

    m = msg("HeaderMutation")
    fld(m, "set_headers", 1, _MSG, _REP, t("HeaderValueOption"))
    fld(m, "remove_headers", 2, _STR, _REP)

    m = msg("BodyMutation")
    m.oneof_decl.add().name = "mutation"
    fld(m, "body", 1, _BYTES, oneof=0)
    fld(m, "clear_body", 2, _BOOL, oneof=0)

    m = msg("HttpStatus")
    fld(m, "code", 1, _INT32)

    m = msg("HttpHeaders")
    fld(m, "headers", 1, _MSG, tname=t("HeaderMap"))
    fld(m, "end_of_stream", 3, _BOOL)

    m = msg("HttpBody")
    fld(m, "body", 1, _BYTES)
    fld(m, "end_of_stream", 2, _BOOL)

    m = msg("HttpTrailers")
    fld(m, "trailers", 1, _MSG, tname=t("HeaderMap"))

    m = msg("ProcessingRequest")
    m.oneof_decl.add().name = "request"
    fld(m, "request_headers", 2, _MSG, tname=t("HttpHeaders"), oneof=0)
    fld(m, "response_headers", 3, _MSG, tname=t("HttpHeaders"), oneof=0)
    fld(m, "request_body", 4, _MSG, tname=t("HttpBody"), oneof=0)
    fld(m, "response_body", 5, _MSG, tname=t("HttpBody"), oneof=0)
    fld(m, "request_trailers", 6, _MSG, tname=t("HttpTrailers"), oneof=0)
    fld(m, "response_trailers", 7, _MSG, tname=t("HttpTrailers"), oneof=0)
    fld(m, "observability_mode", 10, _BOOL)

    m = msg("CommonResponse")
    fld(m, "status", 1, _INT32)
    fld(m, "header_mutation", 2, _MSG, tname=t("HeaderMutation"))
    fld(m, "body_mutation", 3, _MSG, tname=t("BodyMutation"))
    fld(m, "trailers", 4, _MSG, tname=t("HeaderMap"))
    fld(m, "clear_route_cache", 5, _BOOL)

    m = msg("HeadersResponse")
    fld(m, "response", 1, _MSG, tname=t("CommonResponse"))

    m = msg("BodyResponse")
    fld(m, "response", 1, _MSG, tname=t("CommonResponse"))

    m = msg("TrailersResponse")
    fld(m, "header_mutation", 1, _MSG, tname=t("HeaderMutation"))

    m = msg("ImmediateResponse")
    fld(m, "status", 1, _MSG, tname=t("HttpStatus"))
    fld(m, "headers", 2, _MSG, tname=t("HeaderMutation"))
    fld(m, "body", 3, _BYTES)
    fld(m, "details", 5, _STR)

    m = msg("ProcessingResponse")
    m.oneof_decl.add().name = "response"
    fld(m, "request_headers", 1, _MSG, tname=t("HeadersResponse"), oneof=0)
    fld(m, "response_headers", 2, _MSG, tname=t("HeadersResponse"), oneof=0)
    fld(m, "request_body", 3, _MSG, tname=t("BodyResponse"), oneof=0)
    fld(m, "response_body", 4, _MSG, tname=t("BodyResponse"), oneof=0)
    fld(m, "request_trailers", 5, _MSG, tname=t("TrailersResponse"),
        oneof=0)
    fld(m, "response_trailers", 6, _MSG, tname=t("TrailersResponse"),
        oneof=0)
    fld(m, "immediate_response", 7, _MSG, tname=t("ImmediateResponse"),
        oneof=0)
    fld(m, "dynamic_metadata", 8, _MSG, tname=".google.protobuf.Struct")
    return f


from google.protobuf import struct_pb2  # noqa: E402  (registers struct.proto)

_pool = descriptor_pool.Default()
assert struct_pb2.Struct is not None
try:
    _pool.FindFileByName(_FILE)
except KeyError:
    _pool.Add(_build_file())


def _cls(name):
    return message_factory.GetMessageClass(
        _pool.FindMessageTypeByName(f"{_PKG}.{name}"))


HeaderValue = _cls("HeaderValue")
HeaderMap = _cls("HeaderMap")
HeaderValueOption = _cls("HeaderValueOption")
HeaderMutation = _cls("HeaderMutation")
BodyMutation = _cls("BodyMutation")
HttpStatus = _cls("HttpStatus")
HttpHeaders = _cls("HttpHeaders")
HttpBody = _cls("HttpBody")
HttpTrailers = _cls("HttpTrailers")
ProcessingRequest = _cls("ProcessingRequest")
CommonResponse = _cls("CommonResponse")
HeadersResponse = _cls("HeadersResponse")
BodyResponse = _cls("BodyResponse")
TrailersResponse = _cls("TrailersResponse")
ImmediateResponse = _cls("ImmediateResponse")
ProcessingResponse = _cls("ProcessingResponse")


def headers_to_dict(hm) -> dict:
    """HeaderMap -> lowercase dict; raw_value wins over value (Envoy sets
    exactly one of the two)."""
    out = {}
    for hv in hm.headers:
        v = hv.raw_value.decode("utf-8", "replace") if hv.raw_value \
            else hv.value
        out[hv.key.lower()] = v
    return out


def set_header(mutation, key: str, value: str) -> None:
    """Append a set-header op using raw_value (what the reference emits —
    GIE helpers build HeaderValueOption{RawValue}); append_action
    OVERWRITE_IF_EXISTS_OR_ADD=0."""
    opt = mutation.set_headers.add()
    opt.header.key = key
    opt.header.raw_value = value.encode()
