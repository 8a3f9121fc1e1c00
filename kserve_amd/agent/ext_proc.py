"""Envoy external-processor (ext-proc) endpoint picker.

The Gateway API Inference Extension drives endpoint selection through
Envoy's `envoy.service.ext_proc.v3.ExternalProcessor/Process` bidi
stream: the gateway sends request headers/body, the picker answers with
a header mutation naming the pool member
(`x-gateway-destination-endpoint`), and Envoy routes the request there.
The reference deploys llm-d's external EPP image speaking exactly this
protocol (llmisvc/scheduler.go:74-388, grpc :9002 / health :9003).

This is the MI355X-native EPP's gRPC face: a wire-compatible SUBSET of
the ext-proc schema (field numbers per envoy ext_proc.proto v3; unknown
fields in incoming messages are ignored by protobuf, so speaking the
subset is safe), built programmatically because the image has no protoc
— same approach as `protocol/grpc/proto.py`. Selection logic lives in
`agent.endpoint_picker.EndpointPicker`; this module only adapts it to
the stream protocol.
"""

from __future__ import annotations

from typing import Optional

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_F = descriptor_pb2.FieldDescriptorProto
_POOL = descriptor_pool.Default()
_FILE_NAME = "kserve_amd/ext_proc_subset.proto"
_PKG = "envoy.service.ext_proc.v3"

# the header the Gateway API Inference Extension routes on
DESTINATION_HEADER = "x-gateway-destination-endpoint"
SESSION_HEADER = "x-session-id"


def _field(name, number, ftype, label=_F.LABEL_OPTIONAL, type_name=None,
           oneof_index=None):
    f = _F(name=name, number=number, type=ftype, label=label)
    if type_name:
        f.type_name = type_name if type_name.startswith(".") else f".{_PKG}.{type_name}"
    if oneof_index is not None:
        f.oneof_index = oneof_index
    return f


def _build_file() -> descriptor_pb2.FileDescriptorProto:
    fd = descriptor_pb2.FileDescriptorProto()
    fd.name = _FILE_NAME
    fd.package = _PKG
    fd.syntax = "proto3"

    def msg(name):
        m = fd.message_type.add()
        m.name = name
        return m

    # envoy.config.core.v3.HeaderValue / HeaderMap / HeaderValueOption —
    # declared inside our package (type identity is by field number on
    # the wire, not by name, so this stays wire-compatible)
    hv = msg("HeaderValue")
    hv.field.append(_field("key", 1, _F.TYPE_STRING))
    hv.field.append(_field("value", 2, _F.TYPE_STRING))
    hv.field.append(_field("raw_value", 3, _F.TYPE_BYTES))

    hm = msg("HeaderMap")
    hm.field.append(
        _field("headers", 1, _F.TYPE_MESSAGE, _F.LABEL_REPEATED, "HeaderValue")
    )

    hvo = msg("HeaderValueOption")
    hvo.field.append(_field("header", 1, _F.TYPE_MESSAGE, type_name="HeaderValue"))

    hmut = msg("HeaderMutation")
    hmut.field.append(
        _field("set_headers", 1, _F.TYPE_MESSAGE, _F.LABEL_REPEATED,
               "HeaderValueOption")
    )
    hmut.field.append(
        _field("remove_headers", 2, _F.TYPE_STRING, _F.LABEL_REPEATED)
    )

    # ext_proc.proto subset
    hh = msg("HttpHeaders")
    hh.field.append(_field("headers", 1, _F.TYPE_MESSAGE, type_name="HeaderMap"))
    hh.field.append(_field("end_of_stream", 3, _F.TYPE_BOOL))

    hb = msg("HttpBody")
    hb.field.append(_field("body", 1, _F.TYPE_BYTES))
    hb.field.append(_field("end_of_stream", 2, _F.TYPE_BOOL))

    creq = msg("ProcessingRequest")
    creq.oneof_decl.add().name = "request"
    creq.field.append(
        _field("request_headers", 2, _F.TYPE_MESSAGE, type_name="HttpHeaders",
               oneof_index=0))
    creq.field.append(
        _field("response_headers", 3, _F.TYPE_MESSAGE, type_name="HttpHeaders",
               oneof_index=0))
    creq.field.append(
        _field("request_body", 4, _F.TYPE_MESSAGE, type_name="HttpBody",
               oneof_index=0))
    creq.field.append(
        _field("response_body", 5, _F.TYPE_MESSAGE, type_name="HttpBody",
               oneof_index=0))

    common = msg("CommonResponse")
    common.field.append(_field("status", 1, _F.TYPE_INT32))  # 0 = CONTINUE
    common.field.append(
        _field("header_mutation", 2, _F.TYPE_MESSAGE, type_name="HeaderMutation"))

    hresp = msg("HeadersResponse")
    hresp.field.append(
        _field("response", 1, _F.TYPE_MESSAGE, type_name="CommonResponse"))

    bresp = msg("BodyResponse")
    bresp.field.append(
        _field("response", 1, _F.TYPE_MESSAGE, type_name="CommonResponse"))

    cresp = msg("ProcessingResponse")
    cresp.oneof_decl.add().name = "response"
    cresp.field.append(
        _field("request_headers", 1, _F.TYPE_MESSAGE,
               type_name="HeadersResponse", oneof_index=0))
    cresp.field.append(
        _field("response_headers", 2, _F.TYPE_MESSAGE,
               type_name="HeadersResponse", oneof_index=0))
    cresp.field.append(
        _field("request_body", 3, _F.TYPE_MESSAGE, type_name="BodyResponse",
               oneof_index=0))
    cresp.field.append(
        _field("response_body", 4, _F.TYPE_MESSAGE, type_name="BodyResponse",
               oneof_index=0))
    return fd


def _load():
    fd = _build_file()
    try:
        _POOL.Add(fd)
    except Exception:
        pass
    out = {}
    for name in ("HeaderValue", "HeaderMap", "HeaderValueOption",
                 "HeaderMutation", "HttpHeaders", "HttpBody",
                 "ProcessingRequest", "ProcessingResponse",
                 "CommonResponse", "HeadersResponse", "BodyResponse"):
        desc = _POOL.FindMessageTypeByName(f"{_PKG}.{name}")
        out[name] = message_factory.GetMessageClass(desc)
    return out


_M = _load()
ProcessingRequest = _M["ProcessingRequest"]
ProcessingResponse = _M["ProcessingResponse"]
HttpHeaders = _M["HttpHeaders"]
HeaderMap = _M["HeaderMap"]
HeaderValue = _M["HeaderValue"]

SERVICE_NAME = f"{_PKG}.ExternalProcessor"


def _headers_dict(http_headers) -> dict:
    out = {}
    for h in http_headers.headers.headers:
        val = h.value or (h.raw_value.decode("utf-8", "replace")
                          if h.raw_value else "")
        out[h.key.lower()] = val
    return out


def pick_response(picker, headers: dict):
    """Build the ProcessingResponse for a request_headers message: pick
    a member (session-sticky when x-session-id present) and mutate the
    destination header; no healthy member -> CONTINUE without mutation
    (the gateway falls back to its own load balancing)."""
    endpoint: Optional[str] = picker.pick(
        session_id=headers.get(SESSION_HEADER)
    )
    resp = ProcessingResponse()
    common = resp.request_headers.response
    common.status = 0  # CONTINUE
    if endpoint:
        opt = common.header_mutation.set_headers.add()
        opt.header.key = DESTINATION_HEADER
        opt.header.raw_value = endpoint.encode()
    return resp


class ExtProcServicer:
    """Bidi Process() stream: answer each request-phase message."""

    def __init__(self, picker):
        self.picker = picker

    async def Process(self, request_iterator, context):
        async for req in request_iterator:
            which = req.WhichOneof("request")
            if which == "request_headers":
                yield pick_response(
                    self.picker, _headers_dict(req.request_headers)
                )
            elif which == "request_body":
                resp = ProcessingResponse()
                resp.request_body.response.status = 0
                yield resp
            elif which == "response_headers":
                resp = ProcessingResponse()
                resp.response_headers.response.status = 0
                yield resp
            elif which == "response_body":
                resp = ProcessingResponse()
                resp.response_body.response.status = 0
                yield resp


def create_ext_proc_server(picker, port: int):
    """grpc.aio server speaking ExternalProcessor/Process."""
    import grpc

    server = grpc.aio.server()
    handler = grpc.method_handlers_generic_handler(
        SERVICE_NAME,
        {
            "Process": grpc.stream_stream_rpc_method_handler(
                ExtProcServicer(picker).Process,
                request_deserializer=ProcessingRequest.FromString,
                response_serializer=lambda m: m.SerializeToString(),
            )
        },
    )
    server.add_generic_rpc_handlers((handler,))
    server.add_insecure_port(f"[::]:{port}")
    return server
