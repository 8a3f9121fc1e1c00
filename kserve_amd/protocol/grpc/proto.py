"""Open Inference Protocol gRPC message definitions, built programmatically.

There is no protoc in the target image, so the V2 gRPC schema
(reference: python/kserve protocol/grpc/grpc_predict_v2.proto:23-50,
Triton-compatible ``inference.GRPCInferenceService``) is defined here via
``descriptor_pb2`` and realized with ``message_factory`` — wire-compatible
with stock KServe/Triton V2 gRPC clients.
"""

from __future__ import annotations

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_F = descriptor_pb2.FieldDescriptorProto

_POOL = descriptor_pool.Default()

_FILE_NAME = "kserve_amd/grpc_predict_v2.proto"
_PKG = "inference"


def _field(name, number, ftype, label=_F.LABEL_OPTIONAL, type_name=None, oneof_index=None):
    f = _F(name=name, number=number, type=ftype, label=label)
    if type_name:
        f.type_name = f".{_PKG}.{type_name}"
    if oneof_index is not None:
        f.oneof_index = oneof_index
    return f


def _map_field(msg, name, number, value_type_name, scope=None):
    """Add ``map<string, ValueType>`` field: nested MapEntry message + repeated field.

    ``scope`` is the fully-qualified name of ``msg`` (defaults to
    ``<pkg>.<msg.name>`` for top-level messages).
    """
    entry_name = "".join(p.capitalize() for p in name.split("_")) + "Entry"
    entry = msg.nested_type.add()
    entry.name = entry_name
    entry.options.map_entry = True
    entry.field.append(_field("key", 1, _F.TYPE_STRING))
    entry.field.append(
        _field("value", 2, _F.TYPE_MESSAGE, type_name=value_type_name)
    )
    f = msg.field.add()
    f.name = name
    f.number = number
    f.type = _F.TYPE_MESSAGE
    f.label = _F.LABEL_REPEATED
    scope = scope or f"{_PKG}.{msg.name}"
    f.type_name = f".{scope}.{entry_name}"


def _build_file() -> descriptor_pb2.FileDescriptorProto:
    fd = descriptor_pb2.FileDescriptorProto()
    fd.name = _FILE_NAME
    fd.package = _PKG
    fd.syntax = "proto3"

    def msg(name):
        m = fd.message_type.add()
        m.name = name
        return m

    # -- health ----------------------------------------------------------
    msg("ServerLiveRequest")
    m = msg("ServerLiveResponse")
    m.field.append(_field("live", 1, _F.TYPE_BOOL))
    msg("ServerReadyRequest")
    m = msg("ServerReadyResponse")
    m.field.append(_field("ready", 1, _F.TYPE_BOOL))
    m = msg("ModelReadyRequest")
    m.field.append(_field("name", 1, _F.TYPE_STRING))
    m.field.append(_field("version", 2, _F.TYPE_STRING))
    m = msg("ModelReadyResponse")
    m.field.append(_field("ready", 1, _F.TYPE_BOOL))

    # -- metadata --------------------------------------------------------
    msg("ServerMetadataRequest")
    m = msg("ServerMetadataResponse")
    m.field.append(_field("name", 1, _F.TYPE_STRING))
    m.field.append(_field("version", 2, _F.TYPE_STRING))
    m.field.append(_field("extensions", 3, _F.TYPE_STRING, _F.LABEL_REPEATED))
    m = msg("ModelMetadataRequest")
    m.field.append(_field("name", 1, _F.TYPE_STRING))
    m.field.append(_field("version", 2, _F.TYPE_STRING))

    m = msg("ModelMetadataResponse")
    tm = m.nested_type.add()
    tm.name = "TensorMetadata"
    tm.field.append(_field("name", 1, _F.TYPE_STRING))
    tm.field.append(_field("datatype", 2, _F.TYPE_STRING))
    tm.field.append(_field("shape", 3, _F.TYPE_INT64, _F.LABEL_REPEATED))
    m.field.append(_field("name", 1, _F.TYPE_STRING))
    m.field.append(_field("versions", 2, _F.TYPE_STRING, _F.LABEL_REPEATED))
    m.field.append(_field("platform", 3, _F.TYPE_STRING))
    f = _field("inputs", 4, _F.TYPE_MESSAGE, _F.LABEL_REPEATED)
    f.type_name = f".{_PKG}.ModelMetadataResponse.TensorMetadata"
    m.field.append(f)
    f = _field("outputs", 5, _F.TYPE_MESSAGE, _F.LABEL_REPEATED)
    f.type_name = f".{_PKG}.ModelMetadataResponse.TensorMetadata"
    m.field.append(f)

    # -- infer -----------------------------------------------------------
    m = msg("InferParameter")
    m.oneof_decl.add().name = "parameter_choice"
    m.field.append(_field("bool_param", 1, _F.TYPE_BOOL, oneof_index=0))
    m.field.append(_field("int64_param", 2, _F.TYPE_INT64, oneof_index=0))
    m.field.append(_field("string_param", 3, _F.TYPE_STRING, oneof_index=0))
    m.field.append(_field("double_param", 4, _F.TYPE_DOUBLE, oneof_index=0))
    m.field.append(_field("uint64_param", 5, _F.TYPE_UINT64, oneof_index=0))

    m = msg("InferTensorContents")
    m.field.append(_field("bool_contents", 1, _F.TYPE_BOOL, _F.LABEL_REPEATED))
    m.field.append(_field("int_contents", 2, _F.TYPE_INT32, _F.LABEL_REPEATED))
    m.field.append(_field("int64_contents", 3, _F.TYPE_INT64, _F.LABEL_REPEATED))
    m.field.append(_field("uint_contents", 4, _F.TYPE_UINT32, _F.LABEL_REPEATED))
    m.field.append(_field("uint64_contents", 5, _F.TYPE_UINT64, _F.LABEL_REPEATED))
    m.field.append(_field("fp32_contents", 6, _F.TYPE_FLOAT, _F.LABEL_REPEATED))
    m.field.append(_field("fp64_contents", 7, _F.TYPE_DOUBLE, _F.LABEL_REPEATED))
    m.field.append(_field("bytes_contents", 8, _F.TYPE_BYTES, _F.LABEL_REPEATED))

    m = msg("ModelInferRequest")
    it = m.nested_type.add()
    it.name = "InferInputTensor"
    it.field.append(_field("name", 1, _F.TYPE_STRING))
    it.field.append(_field("datatype", 2, _F.TYPE_STRING))
    it.field.append(_field("shape", 3, _F.TYPE_INT64, _F.LABEL_REPEATED))
    _map_field(it, "parameters", 4, "InferParameter", scope=f"{_PKG}.ModelInferRequest.InferInputTensor")
    it.field.append(
        _field("contents", 5, _F.TYPE_MESSAGE, type_name="InferTensorContents")
    )
    ot = m.nested_type.add()
    ot.name = "InferRequestedOutputTensor"
    ot.field.append(_field("name", 1, _F.TYPE_STRING))
    _map_field(ot, "parameters", 2, "InferParameter", scope=f"{_PKG}.ModelInferRequest.InferRequestedOutputTensor")
    m.field.append(_field("model_name", 1, _F.TYPE_STRING))
    m.field.append(_field("model_version", 2, _F.TYPE_STRING))
    m.field.append(_field("id", 3, _F.TYPE_STRING))
    _map_field(m, "parameters", 4, "InferParameter")
    f = _field("inputs", 5, _F.TYPE_MESSAGE, _F.LABEL_REPEATED)
    f.type_name = f".{_PKG}.ModelInferRequest.InferInputTensor"
    m.field.append(f)
    f = _field("outputs", 6, _F.TYPE_MESSAGE, _F.LABEL_REPEATED)
    f.type_name = f".{_PKG}.ModelInferRequest.InferRequestedOutputTensor"
    m.field.append(f)
    m.field.append(_field("raw_input_contents", 7, _F.TYPE_BYTES, _F.LABEL_REPEATED))

    m = msg("ModelInferResponse")
    ot = m.nested_type.add()
    ot.name = "InferOutputTensor"
    ot.field.append(_field("name", 1, _F.TYPE_STRING))
    ot.field.append(_field("datatype", 2, _F.TYPE_STRING))
    ot.field.append(_field("shape", 3, _F.TYPE_INT64, _F.LABEL_REPEATED))
    _map_field(ot, "parameters", 4, "InferParameter", scope=f"{_PKG}.ModelInferResponse.InferOutputTensor")
    ot.field.append(
        _field("contents", 5, _F.TYPE_MESSAGE, type_name="InferTensorContents")
    )
    m.field.append(_field("model_name", 1, _F.TYPE_STRING))
    m.field.append(_field("model_version", 2, _F.TYPE_STRING))
    m.field.append(_field("id", 3, _F.TYPE_STRING))
    _map_field(m, "parameters", 4, "InferParameter")
    f = _field("outputs", 5, _F.TYPE_MESSAGE, _F.LABEL_REPEATED)
    f.type_name = f".{_PKG}.ModelInferResponse.InferOutputTensor"
    m.field.append(f)
    m.field.append(_field("raw_output_contents", 6, _F.TYPE_BYTES, _F.LABEL_REPEATED))

    # -- repository ------------------------------------------------------
    m = msg("RepositoryIndexRequest")
    m.field.append(_field("repository_name", 1, _F.TYPE_STRING))
    m.field.append(_field("ready", 2, _F.TYPE_BOOL))
    m = msg("RepositoryIndexResponse")
    mi = m.nested_type.add()
    mi.name = "ModelIndex"
    mi.field.append(_field("name", 1, _F.TYPE_STRING))
    mi.field.append(_field("version", 2, _F.TYPE_STRING))
    mi.field.append(_field("state", 3, _F.TYPE_STRING))
    mi.field.append(_field("reason", 4, _F.TYPE_STRING))
    f = _field("models", 1, _F.TYPE_MESSAGE, _F.LABEL_REPEATED)
    f.type_name = f".{_PKG}.RepositoryIndexResponse.ModelIndex"
    m.field.append(f)
    m = msg("RepositoryModelLoadRequest")
    m.field.append(_field("repository_name", 1, _F.TYPE_STRING))
    m.field.append(_field("model_name", 2, _F.TYPE_STRING))
    msg("RepositoryModelLoadResponse")
    m = msg("RepositoryModelUnloadRequest")
    m.field.append(_field("repository_name", 1, _F.TYPE_STRING))
    m.field.append(_field("model_name", 2, _F.TYPE_STRING))
    msg("RepositoryModelUnloadResponse")

    return fd


def _load():
    fd = _build_file()
    try:
        file_desc = _POOL.Add(fd)
    except Exception:
        # already registered (module re-import)
        file_desc = _POOL.FindFileByName(_FILE_NAME)
    out = {}
    for name in (
        "ServerLiveRequest",
        "ServerLiveResponse",
        "ServerReadyRequest",
        "ServerReadyResponse",
        "ModelReadyRequest",
        "ModelReadyResponse",
        "ServerMetadataRequest",
        "ServerMetadataResponse",
        "ModelMetadataRequest",
        "ModelMetadataResponse",
        "InferParameter",
        "InferTensorContents",
        "ModelInferRequest",
        "ModelInferResponse",
        "RepositoryIndexRequest",
        "RepositoryIndexResponse",
        "RepositoryModelLoadRequest",
        "RepositoryModelLoadResponse",
        "RepositoryModelUnloadRequest",
        "RepositoryModelUnloadResponse",
    ):
        desc = _POOL.FindMessageTypeByName(f"{_PKG}.{name}")
        out[name] = message_factory.GetMessageClass(desc)
    return out


_MESSAGES = _load()

ServerLiveRequest = _MESSAGES["ServerLiveRequest"]
ServerLiveResponse = _MESSAGES["ServerLiveResponse"]
ServerReadyRequest = _MESSAGES["ServerReadyRequest"]
ServerReadyResponse = _MESSAGES["ServerReadyResponse"]
ModelReadyRequest = _MESSAGES["ModelReadyRequest"]
ModelReadyResponse = _MESSAGES["ModelReadyResponse"]
ServerMetadataRequest = _MESSAGES["ServerMetadataRequest"]
ServerMetadataResponse = _MESSAGES["ServerMetadataResponse"]
ModelMetadataRequest = _MESSAGES["ModelMetadataRequest"]
ModelMetadataResponse = _MESSAGES["ModelMetadataResponse"]
InferParameter = _MESSAGES["InferParameter"]
InferTensorContents = _MESSAGES["InferTensorContents"]
ModelInferRequest = _MESSAGES["ModelInferRequest"]
ModelInferResponse = _MESSAGES["ModelInferResponse"]
RepositoryIndexRequest = _MESSAGES["RepositoryIndexRequest"]
RepositoryIndexResponse = _MESSAGES["RepositoryIndexResponse"]
RepositoryModelLoadRequest = _MESSAGES["RepositoryModelLoadRequest"]
RepositoryModelLoadResponse = _MESSAGES["RepositoryModelLoadResponse"]
RepositoryModelUnloadRequest = _MESSAGES["RepositoryModelUnloadRequest"]
RepositoryModelUnloadResponse = _MESSAGES["RepositoryModelUnloadResponse"]

SERVICE_NAME = "inference.GRPCInferenceService"

# method name -> (request class, response class)
SERVICE_METHODS = {
    "ServerLive": (ServerLiveRequest, ServerLiveResponse),
    "ServerReady": (ServerReadyRequest, ServerReadyResponse),
    "ModelReady": (ModelReadyRequest, ModelReadyResponse),
    "ServerMetadata": (ServerMetadataRequest, ServerMetadataResponse),
    "ModelMetadata": (ModelMetadataRequest, ModelMetadataResponse),
    "ModelInfer": (ModelInferRequest, ModelInferResponse),
    "RepositoryIndex": (RepositoryIndexRequest, RepositoryIndexResponse),
    "RepositoryModelLoad": (RepositoryModelLoadRequest, RepositoryModelLoadResponse),
    "RepositoryModelUnload": (
        RepositoryModelUnloadRequest,
        RepositoryModelUnloadResponse,
    ),
}
