"""Open Inference Protocol (V2) tensor model.

A fresh implementation of the V2 request/response tensor codec with the
binary-tensor extension. Behavioral contract matched against the reference
(python/kserve/kserve/protocol/infer_type.py:36-111 serialize/deserialize,
:593-668 from_bytes, :717 to_rest) — not a code port:

- BYTES tensors serialize as 4-byte little-endian length-prefixed elements,
  row-major.
- Binary extension: HTTP body = JSON prefix of ``inference-content-length``
  bytes, then the raw tensors of every input carrying
  ``parameters.binary_data_size``, concatenated in input order.
- FP16/BF16 tensors must use the binary path; JSON FP16 is rejected
  (reference :627-631).
"""

from __future__ import annotations

import json
import struct
import uuid
from typing import Any, Dict, List, Optional, Union

import numpy as np

from kserve_amd.errors import InvalidInput

# ---------------------------------------------------------------------------
# V2 datatype <-> numpy dtype (reference: utils/numpy_codec.py:18-37)
# ---------------------------------------------------------------------------

_DATATYPE_TO_NP = {
    "BOOL": np.bool_,
    "UINT8": np.uint8,
    "UINT16": np.uint16,
    "UINT32": np.uint32,
    "UINT64": np.uint64,
    "INT8": np.int8,
    "INT16": np.int16,
    "INT32": np.int32,
    "INT64": np.int64,
    "FP16": np.float16,
    "FP32": np.float32,
    "FP64": np.float64,
    "BYTES": np.object_,
}

_NP_TO_DATATYPE = {
    np.dtype(np.bool_): "BOOL",
    np.dtype(np.uint8): "UINT8",
    np.dtype(np.uint16): "UINT16",
    np.dtype(np.uint32): "UINT32",
    np.dtype(np.uint64): "UINT64",
    np.dtype(np.int8): "INT8",
    np.dtype(np.int16): "INT16",
    np.dtype(np.int32): "INT32",
    np.dtype(np.int64): "INT64",
    np.dtype(np.float16): "FP16",
    np.dtype(np.float32): "FP32",
    np.dtype(np.float64): "FP64",
    np.dtype(np.object_): "BYTES",
}

# datatypes that may NOT be carried as JSON numbers (no lossless JSON form)
_BINARY_ONLY_DATATYPES = ("FP16", "BF16")


def to_np_dtype(datatype: str):
    """V2 datatype string -> numpy dtype (BF16 has no numpy dtype -> uint16 view)."""
    if datatype == "BF16":
        return np.uint16  # raw-bits view; torch side reinterprets
    dt = _DATATYPE_TO_NP.get(datatype)
    if dt is None:
        raise InvalidInput(f"Unsupported datatype {datatype}")
    return dt


def from_np_dtype(dtype) -> str:
    datatype = _NP_TO_DATATYPE.get(np.dtype(dtype))
    if datatype is None:
        if np.dtype(dtype).kind in ("S", "U"):
            return "BYTES"
        raise InvalidInput(f"Unsupported numpy dtype {dtype}")
    return datatype


# ---------------------------------------------------------------------------
# BYTES tensor wire format (reference: infer_type.py:36-111)
# ---------------------------------------------------------------------------

def serialize_bytes_tensor(arr: np.ndarray) -> bytes:
    """Row-major concatenation of 4-byte-LE length-prefixed elements."""
    if arr.size == 0:
        return b""
    flat = arr.flatten(order="C")
    out = bytearray()
    for el in flat:
        if isinstance(el, bytes):
            b = el
        elif isinstance(el, str):
            b = el.encode("utf-8")
        elif isinstance(el, np.bytes_):
            b = bytes(el)
        else:
            b = str(el).encode("utf-8")
        out += struct.pack("<I", len(b))
        out += b
    return bytes(out)


def deserialize_bytes_tensor(raw: bytes) -> np.ndarray:
    """Inverse of :func:`serialize_bytes_tensor`; returns 1-D object array."""
    elems: List[bytes] = []
    off = 0
    n = len(raw)
    while off < n:
        if off + 4 > n:
            raise InvalidInput("Truncated BYTES tensor: length prefix cut short")
        (ln,) = struct.unpack_from("<I", raw, off)
        off += 4
        if off + ln > n:
            raise InvalidInput("Truncated BYTES tensor: element cut short")
        elems.append(raw[off : off + ln])
        off += ln
    return np.array(elems, dtype=np.object_)


def _shape_size(shape) -> int:
    size = 1
    for d in shape:
        size *= int(d)
    return size


# ---------------------------------------------------------------------------
# Tensors
# ---------------------------------------------------------------------------

class InferInput:
    """One named input tensor (reference: infer_type.py:113-381)."""

    __slots__ = ("name", "shape", "datatype", "_data", "_raw_data", "parameters")

    def __init__(
        self,
        name: str,
        shape: List[int],
        datatype: str,
        data: Union[List, np.ndarray, None] = None,
        parameters: Optional[Dict[str, Any]] = None,
    ):
        self.name = name
        self.shape = list(shape)
        self.datatype = datatype
        self.parameters = parameters or {}
        self._raw_data: Optional[bytes] = None
        self._data: Optional[List] = None
        if isinstance(data, np.ndarray):
            self.set_data_from_numpy(data)
        elif data is not None:
            self._data = data

    # -- data accessors ----------------------------------------------------
    @property
    def data(self):
        return self._data

    @data.setter
    def data(self, value):
        self._data = value

    @property
    def raw_data(self):
        return self._raw_data

    def set_raw_data(self, raw: bytes):
        self._raw_data = raw
        self._data = None
        self.parameters["binary_data_size"] = len(raw)

    def set_data_from_numpy(self, arr: np.ndarray, binary_data: bool = True):
        expected = from_np_dtype(arr.dtype) if self.datatype != "BF16" else "BF16"
        if self.datatype not in (expected, "BYTES") and not (
            self.datatype == "BF16" and arr.dtype == np.uint16
        ):
            if expected != self.datatype:
                raise InvalidInput(
                    f"got unexpected dtype {arr.dtype} for input {self.name} "
                    f"(datatype {self.datatype})"
                )
        self.shape = list(arr.shape)
        if binary_data:
            if self.datatype == "BYTES":
                self.set_raw_data(serialize_bytes_tensor(arr))
            else:
                self.set_raw_data(np.ascontiguousarray(arr).tobytes())
        else:
            if self.datatype in _BINARY_ONLY_DATATYPES:
                raise InvalidInput(
                    f"{self.datatype} tensors must use binary data "
                    f"(input {self.name})"
                )
            if self.datatype == "BYTES":
                self._data = [
                    el.decode("utf-8") if isinstance(el, (bytes, np.bytes_)) else el
                    for el in arr.flatten(order="C")
                ]
            else:
                self._data = arr.flatten(order="C").tolist()
            self._raw_data = None
            self.parameters.pop("binary_data_size", None)

    def as_numpy(self) -> np.ndarray:
        dtype = to_np_dtype(self.datatype)
        if self._raw_data is not None:
            if self.datatype == "BYTES":
                arr = deserialize_bytes_tensor(self._raw_data)
                return arr.reshape(self.shape)
            arr = np.frombuffer(self._raw_data, dtype=dtype)
            return arr.reshape(self.shape)
        if self._data is None:
            raise InvalidInput(f"Input {self.name} has no data")
        if self.datatype == "BYTES":
            flat = [
                el.encode("utf-8") if isinstance(el, str) else el for el in self._data
            ]
            return np.array(flat, dtype=np.object_).reshape(self.shape)
        return np.array(self._data, dtype=dtype).reshape(self.shape)

    # -- REST dict form ----------------------------------------------------
    def to_dict(self, binary: bool = False) -> Dict[str, Any]:
        d: Dict[str, Any] = {
            "name": self.name,
            "shape": self.shape,
            "datatype": self.datatype,
        }
        params = dict(self.parameters)
        if binary and self._raw_data is not None:
            params["binary_data_size"] = len(self._raw_data)
        else:
            params.pop("binary_data_size", None)
            d["data"] = self._data if self._data is not None else self._json_data()
        if params:
            d["parameters"] = params
        return d

    def _json_data(self):
        if self.datatype in _BINARY_ONLY_DATATYPES:
            raise InvalidInput(
                f"Cannot JSON-encode {self.datatype} tensor {self.name}; "
                "use the binary extension"
            )
        arr = self.as_numpy()
        if self.datatype == "BYTES":
            return [
                el.decode("utf-8", errors="replace")
                if isinstance(el, (bytes, np.bytes_))
                else el
                for el in arr.flatten(order="C")
            ]
        return arr.flatten(order="C").tolist()

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "InferInput":
        try:
            inp = cls(
                name=d["name"],
                shape=d["shape"],
                datatype=d["datatype"],
                parameters=d.get("parameters") or {},
            )
        except KeyError as e:
            raise InvalidInput(f"Missing input field {e}")
        if "data" in d:
            if inp.datatype in _BINARY_ONLY_DATATYPES:
                # reference rejects JSON FP16 (infer_type.py:627-631)
                raise InvalidInput(
                    f"{inp.datatype} input {inp.name} must use binary data"
                )
            inp._data = d["data"]
        return inp

    def __eq__(self, other):
        if not isinstance(other, InferInput):
            return False
        if (self.name, self.datatype, list(self.shape)) != (
            other.name,
            other.datatype,
            list(other.shape),
        ):
            return False
        try:
            return np.array_equal(self.as_numpy(), other.as_numpy())
        except InvalidInput:
            return self._data == other._data and self._raw_data == other._raw_data

    def __repr__(self):
        return (
            f"InferInput(name={self.name!r}, shape={self.shape}, "
            f"datatype={self.datatype!r})"
        )


class RequestedOutput:
    """Requested output with params, e.g. binary_data (reference: infer_type.py)."""

    __slots__ = ("name", "parameters")

    def __init__(self, name: str, parameters: Optional[Dict[str, Any]] = None):
        self.name = name
        self.parameters = parameters or {}

    @property
    def binary_data(self) -> bool:
        return bool(self.parameters.get("binary_data", False))

    def to_dict(self):
        d = {"name": self.name}
        if self.parameters:
            d["parameters"] = self.parameters
        return d

    @classmethod
    def from_dict(cls, d):
        return cls(name=d["name"], parameters=d.get("parameters") or {})


class InferOutput(InferInput):
    """Output tensor — same wire shape as an input tensor."""

    def __repr__(self):
        return (
            f"InferOutput(name={self.name!r}, shape={self.shape}, "
            f"datatype={self.datatype!r})"
        )


# ---------------------------------------------------------------------------
# Request / Response
# ---------------------------------------------------------------------------

class InferRequest:
    """V2 inference request (reference: infer_type.py:480-918)."""

    def __init__(
        self,
        model_name: str,
        infer_inputs: List[InferInput],
        request_id: Optional[str] = None,
        request_outputs: Optional[List[RequestedOutput]] = None,
        parameters: Optional[Dict[str, Any]] = None,
        from_grpc: bool = False,
    ):
        self.model_name = model_name
        self.inputs = infer_inputs
        self.id = request_id or str(uuid.uuid4())
        self.request_outputs = request_outputs or []
        self.parameters = parameters or {}
        self.from_grpc = from_grpc

    # -- REST decode -------------------------------------------------------
    @classmethod
    def from_inference_request(cls, model_name: str, body: Dict[str, Any]) -> "InferRequest":
        try:
            inputs = [InferInput.from_dict(i) for i in body["inputs"]]
        except KeyError:
            raise InvalidInput("Missing 'inputs' in request body")
        outs = [RequestedOutput.from_dict(o) for o in body.get("outputs", [])]
        return cls(
            model_name=model_name,
            infer_inputs=inputs,
            request_id=body.get("id"),
            request_outputs=outs,
            parameters=body.get("parameters") or {},
        )

    @classmethod
    def from_bytes(cls, body: bytes, json_length: int, model_name: str) -> "InferRequest":
        """Decode the binary extension: JSON prefix + concatenated raw tensors.

        Reference semantics: infer_type.py:593-668.
        """
        if json_length > len(body):
            raise InvalidInput(
                "inference-content-length larger than request body"
            )
        try:
            meta = json.loads(body[:json_length])
        except json.JSONDecodeError as e:
            raise InvalidInput(f"Invalid JSON prefix: {e}")
        req = cls.from_inference_request(model_name, meta)
        off = json_length
        for inp in req.inputs:
            bsz = inp.parameters.get("binary_data_size")
            if bsz is None:
                continue
            bsz = int(bsz)
            if off + bsz > len(body):
                raise InvalidInput(
                    f"Truncated binary tensor for input {inp.name}"
                )
            inp.set_raw_data(body[off : off + bsz])
            off += bsz
        # re-decode: JSON 'data' for FP16/BF16 was rejected in from_dict only
        # when no binary size present; validate remaining binary-only inputs
        for inp in req.inputs:
            if inp.datatype in _BINARY_ONLY_DATATYPES and inp.raw_data is None:
                raise InvalidInput(
                    f"{inp.datatype} input {inp.name} requires binary data"
                )
        return req

    # -- REST encode -------------------------------------------------------
    def to_rest(self):
        """Returns ``(body, json_length)``.

        ``body`` is a dict when no input carries raw data, else ``bytes`` of
        JSON prefix + raw tensors and ``json_length`` of the prefix
        (reference: infer_type.py:717-790).
        """
        use_binary = any(i.raw_data is not None for i in self.inputs)
        d: Dict[str, Any] = {
            "id": self.id,
            "inputs": [i.to_dict(binary=i.raw_data is not None) for i in self.inputs],
        }
        if self.request_outputs:
            d["outputs"] = [o.to_dict() for o in self.request_outputs]
        if self.parameters:
            d["parameters"] = _clean_params(self.parameters)
        if not use_binary:
            return d, None
        prefix = json.dumps(d).encode("utf-8")
        chunks = [prefix]
        for i in self.inputs:
            if i.raw_data is not None:
                chunks.append(i.raw_data)
        return b"".join(chunks), len(prefix)

    def as_dataframe(self):
        """Columns = inputs (reference: infer_type.py:849)."""
        import pandas as pd

        cols = {}
        for inp in self.inputs:
            arr = inp.as_numpy()
            if arr.dtype == np.object_:
                arr = np.array(
                    [
                        el.decode("utf-8") if isinstance(el, (bytes, np.bytes_)) else el
                        for el in arr.flatten(order="C")
                    ]
                ).reshape(arr.shape)
            cols[inp.name] = arr.flatten(order="C") if arr.ndim <= 1 else list(arr)
        return pd.DataFrame(cols)

    def get_input_by_name(self, name: str) -> Optional[InferInput]:
        for i in self.inputs:
            if i.name == name:
                return i
        return None

    def __eq__(self, other):
        return (
            isinstance(other, InferRequest)
            and self.model_name == other.model_name
            and self.inputs == other.inputs
        )

    def __repr__(self):
        return (
            f"InferRequest(model_name={self.model_name!r}, id={self.id!r}, "
            f"inputs={self.inputs!r})"
        )


def _clean_params(params: Dict[str, Any]) -> Dict[str, Any]:
    return {k: v for k, v in params.items() if not k.startswith("_")}


class InferResponse:
    """V2 inference response (reference: infer_type.py:1162+)."""

    def __init__(
        self,
        response_id: str,
        model_name: str,
        infer_outputs: List[InferOutput],
        model_version: Optional[str] = None,
        parameters: Optional[Dict[str, Any]] = None,
        from_grpc: bool = False,
    ):
        self.id = response_id
        self.model_name = model_name
        self.model_version = model_version
        self.outputs = infer_outputs
        self.parameters = parameters or {}
        self.from_grpc = from_grpc

    @classmethod
    def from_rest(cls, model_name: str, response: Dict[str, Any]) -> "InferResponse":
        outs = [InferOutput.from_dict(o) for o in response.get("outputs", [])]
        return cls(
            response_id=response.get("id", ""),
            model_name=response.get("model_name", model_name),
            model_version=response.get("model_version"),
            infer_outputs=outs,
            parameters=response.get("parameters") or {},
        )

    @classmethod
    def from_bytes(cls, body: bytes, json_length: int) -> "InferResponse":
        try:
            meta = json.loads(body[:json_length])
        except json.JSONDecodeError as e:
            raise InvalidInput(f"Invalid JSON prefix: {e}")
        resp = cls.from_rest(meta.get("model_name", ""), meta)
        off = json_length
        for out in resp.outputs:
            bsz = out.parameters.get("binary_data_size")
            if bsz is None:
                continue
            bsz = int(bsz)
            if off + bsz > len(body):
                # mirror the request-side truncation check: fail loudly here
                # instead of with an unrelated reshape error downstream
                raise InvalidInput(
                    f"Truncated binary output {out.name}: need {bsz} bytes "
                    f"at offset {off}, body has {len(body)}"
                )
            out.set_raw_data(body[off : off + bsz])
            off += bsz
        return resp

    def to_rest(self, requested_outputs: Optional[List[RequestedOutput]] = None):
        """Returns ``(body, json_length)`` like :meth:`InferRequest.to_rest`.

        An output goes binary when the caller requested ``binary_data`` for it
        or when it already carries raw data and the caller did not explicitly
        request JSON.
        """
        want_binary: Dict[str, bool] = {}
        if requested_outputs:
            for ro in requested_outputs:
                want_binary[ro.name] = ro.binary_data

        binary_outputs: List[InferOutput] = []
        dicts = []
        for out in self.outputs:
            binary = want_binary.get(out.name, out.raw_data is not None)
            if binary and out.raw_data is None:
                out.set_data_from_numpy(out.as_numpy(), binary_data=True)
            if not binary and out.raw_data is not None and out.datatype not in _BINARY_ONLY_DATATYPES:
                out.set_data_from_numpy(out.as_numpy(), binary_data=False)
                # keep non-binary
            d = out.to_dict(binary=binary)
            dicts.append(d)
            if binary:
                binary_outputs.append(out)
        body: Dict[str, Any] = {
            "id": self.id,
            "model_name": self.model_name,
            "outputs": dicts,
        }
        if self.model_version:
            body["model_version"] = self.model_version
        if self.parameters:
            body["parameters"] = _clean_params(self.parameters)
        if not binary_outputs:
            return body, None
        prefix = json.dumps(body).encode("utf-8")
        chunks = [prefix] + [o.raw_data for o in binary_outputs]
        return b"".join(chunks), len(prefix)

    def get_output_by_name(self, name: str) -> Optional[InferOutput]:
        for o in self.outputs:
            if o.name == name:
                return o
        return None

    def __eq__(self, other):
        return (
            isinstance(other, InferResponse)
            and self.model_name == other.model_name
            and self.outputs == other.outputs
        )

    def __repr__(self):
        return (
            f"InferResponse(id={self.id!r}, model_name={self.model_name!r}, "
            f"outputs={self.outputs!r})"
        )
