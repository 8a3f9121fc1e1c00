"""Protocol-agnostic data-plane core.

Reference parity: python/kserve/kserve/protocol/dataplane.py:49-509 —
model registry access, liveness/readiness, decode (JSON / V2 binary) ->
infer -> encode. CloudEvents decode is a thin optional layer (structured
JSON events only; binary-mode ce headers passed through).
"""

from __future__ import annotations

import json
from typing import Any, Dict, Optional, Tuple, Union

from kserve_amd import constants
from kserve_amd.errors import InvalidInput, ModelNotFound, ModelNotReady
from kserve_amd.model import BaseModel, InferenceModel
from kserve_amd.model_repository import ModelRepository
from kserve_amd.protocol.infer_type import InferRequest, InferResponse

JSON_HEADERS = ("application/json", "application/cloudevents+json", "application/ld+json")


class DataPlane:
    def __init__(self, model_registry: ModelRepository):
        self._model_registry = model_registry
        self._server_name = "kserve-amd"
        self._server_version = _package_version()

    @property
    def model_registry(self) -> ModelRepository:
        return self._model_registry

    def get_model_from_registry(self, name: str) -> BaseModel:
        model = self._model_registry.get_model(name)
        if model is None:
            raise ModelNotFound(name)
        return model

    def get_model(self, name: str) -> BaseModel:
        model = self.get_model_from_registry(name)
        if not self._model_registry.is_model_ready(name):
            raise ModelNotReady(name)
        return model

    # -- server metadata ---------------------------------------------------
    async def metadata(self) -> Dict[str, Any]:
        return {
            "name": self._server_name,
            "version": self._server_version,
            "extensions": [
                "model_repository_extension",
                "binary_tensor_data_extension",
            ],
        }

    async def model_metadata(self, model_name: str) -> Dict[str, Any]:
        model = self.get_model_from_registry(model_name)
        inputs, outputs = [], []
        if isinstance(model, InferenceModel):
            inputs = await model.get_input_types()
            outputs = await model.get_output_types()
        return {
            "name": model_name,
            "platform": "kserve_amd",
            "inputs": inputs,
            "outputs": outputs,
        }

    # -- health -------------------------------------------------------------
    async def live(self) -> bool:
        return True

    async def ready(self) -> bool:
        models = self._model_registry.get_models().values()
        return all(m.ready for m in models) if models else True

    async def model_ready(self, model_name: str) -> bool:
        if self._model_registry.get_model(model_name) is None:
            raise ModelNotFound(model_name)
        return self._model_registry.is_model_ready(model_name)

    # -- decode/encode -------------------------------------------------------
    def decode(
        self,
        body: Union[bytes, Dict],
        headers: Optional[Dict[str, str]] = None,
        model_name: str = "",
    ) -> Tuple[Union[Dict, InferRequest], Dict[str, Any]]:
        """Bytes/dict -> dict or InferRequest (reference dataplane.py:332-366)."""
        attributes: Dict[str, Any] = {}
        if isinstance(body, dict):
            return self._maybe_unwrap_cloudevent(body, attributes), attributes
        if not isinstance(body, (bytes, bytearray)):
            return body, attributes
        headers = headers or {}
        json_length = headers.get(constants.INFERENCE_CONTENT_LENGTH_HEADER)
        if json_length is not None:
            # V2 binary tensor extension (reference :393-405)
            try:
                req = InferRequest.from_bytes(
                    bytes(body), int(json_length), model_name
                )
            except ValueError as e:
                raise InvalidInput(str(e))
            return req, attributes
        if len(body) == 0:
            return {}, attributes
        try:
            decoded = json.loads(body)
        except json.JSONDecodeError as e:
            raise InvalidInput(f"Unrecognized request format: {e}")
        return self._maybe_unwrap_cloudevent(decoded, attributes), attributes

    @staticmethod
    def _maybe_unwrap_cloudevent(decoded: Dict, attributes: Dict[str, Any]):
        """Structured-mode CloudEvent unwrap (reference :128-161, simplified:
        structured JSON events only)."""
        if (
            isinstance(decoded, dict)
            and "specversion" in decoded
            and "data" in decoded
            and "type" in decoded
        ):
            for k, v in decoded.items():
                if k != "data":
                    attributes[k] = v
            return decoded["data"]
        return decoded

    def encode(
        self,
        model_name: str,
        response: Any,
        headers: Optional[Dict[str, str]] = None,
        req_attributes: Optional[Dict[str, Any]] = None,
    ) -> Tuple[Any, Dict[str, str]]:
        """Response object -> wire form + response headers
        (reference dataplane.py:407-437)."""
        response_headers: Dict[str, str] = {}
        if isinstance(response, InferResponse):
            requested = None
            body, json_length = response.to_rest(requested)
            if json_length is not None:
                response_headers[constants.INFERENCE_CONTENT_LENGTH_HEADER] = str(
                    json_length
                )
                response_headers["content-type"] = "application/octet-stream"
            return body, response_headers
        return response, response_headers

    # -- infer/explain -------------------------------------------------------
    async def infer(
        self,
        model_name: str,
        request: Union[Dict, InferRequest],
        headers: Optional[Dict[str, str]] = None,
    ):
        model = self.get_model(model_name)
        return await model(request, headers=headers, verb="predict")

    async def explain(
        self,
        model_name: str,
        request: Union[Dict, InferRequest],
        headers: Optional[Dict[str, str]] = None,
    ):
        model = self.get_model(model_name)
        return await model(request, headers=headers, verb="explain")


def _package_version() -> str:
    try:
        from kserve_amd import __version__

        return __version__
    except Exception:
        return "0.1.0"
