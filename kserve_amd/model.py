"""Model ABCs for the data plane.

Reference parity: python/kserve/kserve/model.py — ``BaseKServeModel`` lifecycle
(:69-121), ``Model.__call__`` pre/predict/post pipeline with per-stage metrics
(:198-284), transformer-mode forwarding to a downstream predictor (:393-456).
Re-designed: async-first, no sync/async dual paths.
"""

from __future__ import annotations

import inspect
import time
from enum import Enum
from typing import Any, AsyncIterator, Dict, Optional, Union

import httpx

from kserve_amd import constants
from kserve_amd.errors import InvalidInput
from kserve_amd.logging import logger, trace_logger

# per-request latency trace lines (reference ModelServer
# --enable_latency_logging); module-level so ModelServer's flag reaches
# every Model instance
_LATENCY_LOGGING = True


def set_latency_logging(enabled: bool) -> None:
    global _LATENCY_LOGGING
    _LATENCY_LOGGING = enabled


def enable_latency_logging() -> bool:
    return _LATENCY_LOGGING
from kserve_amd.metrics import get_labeled_histograms
from kserve_amd.protocol.infer_type import InferRequest, InferResponse


class PredictorProtocol(Enum):
    REST_V1 = "v1"
    REST_V2 = "v2"
    GRPC_V2 = "grpc-v2"


class BaseModel:
    """Lifecycle base: load/start/stop/healthy (reference model.py:69-121)."""

    def __init__(self, name: str):
        self.name = name
        self.ready = False
        # engine hook: models owning a background engine (the native LLM
        # engine) set this; ModelServer awaits start_engine() at startup
        # (reference model_server.py:441-459).
        self.engine = False

    def load(self) -> bool:
        """Synchronous weight load; sets self.ready."""
        self.ready = True
        return self.ready

    async def start_engine(self) -> None:
        """Start a background engine loop (native LLM engine)."""

    def start(self):
        pass

    def stop(self):
        self.ready = False

    async def healthy(self) -> bool:
        return self.ready


class InferenceModel(BaseModel):
    """V1/V2 inference-capable model."""

    def __init__(self, name: str):
        super().__init__(name)

    async def get_input_types(self):
        return []

    async def get_output_types(self):
        return []


class PredictorConfig:
    """Downstream predictor target for transformer mode
    (reference: predictor_config.py:20)."""

    def __init__(
        self,
        predictor_host: str,
        predictor_protocol: str = PredictorProtocol.REST_V1.value,
        predictor_use_ssl: bool = False,
        predictor_request_timeout_seconds: int = 600,
        predictor_request_retries: int = 0,
    ):
        self.predictor_host = predictor_host
        self.predictor_protocol = predictor_protocol
        self.predictor_use_ssl = predictor_use_ssl
        self.predictor_request_timeout_seconds = predictor_request_timeout_seconds
        self.predictor_request_retries = predictor_request_retries

    @property
    def predictor_base_url(self) -> str:
        scheme = "https" if self.predictor_use_ssl else "http"
        return f"{scheme}://{self.predictor_host}"


class Model(InferenceModel):
    """User-facing model: preprocess -> predict -> postprocess with per-stage
    Prometheus histograms (reference model.py:198-284)."""

    def __init__(self, name: str, predictor_config: Optional[PredictorConfig] = None):
        super().__init__(name)
        self.predictor_config = predictor_config
        self._histograms = get_labeled_histograms(name)
        self._http_client: Optional[httpx.AsyncClient] = None

    # -- pipeline ----------------------------------------------------------
    async def __call__(
        self,
        body: Union[Dict, InferRequest],
        headers: Optional[Dict[str, str]] = None,
        verb: str = "predict",
    ):
        request_id = (headers or {}).get(constants.REQUEST_ID_HEADER, "N.A.")
        t0 = time.perf_counter()
        payload = await self._maybe_await(self.preprocess(body, headers))
        t1 = time.perf_counter()
        self._histograms["preprocess"].observe(t1 - t0)
        payload = self.validate(payload)
        if verb == "explain":
            result = await self._maybe_await(self.explain(payload, headers))
            t2 = time.perf_counter()
            self._histograms["explain"].observe(t2 - t1)
        else:
            result = await self._maybe_await(self.predict(payload, headers))
            t2 = time.perf_counter()
            self._histograms["predict"].observe(t2 - t1)
        response = await self._maybe_await(self.postprocess(result, headers))
        t3 = time.perf_counter()
        self._histograms["postprocess"].observe(t3 - t2)
        if enable_latency_logging():
            trace_logger.info(
                "requestId: %s, preprocess_ms: %.3f, explain_ms: %.3f, "
                "predict_ms: %.3f, postprocess_ms: %.3f",
                request_id,
                (t1 - t0) * 1000,
                (t2 - t1) * 1000 if verb == "explain" else 0,
                (t2 - t1) * 1000 if verb != "explain" else 0,
                (t3 - t2) * 1000,
            )
        return response

    @staticmethod
    async def _maybe_await(v):
        if inspect.isawaitable(v):
            return await v
        return v

    # -- overridable stages ------------------------------------------------
    def preprocess(self, body, headers=None):
        return body

    def validate(self, payload):
        if isinstance(payload, InferRequest):
            return payload
        if isinstance(payload, dict):
            if "instances" in payload and not isinstance(payload["instances"], list):
                raise InvalidInput('Expected "instances" to be a list')
            if "inputs" in payload and not isinstance(payload["inputs"], list):
                raise InvalidInput('Expected "inputs" to be a list')
        return payload

    def predict(self, payload, headers=None):
        """Default predict: forward to the configured predictor host
        (transformer mode, reference model.py:393-456); otherwise override."""
        if self.predictor_config is None:
            raise NotImplementedError("predict() not implemented")
        return self._forward_predict(payload, headers, verb="predict")

    def explain(self, payload, headers=None):
        if self.predictor_config is None:
            raise NotImplementedError("explain() not implemented")
        return self._forward_predict(payload, headers, verb="explain")

    def postprocess(self, result, headers=None):
        return result

    # -- transformer-mode HTTP forwarding ----------------------------------
    def _forward_headers(self, headers: Optional[Dict[str, str]]) -> Dict[str, str]:
        out = {}
        for h in constants.FORWARDABLE_HEADERS:
            if headers and h in headers:
                out[h] = headers[h]
        return out

    async def _forward_predict(self, payload, headers, verb="predict"):
        cfg = self.predictor_config
        if self._http_client is None:
            self._http_client = httpx.AsyncClient(
                timeout=cfg.predictor_request_timeout_seconds,
                transport=httpx.AsyncHTTPTransport(
                    retries=cfg.predictor_request_retries
                ),
            )
        fwd = self._forward_headers(headers)
        if cfg.predictor_protocol == PredictorProtocol.REST_V2.value:
            if isinstance(payload, InferRequest):
                body, json_len = payload.to_rest()
            else:
                body, json_len = payload, None
            url = f"{cfg.predictor_base_url}/v2/models/{self.name}/infer"
            if json_len is not None:
                fwd[constants.INFERENCE_CONTENT_LENGTH_HEADER] = str(json_len)
                fwd["content-type"] = "application/octet-stream"
                r = await self._http_client.post(url, content=body, headers=fwd)
            else:
                r = await self._http_client.post(url, json=body, headers=fwd)
            r.raise_for_status()
            cl = r.headers.get(constants.INFERENCE_CONTENT_LENGTH_HEADER)
            if cl is not None:
                return InferResponse.from_bytes(r.content, int(cl))
            return r.json()
        # V1
        url = f"{cfg.predictor_base_url}/v1/models/{self.name}:{verb}"
        r = await self._http_client.post(url, json=payload, headers=fwd)
        r.raise_for_status()
        return r.json()


class OpenAIModel(BaseModel):
    """ABC for OpenAI-protocol models (reference: openai_model.py:42-134)."""

    async def create_completion(self, request, raw_request=None, context=None):
        raise NotImplementedError

    async def create_chat_completion(self, request, raw_request=None, context=None):
        raise NotImplementedError

    async def create_embedding(self, request, raw_request=None, context=None):
        raise NotImplementedError

    async def create_rerank(self, request, raw_request=None, context=None):
        raise NotImplementedError

    async def models(self):
        return [self.name]
