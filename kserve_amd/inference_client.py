"""Async inference clients used by transformers, the graph router, and tests.

Reference parity: python/kserve/kserve/inference_client.py —
InferenceRESTClient (:390-707: V1+V2 url construction, binary tensor
support, retries, health) and InferenceGRPCClient (:61-333).
"""

from __future__ import annotations

import json
from typing import Any, Dict, List, Optional, Union

import httpx

from kserve_amd import constants
from kserve_amd.errors import InferenceError
from kserve_amd.protocol.infer_type import InferRequest, InferResponse


class RESTConfig:
    def __init__(
        self,
        protocol: str = "v1",
        retries: int = 3,
        timeout: float = 600,
        verify: bool = True,
        transport: Optional[httpx.AsyncBaseTransport] = None,
    ):
        self.protocol = protocol
        self.retries = retries
        self.timeout = timeout
        self.verify = verify
        self.transport = transport


class InferenceRESTClient:
    def __init__(self, config: Optional[RESTConfig] = None):
        self._config = config or RESTConfig()
        self._client = httpx.AsyncClient(
            timeout=self._config.timeout,
            verify=self._config.verify,
            transport=self._config.transport
            or httpx.AsyncHTTPTransport(retries=self._config.retries),
        )

    async def close(self):
        await self._client.aclose()

    def _url(self, base_url: str, model_name: str, verb: str) -> str:
        base = str(base_url).rstrip("/")
        if self._config.protocol == "v2":
            return f"{base}/v2/models/{model_name}/{verb}"
        return f"{base}/v1/models/{model_name}:{verb}"

    # -- health ------------------------------------------------------------
    async def is_server_ready(self, base_url: str) -> bool:
        r = await self._client.get(f"{str(base_url).rstrip('/')}/v2/health/ready")
        return r.status_code == 200 and r.json().get("ready", False)

    async def is_server_live(self, base_url: str) -> bool:
        r = await self._client.get(f"{str(base_url).rstrip('/')}/v2/health/live")
        return r.status_code == 200 and r.json().get("live", False)

    async def is_model_ready(self, base_url: str, model_name: str) -> bool:
        base = str(base_url).rstrip("/")
        if self._config.protocol == "v2":
            r = await self._client.get(f"{base}/v2/models/{model_name}/ready")
        else:
            r = await self._client.get(f"{base}/v1/models/{model_name}")
        return r.status_code == 200

    # -- infer ----------------------------------------------------------------
    async def infer(
        self,
        base_url: str,
        data: Union[Dict, InferRequest],
        model_name: str,
        headers: Optional[Dict[str, str]] = None,
    ) -> Union[Dict, InferResponse]:
        headers = dict(headers or {})
        if isinstance(data, InferRequest):
            body, json_len = data.to_rest()
            url = self._url(base_url, model_name, "infer")
            if json_len is not None:
                headers[constants.INFERENCE_CONTENT_LENGTH_HEADER] = str(json_len)
                headers["content-type"] = "application/octet-stream"
                r = await self._client.post(url, content=body, headers=headers)
            else:
                r = await self._client.post(url, json=body, headers=headers)
            self._check(r)
            cl = r.headers.get(constants.INFERENCE_CONTENT_LENGTH_HEADER)
            if cl is not None:
                return InferResponse.from_bytes(r.content, int(cl))
            return InferResponse.from_rest(model_name, r.json())
        url = self._url(base_url, model_name, "predict")
        r = await self._client.post(url, json=data, headers=headers)
        self._check(r)
        return r.json()

    async def explain(
        self,
        base_url: str,
        model_name: str,
        data: Dict,
        headers: Optional[Dict[str, str]] = None,
    ) -> Dict:
        url = self._url(base_url, model_name, "explain")
        r = await self._client.post(url, json=data, headers=headers)
        self._check(r)
        return r.json()

    @staticmethod
    def _check(r: httpx.Response):
        if r.status_code >= 400:
            raise InferenceError(
                f"HTTP {r.status_code}: {r.text[:500]}", status=str(r.status_code)
            )


class InferenceGRPCClient:
    """Thin async client over the programmatic V2 proto."""

    def __init__(self, url: str, timeout: float = 60.0):
        import grpc

        from kserve_amd.protocol.grpc import proto

        self._proto = proto
        self._channel = grpc.aio.insecure_channel(url)
        self._timeout = timeout

    def _method(self, name, req_cls, resp_cls):
        return self._channel.unary_unary(
            f"/{self._proto.SERVICE_NAME}/{name}",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=resp_cls.FromString,
        )

    async def close(self):
        await self._channel.close()

    async def is_server_ready(self) -> bool:
        p = self._proto
        resp = await self._method("ServerReady", p.ServerReadyRequest, p.ServerReadyResponse)(
            p.ServerReadyRequest(), timeout=self._timeout
        )
        return resp.ready

    async def is_model_ready(self, model_name: str) -> bool:
        p = self._proto
        resp = await self._method("ModelReady", p.ModelReadyRequest, p.ModelReadyResponse)(
            p.ModelReadyRequest(name=model_name), timeout=self._timeout
        )
        return resp.ready

    async def infer(
        self, infer_request: InferRequest, timeout: Optional[float] = None
    ) -> InferResponse:
        from kserve_amd.protocol.grpc.servicer import (
            _dict_to_params,
            serialize_bytes_tensor,
        )
        import numpy as np

        p = self._proto
        req = p.ModelInferRequest()
        req.model_name = infer_request.model_name
        req.id = infer_request.id or ""
        for inp in infer_request.inputs:
            t = req.inputs.add()
            t.name = inp.name
            t.datatype = inp.datatype
            t.shape.extend(int(s) for s in inp.shape)
            _dict_to_params(
                {k: v for k, v in inp.parameters.items() if k != "binary_data_size"},
                t.parameters,
            )
            if inp.raw_data is not None:
                req.raw_input_contents.append(inp.raw_data)
            else:
                arr = inp.as_numpy()
                if inp.datatype == "BYTES":
                    req.raw_input_contents.append(serialize_bytes_tensor(arr))
                else:
                    req.raw_input_contents.append(
                        np.ascontiguousarray(arr).tobytes()
                    )
        resp = await self._method("ModelInfer", p.ModelInferRequest, p.ModelInferResponse)(
            req, timeout=timeout or self._timeout
        )
        from kserve_amd.protocol.infer_type import InferOutput

        outputs = []
        for i, o in enumerate(resp.outputs):
            out = InferOutput(o.name, list(o.shape), o.datatype)
            if i < len(resp.raw_output_contents):
                out.set_raw_data(resp.raw_output_contents[i])
            outputs.append(out)
        return InferResponse(
            resp.id, resp.model_name, outputs, model_version=resp.model_version or None
        )
