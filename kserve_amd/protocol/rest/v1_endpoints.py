"""V1 REST protocol (reference parity: python/kserve v1_endpoints.py:30-174).

Routes: GET /v1/models, GET /v1/models/{model_name},
POST /v1/models/{model_name}:predict, POST /v1/models/{model_name}:explain
"""

from __future__ import annotations

from typing import Optional

from fastapi import APIRouter, Request, Response

from kserve_amd.errors import ModelNotReady
from kserve_amd.protocol.dataplane import DataPlane


class V1Endpoints:
    def __init__(self, dataplane: DataPlane):
        self.dataplane = dataplane

    async def models(self):
        return {"models": list(self.dataplane.model_registry.get_models().keys())}

    async def model_ready(self, model_name: str):
        ready = await self.dataplane.model_ready(model_name)
        if not ready:
            raise ModelNotReady(model_name)
        return {"name": model_name, "ready": ready}

    async def predict(self, model_name: str, request: Request) -> Response:
        body = await request.body()
        headers = dict(request.headers.items())
        payload, attributes = self.dataplane.decode(body, headers, model_name)
        result = await self.dataplane.infer(model_name, payload, headers)
        response, response_headers = self.dataplane.encode(
            model_name, result, headers, attributes
        )
        if isinstance(response, (bytes, bytearray)):
            return Response(content=response, headers=response_headers)
        import json

        response_headers.setdefault("content-type", "application/json")
        return Response(content=json.dumps(response), headers=response_headers)

    async def explain(self, model_name: str, request: Request) -> Response:
        body = await request.body()
        headers = dict(request.headers.items())
        payload, attributes = self.dataplane.decode(body, headers, model_name)
        result = await self.dataplane.explain(model_name, payload, headers)
        response, response_headers = self.dataplane.encode(
            model_name, result, headers, attributes
        )
        import json

        response_headers.setdefault("content-type", "application/json")
        return Response(content=json.dumps(response), headers=response_headers)


def register_v1_endpoints(app, dataplane: DataPlane):
    """Route table per reference v1_endpoints.py:155-171."""
    v1 = V1Endpoints(dataplane)
    router = APIRouter(tags=["V1"])
    router.add_api_route("/v1/models", v1.models, methods=["GET"])
    router.add_api_route("/v1/models/{model_name}", v1.model_ready, methods=["GET"])
    # ':predict' / ':explain' verb-suffixed paths
    router.add_api_route(
        "/v1/models/{model_name}:predict", v1.predict, methods=["POST"]
    )
    router.add_api_route(
        "/v1/models/{model_name}:explain", v1.explain, methods=["POST"]
    )
    app.include_router(router)
    return v1
