"""V2 (Open Inference Protocol) REST endpoints.

Reference parity: python/kserve v2_endpoints.py:132-305 — server/model
metadata, health, infer (incl. binary tensor extension), repository
load/unload.
"""

from __future__ import annotations

import json
from typing import Optional

from fastapi import APIRouter, Request, Response

from kserve_amd import constants
from kserve_amd.errors import ModelNotReady
from kserve_amd.protocol.dataplane import DataPlane
from kserve_amd.protocol.infer_type import InferRequest, InferResponse


class V2Endpoints:
    def __init__(self, dataplane: DataPlane, model_repository_extension=None):
        self.dataplane = dataplane
        self.model_repository_extension = model_repository_extension

    # -- metadata ----------------------------------------------------------
    async def metadata(self):
        return await self.dataplane.metadata()

    async def health_live(self):
        return {"live": await self.dataplane.live()}

    async def health_ready(self):
        return {"ready": await self.dataplane.ready()}

    async def models(self):
        return {"models": list(self.dataplane.model_registry.get_models().keys())}

    async def model_metadata(self, model_name: str, model_version: Optional[str] = None):
        return await self.dataplane.model_metadata(model_name)

    async def model_ready(self, model_name: str, model_version: Optional[str] = None):
        ready = await self.dataplane.model_ready(model_name)
        if not ready:
            raise ModelNotReady(model_name)
        return {"name": model_name, "ready": ready}

    # -- infer (reference v2_endpoints.py:132-194) ---------------------------
    async def infer(
        self, model_name: str, request: Request, model_version: Optional[str] = None
    ) -> Response:
        body = await request.body()
        headers = dict(request.headers.items())
        payload, attributes = self.dataplane.decode(body, headers, model_name)
        if isinstance(payload, dict):
            payload = InferRequest.from_inference_request(model_name, payload)
        result = await self.dataplane.infer(model_name, payload, headers)
        if isinstance(result, InferResponse):
            requested = payload.request_outputs if isinstance(payload, InferRequest) else None
            response_body, json_length = result.to_rest(requested)
            if json_length is not None:
                return Response(
                    content=response_body,
                    headers={
                        constants.INFERENCE_CONTENT_LENGTH_HEADER: str(json_length),
                        "content-type": "application/octet-stream",
                    },
                )
            return Response(
                content=json.dumps(response_body),
                headers={"content-type": "application/json"},
            )
        return Response(
            content=json.dumps(result), headers={"content-type": "application/json"}
        )

    # -- repository extension ------------------------------------------------
    async def load(self, model_name: str):
        if self.model_repository_extension is not None:
            await self.model_repository_extension.load(model_name)
        else:
            ok = self.dataplane.model_registry.load(model_name)
            if not ok:
                from kserve_amd.errors import ModelNotFound

                raise ModelNotFound(model_name)
        return {"name": model_name, "load": True}

    async def unload(self, model_name: str):
        if self.model_repository_extension is not None:
            await self.model_repository_extension.unload(model_name)
        else:
            self.dataplane.model_registry.unload(model_name)
        return {"name": model_name, "unload": True}

    async def repository_index(self, request: Request):
        return self.dataplane.model_registry.index()


def register_v2_endpoints(app, dataplane: DataPlane, model_repository_extension=None):
    """Route table per reference v2_endpoints.py:236-305."""
    v2 = V2Endpoints(dataplane, model_repository_extension)
    router = APIRouter(tags=["V2"])
    router.add_api_route("/v2", v2.metadata, methods=["GET"])
    router.add_api_route("/v2/health/live", v2.health_live, methods=["GET"])
    router.add_api_route("/v2/health/ready", v2.health_ready, methods=["GET"])
    router.add_api_route("/v2/models", v2.models, methods=["GET"])
    router.add_api_route("/v2/models/{model_name}", v2.model_metadata, methods=["GET"])
    router.add_api_route(
        "/v2/models/{model_name}/ready", v2.model_ready, methods=["GET"]
    )
    router.add_api_route(
        "/v2/models/{model_name}/infer", v2.infer, methods=["POST"]
    )
    router.add_api_route(
        "/v2/models/{model_name}/versions/{model_version}/infer",
        v2.infer,
        methods=["POST"],
    )
    router.add_api_route(
        "/v2/models/{model_name}/versions/{model_version}/ready",
        v2.model_ready,
        methods=["GET"],
    )
    router.add_api_route(
        "/v2/repository/models/{model_name}/load", v2.load, methods=["POST"]
    )
    router.add_api_route(
        "/v2/repository/models/{model_name}/unload", v2.unload, methods=["POST"]
    )
    router.add_api_route(
        "/v2/repository/index", v2.repository_index, methods=["POST"]
    )
    app.include_router(router)
    return v2
