"""OpenAI REST routes.

Reference parity: python/kserve protocol/rest/openai/endpoints.py:52-301 —
/openai/v1/{completions,chat/completions,embeddings,rerank,models,models/{m}}
with SSE streaming. Also mounted at /v1/* aliases (completions and
chat/completions do not collide with the V1 predict routes).
"""

from __future__ import annotations

import json
from typing import AsyncIterator, Dict, List, Optional, Union

from fastapi import APIRouter, Request
from fastapi.responses import JSONResponse, Response, StreamingResponse

from kserve_amd.errors import InvalidInput, ModelNotFound
from kserve_amd.logging import logger
from kserve_amd.model import OpenAIModel
from kserve_amd.protocol.dataplane import DataPlane
from kserve_amd.protocol.rest.openai.types import (
    ChatCompletion,
    ChatCompletionRequest,
    Completion,
    CompletionRequest,
    EmbeddingRequest,
    ModelCard,
    ModelList,
    OpenAIErrorResponse,
    RerankRequest,
)


def create_error_response(message: str, status_code: int = 400) -> JSONResponse:
    return JSONResponse(
        status_code=status_code,
        content={"error": OpenAIErrorResponse(message=message, code=status_code).model_dump()},
    )


async def with_cancellation(handler_coro, raw_request: Request):
    """Race the handler against client disconnect (reference: vLLM's
    with_cancellation decorator applied to every OpenAI route,
    openai/config.py route registration). If the client goes away while
    the request is still generating, the handler task is cancelled;
    engine streams abort themselves from their finally blocks, so the
    scheduler stops spending GPU on the dead request. Returns the
    handler result, or a 499 response on disconnect."""
    import asyncio
    import contextlib

    handler = asyncio.ensure_future(handler_coro)

    async def watch_disconnect():
        # the body is already consumed, so the only message left on the
        # ASGI receive channel is http.disconnect
        while True:
            message = await raw_request.receive()
            if message["type"] == "http.disconnect":
                return

    watcher = asyncio.ensure_future(watch_disconnect())
    try:
        done, _ = await asyncio.wait(
            {handler, watcher}, return_when=asyncio.FIRST_COMPLETED
        )
        if handler in done:
            return handler.result()
        handler.cancel()
        with contextlib.suppress(asyncio.CancelledError):
            await handler
        return Response(status_code=499)  # client closed request
    finally:
        watcher.cancel()
        with contextlib.suppress(asyncio.CancelledError):
            await watcher


class OpenAIEndpoints:
    def __init__(self, dataplane: DataPlane, models: List[OpenAIModel]):
        self.dataplane = dataplane
        # a model may serve several ids (base name + LoRA adapter names)
        self._models: Dict[str, OpenAIModel] = {}
        for m in models:
            for name in getattr(m, "served_names", [m.name]):
                self._models[name] = m

    def _get_model(self, name: str) -> OpenAIModel:
        model = self._models.get(name)
        if model is None:
            raise ModelNotFound(name)
        return model

    # -- completions (reference endpoints.py:57-99) ------------------------
    async def create_completion(self, raw_request: Request):
        body = await raw_request.json()
        try:
            request = CompletionRequest.model_validate(body)
        except Exception as e:
            return create_error_response(f"Invalid completion request: {e}")
        try:
            model = self._get_model(request.model)
        except ModelNotFound as e:
            return create_error_response(str(e), 404)
        result = await with_cancellation(
            model.create_completion(request, raw_request), raw_request
        )
        if isinstance(result, Response) and result.status_code == 499:
            return result
        if hasattr(result, "__anext__"):
            return StreamingResponse(
                _sse(result), media_type="text/event-stream"
            )
        if isinstance(result, JSONResponse):
            return result
        return JSONResponse(content=result.model_dump(exclude_none=True))

    async def create_chat_completion(self, raw_request: Request):
        body = await raw_request.json()
        try:
            request = ChatCompletionRequest.model_validate(body)
        except Exception as e:
            return create_error_response(f"Invalid chat completion request: {e}")
        try:
            model = self._get_model(request.model)
        except ModelNotFound as e:
            return create_error_response(str(e), 404)
        result = await with_cancellation(
            model.create_chat_completion(request, raw_request), raw_request
        )
        if isinstance(result, Response) and result.status_code == 499:
            return result
        if hasattr(result, "__anext__"):
            return StreamingResponse(_sse(result), media_type="text/event-stream")
        if isinstance(result, JSONResponse):
            return result
        return JSONResponse(content=result.model_dump(exclude_none=True))

    async def create_embedding(self, raw_request: Request):
        body = await raw_request.json()
        try:
            request = EmbeddingRequest.model_validate(body)
        except Exception as e:
            return create_error_response(f"Invalid embedding request: {e}")
        try:
            model = self._get_model(request.model)
        except ModelNotFound as e:
            return create_error_response(str(e), 404)
        result = await with_cancellation(
            model.create_embedding(request, raw_request), raw_request
        )
        if isinstance(result, Response) and result.status_code == 499:
            return result
        if isinstance(result, JSONResponse):
            return result
        return JSONResponse(content=result.model_dump(exclude_none=True))

    async def create_rerank(self, raw_request: Request):
        body = await raw_request.json()
        try:
            request = RerankRequest.model_validate(body)
        except Exception as e:
            return create_error_response(f"Invalid rerank request: {e}")
        try:
            model = self._get_model(request.model)
        except ModelNotFound as e:
            return create_error_response(str(e), 404)
        result = await with_cancellation(
            model.create_rerank(request, raw_request), raw_request
        )
        if isinstance(result, Response) and result.status_code == 499:
            return result
        if isinstance(result, JSONResponse):
            return result
        return JSONResponse(content=result.model_dump(exclude_none=True))

    async def models(self):
        cards = []
        for name in self._models:
            cards.append(ModelCard(id=name))
        return JSONResponse(content=ModelList(data=cards).model_dump())

    async def get_model(self, model_name: str):
        if model_name not in self._models:
            return create_error_response(f"Model {model_name} not found", 404)
        return JSONResponse(content=ModelCard(id=model_name).model_dump())


async def _sse(gen) -> AsyncIterator[str]:
    """Serialize an async generator of pydantic chunks / dicts / raw strings
    as server-sent events."""
    try:
        async for chunk in gen:
            if isinstance(chunk, str):
                yield f"data: {chunk}\n\n"
            elif isinstance(chunk, bytes):
                yield b"data: " + chunk + b"\n\n"
            elif hasattr(chunk, "model_dump_json"):
                yield f"data: {chunk.model_dump_json(exclude_none=True)}\n\n"
            else:
                yield f"data: {json.dumps(chunk)}\n\n"
    except Exception as e:  # stream already started; emit error event
        logger.exception("Streaming error")
        yield f"data: {json.dumps({'error': str(e)})}\n\n"
    yield "data: [DONE]\n\n"


def register_openai_endpoints(app, dataplane: DataPlane, models: List[OpenAIModel]):
    """Routes per reference endpoints.py:260-301 (+ bare /v1 aliases used by
    OpenAI SDK defaults)."""
    ep = OpenAIEndpoints(dataplane, models)
    router = APIRouter(tags=["OpenAI"])
    for prefix in ("/openai/v1", "/v1"):
        router.add_api_route(
            f"{prefix}/completions", ep.create_completion, methods=["POST"]
        )
        router.add_api_route(
            f"{prefix}/chat/completions", ep.create_chat_completion, methods=["POST"]
        )
        router.add_api_route(
            f"{prefix}/embeddings", ep.create_embedding, methods=["POST"]
        )
        router.add_api_route(f"{prefix}/rerank", ep.create_rerank, methods=["POST"])
    router.add_api_route("/openai/v1/models", ep.models, methods=["GET"])
    router.add_api_route("/openai/v1/models/{model_name}", ep.get_model, methods=["GET"])
    app.include_router(router)
    return ep
