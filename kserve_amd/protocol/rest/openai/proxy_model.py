"""OpenAI proxy model: forward /v1/completions and /v1/chat/completions to
an upstream OpenAI-compatible server, passing streams through.

Reference parity: python/kserve/kserve/protocol/rest/openai/
openai_proxy_model.py — the transformer-container pattern for generative
models (preprocess hook -> upstream predictor -> postprocess hook), with
SSE chunks relayed verbatim and upstream errors surfaced as OpenAIError
responses.
"""

from __future__ import annotations

import json
from typing import AsyncIterator, Dict, Optional

import httpx

from kserve_amd.logging import logger
from kserve_amd.model import OpenAIModel

COMPLETIONS_ENDPOINT = "/v1/completions"
CHAT_COMPLETIONS_ENDPOINT = "/v1/chat/completions"


class OpenAIProxyModel(OpenAIModel):
    """Proxy an OpenAI-protocol endpoint; subclass and override
    ``preprocess_completion_request`` / ``postprocess_*`` for transformer
    behavior."""

    def __init__(
        self,
        name: str,
        predictor_url: str,
        http_client: Optional[httpx.AsyncClient] = None,
        timeout: float = 300.0,
    ):
        super().__init__(name)
        self.predictor_url = predictor_url.rstrip("/")
        self._client = http_client or httpx.AsyncClient(
            timeout=httpx.Timeout(timeout)
        )
        self.ready = True

    # -- hooks (identity by default) ----------------------------------------
    async def preprocess_completion_request(self, body: Dict) -> Dict:
        return body

    async def postprocess_completion(self, body: Dict) -> Dict:
        return body

    async def preprocess_chat_completion_request(self, body: Dict) -> Dict:
        return body

    async def postprocess_chat_completion(self, body: Dict) -> Dict:
        return body

    # -- proxying ------------------------------------------------------------
    async def _relay_stream(self, response: httpx.Response) -> AsyncIterator[str]:
        """Pass upstream SSE data lines through (endpoints re-frame them)."""
        try:
            async for line in response.aiter_lines():
                if not line:
                    continue
                if line.startswith("data: "):
                    payload = line[len("data: "):]
                    if payload.strip() == "[DONE]":
                        break
                    yield payload
        finally:
            await response.aclose()

    async def _proxy(self, endpoint: str, body: Dict, pre, post):
        body = await pre(dict(body))
        url = self.predictor_url + endpoint
        if body.get("stream"):
            req = self._client.build_request("POST", url, json=body)
            response = await self._client.send(req, stream=True)
            if response.status_code != 200:
                text = (await response.aread()).decode(errors="replace")
                await response.aclose()
                from fastapi.responses import JSONResponse

                logger.warning("upstream %s -> %d", url, response.status_code)
                try:
                    return JSONResponse(
                        content=json.loads(text), status_code=response.status_code
                    )
                except Exception:
                    return JSONResponse(
                        content={"error": text}, status_code=response.status_code
                    )
            return self._relay_stream(response)
        response = await self._client.post(url, json=body)
        from fastapi.responses import JSONResponse

        if response.status_code != 200:
            logger.warning("upstream %s -> %d", url, response.status_code)
            try:
                return JSONResponse(
                    content=response.json(), status_code=response.status_code
                )
            except Exception:
                return JSONResponse(
                    content={"error": response.text},
                    status_code=response.status_code,
                )
        return JSONResponse(content=await post(response.json()))

    async def create_completion(self, request, raw_request=None, context=None):
        body = request.model_dump(exclude_none=True)
        result = await self._proxy(
            COMPLETIONS_ENDPOINT,
            body,
            self.preprocess_completion_request,
            self.postprocess_completion,
        )
        return result

    async def create_chat_completion(self, request, raw_request=None, context=None):
        body = request.model_dump(exclude_none=True)
        return await self._proxy(
            CHAT_COMPLETIONS_ENDPOINT,
            body,
            self.preprocess_chat_completion_request,
            self.postprocess_chat_completion,
        )

    async def healthy(self) -> bool:
        return self.ready

    async def close(self):
        await self._client.aclose()
