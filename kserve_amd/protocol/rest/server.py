"""REST server: FastAPI app factory + uvicorn runner.

Reference parity: python/kserve protocol/rest/server.py:69-189 — root route,
/metrics, V1+V2+OpenAI registration, timing middleware, exception handler
table.
"""

from __future__ import annotations

import json
import time
from typing import Optional

from fastapi import FastAPI, Request, Response
from fastapi.responses import JSONResponse
from prometheus_client import CONTENT_TYPE_LATEST, generate_latest

from kserve_amd import errors as kerr
from kserve_amd.logging import logger, trace_logger
from kserve_amd.protocol.dataplane import DataPlane
from kserve_amd.protocol.rest.v1_endpoints import register_v1_endpoints
from kserve_amd.protocol.rest.v2_endpoints import register_v2_endpoints


def _error_json(status: int, message: str) -> JSONResponse:
    return JSONResponse(status_code=status, content={"error": message})


def install_exception_handlers(app: FastAPI):
    """Exception table per reference rest/server.py."""

    @app.exception_handler(kerr.InvalidInput)
    async def invalid_input_handler(request, exc):
        return _error_json(400, str(exc))

    @app.exception_handler(kerr.ModelNotFound)
    async def model_not_found_handler(request, exc):
        return _error_json(404, str(exc))

    @app.exception_handler(kerr.ModelNotReady)
    async def model_not_ready_handler(request, exc):
        return _error_json(503, str(exc))

    @app.exception_handler(kerr.InferenceError)
    async def inference_error_handler(request, exc):
        return _error_json(500, str(exc))

    @app.exception_handler(kerr.UnsupportedProtocol)
    async def unsupported_protocol_handler(request, exc):
        return _error_json(400, str(exc))

    @app.exception_handler(kerr.EngineDead)
    async def engine_dead_handler(request, exc):
        return _error_json(500, "Engine is dead: " + str(exc))

    @app.exception_handler(NotImplementedError)
    async def not_implemented_handler(request, exc):
        return _error_json(501, str(exc) or "Not implemented")

    @app.exception_handler(Exception)
    async def generic_handler(request, exc):
        logger.exception("Unhandled server error")
        return _error_json(500, f"{type(exc).__name__}: {exc}")


class TimingMiddleware:
    """Per-route wall-time logging (reference rest/server.py:147-152)."""

    def __init__(self, app):
        self.app = app

    async def __call__(self, scope, receive, send):
        if scope["type"] != "http":
            await self.app(scope, receive, send)
            return
        start = time.perf_counter()
        try:
            await self.app(scope, receive, send)
        finally:
            elapsed = (time.perf_counter() - start) * 1000
            trace_logger.debug("%s %s %.2f ms", scope.get("method"), scope.get("path"), elapsed)


def create_app(
    dataplane: DataPlane,
    model_repository_extension=None,
    enable_docs: bool = False,
) -> FastAPI:
    app = FastAPI(
        title="kserve-amd",
        docs_url="/docs" if enable_docs else None,
        redoc_url=None,
        default_response_class=JSONResponse,
    )

    @app.get("/")
    async def root():
        return {}

    @app.get("/metrics")
    async def metrics():
        return Response(content=generate_latest(), media_type=CONTENT_TYPE_LATEST)

    register_v1_endpoints(app, dataplane)
    register_v2_endpoints(app, dataplane, model_repository_extension)
    from kserve_amd.protocol.rest.timeseries import register_timeseries_endpoints

    register_timeseries_endpoints(app, dataplane)
    install_exception_handlers(app)
    return app


class RESTServer:
    """uvicorn wrapper (single process)."""

    def __init__(
        self,
        app: FastAPI,
        http_port: int = 8080,
        host: str = "0.0.0.0",
        access_log: bool = False,
        workers: int = 1,
    ):
        import uvicorn

        self.app = app
        cfg = uvicorn.Config(
            app,
            host=host,
            port=http_port,
            log_config=None,
            access_log=access_log,
        )
        self.server = uvicorn.Server(cfg)

    async def start(self):
        await self.server.serve()

    async def stop(self):
        self.server.should_exit = True
