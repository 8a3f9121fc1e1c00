"""ModelServer: process entry for the data plane.

Reference parity: python/kserve/kserve/model_server.py:219-461 — arg parser,
model registration with at-least-one-ready check, engine startup hook, REST +
gRPC servers, signal handling.
"""

from __future__ import annotations

import argparse
import asyncio
import signal
import sys
from typing import Dict, List, Optional, Union

from kserve_amd import constants
from kserve_amd.logging import configure_logging, logger
from kserve_amd.model import BaseModel
from kserve_amd.model_repository import ModelRepository
from kserve_amd.protocol.dataplane import DataPlane
from kserve_amd.protocol.rest.server import RESTServer, create_app


def build_arg_parser() -> argparse.ArgumentParser:
    """Shared CLI surface (reference model_server.py:48-208)."""
    parser = argparse.ArgumentParser(add_help=True, description="kserve-amd ModelServer")
    parser.add_argument("--http_port", default=constants.HTTP_PORT, type=int)
    parser.add_argument("--grpc_port", default=constants.GRPC_PORT, type=int)
    parser.add_argument("--workers", default=1, type=int)
    parser.add_argument("--max_threads", default=4, type=int)
    parser.add_argument("--model_name", default=constants.DEFAULT_MODEL_NAME, type=str)
    parser.add_argument("--model_dir", default=constants.MODEL_MOUNT_PATH, type=str)
    parser.add_argument("--enable_grpc", default=True, type=lambda x: str(x).lower() == "true")
    parser.add_argument("--enable_docs_url", default=False, type=lambda x: str(x).lower() == "true")
    parser.add_argument("--access_log", default=False, type=lambda x: str(x).lower() == "true")
    parser.add_argument("--enable_latency_logging", default=True,
                        type=lambda x: str(x).lower() == "true")
    # transformer mode
    parser.add_argument("--predictor_host", default=None, type=str)
    parser.add_argument("--predictor_protocol", default="v1", type=str)
    parser.add_argument("--predictor_use_ssl", default=False, type=lambda x: str(x).lower() == "true")
    parser.add_argument("--predictor_request_timeout_seconds", default=600, type=int)
    parser.add_argument("--predictor_request_retries", default=0, type=int)
    return parser


class ModelServer:
    def __init__(
        self,
        http_port: int = constants.HTTP_PORT,
        grpc_port: int = constants.GRPC_PORT,
        registered_models: Optional[ModelRepository] = None,
        enable_grpc: bool = True,
        enable_docs_url: bool = False,
        workers: int = 1,
        max_threads: int = 4,
        access_log: bool = False,
        enable_latency_logging: bool = True,
    ):
        configure_logging()
        from kserve_amd.model import set_latency_logging

        set_latency_logging(enable_latency_logging)
        self.http_port = http_port
        self.grpc_port = grpc_port
        self.workers = workers
        self.max_threads = max_threads
        self.enable_grpc = enable_grpc
        self.enable_docs_url = enable_docs_url
        self.access_log = access_log
        self.registered_models = registered_models or ModelRepository()
        self.dataplane = DataPlane(self.registered_models)
        self.app = create_app(self.dataplane, enable_docs=enable_docs_url)
        self._rest_server: Optional[RESTServer] = None
        self._grpc_server = None
        self._engine_tasks: List[asyncio.Task] = []
        self._custom_tasks: List = []

    # -- registration (reference :441-459) ---------------------------------
    def register_model(self, model: BaseModel, name: Optional[str] = None):
        if not (name or model.name):
            raise RuntimeError("Failed to register model: model name must be provided")
        self.registered_models.update_handle(model, name)
        logger.info("Registering model: %s", name or model.name)

    def _register_and_check(self, models: List[BaseModel]):
        if isinstance(models, dict):
            for name, m in models.items():
                self.register_model(m, name)
            models = list(models.values())
        else:
            for m in models:
                self.register_model(m)
        at_least_one_ready = any(m.ready or m.engine for m in models)
        if not at_least_one_ready and models:
            raise RuntimeError("At least one model must be ready (or own an engine)")
        return models

    def register_openai_routes(self, models: List[BaseModel]):
        from kserve_amd.model import OpenAIModel
        from kserve_amd.protocol.rest.openai.endpoints import (
            register_openai_endpoints,
        )

        openai_models = [m for m in models if isinstance(m, OpenAIModel)]
        if openai_models:
            register_openai_endpoints(self.app, self.dataplane, openai_models)

    # -- lifecycle ----------------------------------------------------------
    async def _serve(self, models: List[BaseModel]):
        models = self._register_and_check(models)
        self.register_openai_routes(models)

        loop = asyncio.get_running_loop()
        stop_event = asyncio.Event()
        for sig in (signal.SIGINT, signal.SIGTERM):
            try:
                loop.add_signal_handler(sig, stop_event.set)
            except NotImplementedError:
                pass

        # start model engines in the server loop (reference :454-455)
        for m in models:
            if m.engine:
                self._engine_tasks.append(asyncio.create_task(m.start_engine()))

        servers = []
        self._rest_server = RESTServer(
            self.app, http_port=self.http_port, access_log=self.access_log
        )
        servers.append(asyncio.create_task(self._rest_server.start()))

        if self.enable_grpc:
            from kserve_amd.protocol.grpc.server import GRPCServer

            self._grpc_server = GRPCServer(self.dataplane, port=self.grpc_port)
            servers.append(asyncio.create_task(self._grpc_server.start()))

        stop_task = asyncio.create_task(stop_event.wait())
        done, pending = await asyncio.wait(
            servers + [stop_task], return_when=asyncio.FIRST_COMPLETED
        )
        await self.stop()
        for t in pending:
            t.cancel()

    def start(self, models: List[BaseModel]):
        """Blocking entry (reference model_server.py:332-377)."""
        asyncio.run(self._serve(models))

    async def stop(self):
        for m in self.registered_models.get_models().values():
            try:
                m.stop()
            except Exception:
                logger.exception("Error stopping model %s", m.name)
        for t in self._engine_tasks:
            t.cancel()
        if self._rest_server is not None:
            await self._rest_server.stop()
        if self._grpc_server is not None:
            await self._grpc_server.stop()
