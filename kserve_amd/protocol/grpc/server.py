"""grpc.aio server wired with generic handlers (no generated stubs).

Reference parity: python/kserve protocol/grpc/server.py:30-103.
"""

from __future__ import annotations

import grpc

from kserve_amd.logging import logger
from kserve_amd.protocol.dataplane import DataPlane
from kserve_amd.protocol.grpc import proto
from kserve_amd.protocol.grpc.servicer import InferenceServicer

MAX_GRPC_MESSAGE_LENGTH = 8388608  # 8 MiB default, same as reference


class GRPCServer:
    def __init__(
        self,
        dataplane: DataPlane,
        port: int = 8081,
        model_repository_extension=None,
        max_message_length: int = MAX_GRPC_MESSAGE_LENGTH,
    ):
        self.port = port
        self.servicer = InferenceServicer(dataplane, model_repository_extension)
        self._server = grpc.aio.server(
            options=[
                ("grpc.max_send_message_length", max_message_length),
                ("grpc.max_receive_message_length", max_message_length),
            ]
        )
        handlers = {}
        for method, (req_cls, resp_cls) in proto.SERVICE_METHODS.items():
            handlers[method] = grpc.unary_unary_rpc_method_handler(
                getattr(self.servicer, method),
                request_deserializer=req_cls.FromString,
                response_serializer=lambda m: m.SerializeToString(),
            )
        self._server.add_generic_rpc_handlers(
            (grpc.method_handlers_generic_handler(proto.SERVICE_NAME, handlers),)
        )
        self._bound = self._server.add_insecure_port(f"[::]:{port}")

    @property
    def bound_port(self) -> int:
        return self._bound

    async def start(self):
        await self._server.start()
        logger.info("gRPC server started on port %d", self._bound)
        await self._server.wait_for_termination()

    async def stop(self, grace: float = 5.0):
        await self._server.stop(grace)
