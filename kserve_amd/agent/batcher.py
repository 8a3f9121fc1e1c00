"""Micro-batching for V1 ``:predict`` requests.

Reference parity: pkg/batcher/handler.go — accumulate ``instances`` across
concurrent requests, flush on MaxBatchSize (32) or MaxLatency (5000 ms)
(:33-37 defaults, :157-188 batch loop, :99-155 batchPredict scatter).
Implemented as an asyncio accumulator usable standalone (agent proxy) or
in-process in front of a model.
"""

from __future__ import annotations

import asyncio
import json
import time
import uuid as _uuid
from dataclasses import dataclass, field
from typing import Any, Awaitable, Callable, Dict, List, Optional

import httpx
from fastapi import FastAPI, Request, Response

from kserve_amd.agent.payload_logger import LogEntry as _LogEntry
from kserve_amd.constants import DEFAULT_MAX_BATCH_SIZE, DEFAULT_MAX_LATENCY_MS
from kserve_amd.logging import logger


@dataclass
class _Pending:
    instances: List[Any]
    future: asyncio.Future = field(default_factory=asyncio.Future)


class Batcher:
    """Accumulates V1 instances; ``predict_fn(batch_instances)`` is invoked
    once per flush and predictions are scattered back per caller."""

    def __init__(
        self,
        predict_fn: Callable[[List[Any]], Awaitable[Dict]],
        max_batch_size: int = DEFAULT_MAX_BATCH_SIZE,
        max_latency_ms: int = DEFAULT_MAX_LATENCY_MS,
    ):
        self.predict_fn = predict_fn
        self.max_batch_size = max_batch_size
        self.max_latency_ms = max_latency_ms
        self._queue: List[_Pending] = []
        self._count = 0
        self._lock = asyncio.Lock()
        self._flush_task: Optional[asyncio.Task] = None

    async def predict(self, instances: List[Any]) -> Dict:
        if len(instances) >= self.max_batch_size:
            # oversized request: straight through
            return await self.predict_fn(instances)
        entry = _Pending(instances=instances)
        async with self._lock:
            self._queue.append(entry)
            self._count += len(instances)
            if self._count >= self.max_batch_size:
                await self._flush_locked()
            elif self._flush_task is None:
                self._flush_task = asyncio.create_task(self._deadline_flush())
        return await entry.future

    async def _deadline_flush(self):
        await asyncio.sleep(self.max_latency_ms / 1000.0)
        async with self._lock:
            await self._flush_locked()

    async def _flush_locked(self):
        if self._flush_task is not None:
            if self._flush_task is not asyncio.current_task():
                self._flush_task.cancel()
            self._flush_task = None
        batch = self._queue
        self._queue = []
        self._count = 0
        if not batch:
            return
        all_instances: List[Any] = []
        for p in batch:
            all_instances.extend(p.instances)
        t0 = time.perf_counter()
        try:
            result = await self.predict_fn(all_instances)
        except Exception as e:
            for p in batch:
                if not p.future.done():
                    p.future.set_exception(e)
            return
        elapsed_ms = (time.perf_counter() - t0) * 1000
        predictions = result.get("predictions", [])
        if len(predictions) != len(all_instances):
            err = RuntimeError(
                f"Batcher: {len(all_instances)} instances but "
                f"{len(predictions)} predictions"
            )
            for p in batch:
                if not p.future.done():
                    p.future.set_exception(err)
            return
        off = 0
        batch_id = f"batch-{int(time.time() * 1000)}"
        for p in batch:
            n = len(p.instances)
            resp = {
                "predictions": predictions[off : off + n],
                "batchId": batch_id,
                "instanceCount": n,
                "latencyMs": round(elapsed_ms, 3),
            }
            off += n
            if not p.future.done():
                p.future.set_result(resp)


def create_batcher_proxy_app(
    backend_url: str,
    model_name: str,
    max_batch_size: int = DEFAULT_MAX_BATCH_SIZE,
    max_latency_ms: int = DEFAULT_MAX_LATENCY_MS,
    transport=None,
    payload_logger=None,
):
    """Agent-style reverse proxy: batches ``POST /v1/models/{m}:predict``,
    passes everything else through, and (when a PayloadLogger is given)
    captures request/response payloads as CloudEvents exactly like the
    reference agent chain (cmd/agent/main.go:429-449: Drainer -> logger
    -> batcher -> reverse proxy)."""
    app = FastAPI()
    if payload_logger is not None:
        @app.on_event("startup")
        async def _start_logger():
            await payload_logger.start()
    client = httpx.AsyncClient(base_url=backend_url, transport=transport, timeout=60)

    async def call_backend(instances: List[Any]) -> Dict:
        r = await client.post(
            f"/v1/models/{model_name}:predict", json={"instances": instances}
        )
        r.raise_for_status()
        return r.json()

    batcher = Batcher(call_backend, max_batch_size, max_latency_ms)
    app.state.batcher = batcher
    # Drainer (outermost handler of the reference agent chain,
    # cmd/agent/main.go:429-449): readiness passes through to the user
    # container until drain starts; on drain, in-flight requests finish
    # while the probe reports not-ready so the endpoint is removed first.
    app.state.draining = False
    app.state.inflight = 0

    @app.get("/readyz")
    async def readyz():
        if app.state.draining:
            return Response(content="draining", status_code=503)
        try:
            r = await client.get("/")
            return Response(status_code=200 if r.status_code < 500 else 503)
        except Exception:
            return Response(content="backend unreachable", status_code=503)

    @app.post("/drain")
    async def drain():
        app.state.draining = True
        import asyncio as _aio

        # wait for in-flight work (bounded) before reporting drained
        for _ in range(300):
            if app.state.inflight == 0:
                break
            await _aio.sleep(0.1)
        return {"draining": True, "inflight": app.state.inflight}

    @app.post("/v1/models/{name}:predict")
    async def predict(name: str, request: Request):
        raw = await request.body()
        try:
            body = json.loads(raw)
        except json.JSONDecodeError:
            return Response(
                content=json.dumps({"error": "invalid JSON"}),
                status_code=400,
            )
        instances = body.get("instances")
        if not isinstance(instances, list):
            return Response(
                content=json.dumps({"error": "instances must be a list"}),
                status_code=400,
            )
        rid = request.headers.get("x-request-id") or str(_uuid.uuid4())
        if payload_logger is not None:
            await payload_logger.log(_LogEntry(
                request_id=rid,
                event_type="org.kubeflow.serving.inference.request",
                model_name=name,
                payload=raw,
            ))
        app.state.inflight += 1
        try:
            result = await batcher.predict(instances)
        finally:
            app.state.inflight -= 1
        if payload_logger is not None:
            await payload_logger.log(_LogEntry(
                request_id=rid,
                event_type="org.kubeflow.serving.inference.response",
                model_name=name,
                payload=json.dumps(result).encode(),
            ))
        return result

    @app.api_route(
        "/{path:path}", methods=["GET", "POST", "PUT", "DELETE"]
    )
    async def passthrough(path: str, request: Request):
        body = await request.body()
        r = await client.request(
            request.method,
            f"/{path}",
            content=body,
            headers={
                k: v
                for k, v in request.headers.items()
                if k.lower() not in ("host", "content-length")
            },
        )
        return Response(
            content=r.content,
            status_code=r.status_code,
            media_type=r.headers.get("content-type"),
        )

    return app
