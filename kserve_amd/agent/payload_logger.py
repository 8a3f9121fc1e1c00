"""Async payload logging (CloudEvents).

Reference parity: pkg/logger — LoggerHandler (handler.go:76-92), worker pool
(worker.go:68-79), blob store + marshaller pipeline (store.go:86-104,
marshaller_{json,csv}.go, batch_{immediate,size,timed}.go). Python version:
asyncio queue + workers emitting binary-mode CloudEvents over HTTP and/or
writing marshalled files to a store path.
"""

from __future__ import annotations

import asyncio
import csv
import io
import json
import os
import time
import uuid
from dataclasses import dataclass, field
from enum import Enum
from typing import Any, Dict, List, Optional

import httpx

from kserve_amd.logging import logger


class LogMode(str, Enum):
    all = "all"
    request = "request"
    response = "response"


@dataclass
class LogEntry:
    request_id: str
    event_type: str  # "org.kubeflow.serving.inference.request" | ".response"
    model_name: str
    payload: bytes
    content_type: str = "application/json"
    metadata: Dict[str, str] = field(default_factory=dict)
    timestamp: float = field(default_factory=time.time)


class Marshaller:
    """JSON (pass-through) or CSV (instances/predictions flattening)."""

    def __init__(self, fmt: str = "json"):
        assert fmt in ("json", "csv")
        self.fmt = fmt

    def marshal(self, entry: LogEntry) -> bytes:
        if self.fmt == "json":
            return entry.payload
        doc = json.loads(entry.payload)
        rows = doc.get("instances") or doc.get("predictions") or []
        buf = io.StringIO()
        w = csv.writer(buf)
        for row in rows:
            w.writerow(row if isinstance(row, (list, tuple)) else [row])
        return buf.getvalue().encode()

    @property
    def extension(self) -> str:
        return ".json" if self.fmt == "json" else ".csv"


class PayloadLogger:
    """Reference agent logger roles: CloudEvents to ``url`` and/or files
    under ``store_path`` (the blob-store stand-in for offline use)."""

    def __init__(
        self,
        url: Optional[str] = None,
        store_path: Optional[str] = None,
        mode: LogMode = LogMode.all,
        marshaller: str = "json",
        num_workers: int = 2,
        source: str = "kserve-amd-agent",
        transport=None,
    ):
        self.url = url
        self.store_path = store_path
        self.mode = LogMode(mode)
        self.marshaller = Marshaller(marshaller)
        self.source = source
        self.num_workers = num_workers
        self._queue: asyncio.Queue = asyncio.Queue(maxsize=1024)
        self._workers: List[asyncio.Task] = []
        self._client = httpx.AsyncClient(transport=transport, timeout=10)

    # -- API ---------------------------------------------------------------
    def should_log(self, event_type: str) -> bool:
        if self.mode == LogMode.all:
            return True
        return self.mode.value in event_type

    async def log(self, entry: LogEntry):
        if not self.should_log(entry.event_type):
            return
        try:
            self._queue.put_nowait(entry)
        except asyncio.QueueFull:
            logger.warning("Payload log queue full; dropping entry")

    async def start(self):
        for _ in range(self.num_workers):
            self._workers.append(asyncio.create_task(self._worker()))

    async def stop(self):
        await self._queue.join()
        for w in self._workers:
            w.cancel()
        await self._client.aclose()

    # -- workers -----------------------------------------------------------
    async def _worker(self):
        while True:
            entry = await self._queue.get()
            try:
                await self._emit(entry)
            except Exception:
                logger.exception("Payload log emit failed")
            finally:
                self._queue.task_done()

    async def _emit(self, entry: LogEntry):
        if self.url:
            # binary-mode CloudEvent (ce-* headers), CloudEvents v1.0
            headers = {
                "ce-specversion": "1.0",
                "ce-id": str(uuid.uuid4()),
                "ce-type": entry.event_type,
                "ce-source": self.source,
                "ce-inferenceservicename": entry.model_name,
                "ce-requestid": entry.request_id,
                "content-type": entry.content_type,
                **{f"ce-{k}": v for k, v in entry.metadata.items()},
            }
            r = await self._client.post(self.url, content=entry.payload, headers=headers)
            if r.status_code >= 400:
                logger.warning("Payload log sink returned %d", r.status_code)
        if self.store_path:
            os.makedirs(self.store_path, exist_ok=True)
            kind = "request" if "request" in entry.event_type else "response"
            fname = f"{entry.request_id}-{kind}{self.marshaller.extension}"
            data = self.marshaller.marshal(entry)
            with open(os.path.join(self.store_path, fname), "wb") as f:
                f.write(data)
