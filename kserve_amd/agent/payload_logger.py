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
    """JSON (pass-through), CSV (instances/predictions flattening) or
    Parquet (pyarrow table per batch; reference marshaller_parquet.go)."""

    def __init__(self, fmt: str = "json"):
        assert fmt in ("json", "csv", "parquet")
        self.fmt = fmt

    def marshal(self, entry: LogEntry) -> bytes:
        if self.fmt == "json":
            return entry.payload
        if self.fmt == "parquet":
            return self.marshal_batch([entry])
        doc = json.loads(entry.payload)
        rows = doc.get("instances") or doc.get("predictions") or []
        buf = io.StringIO()
        w = csv.writer(buf)
        for row in rows:
            w.writerow(row if isinstance(row, (list, tuple)) else [row])
        return buf.getvalue().encode()

    def marshal_batch(self, entries: List["LogEntry"]) -> bytes:
        """One file for many entries (size/timed batch strategies). Parquet
        columns: request_id, event_type, model_name, timestamp, payload."""
        if self.fmt == "parquet":
            import pyarrow as pa
            import pyarrow.parquet as pq

            table = pa.table(
                {
                    "request_id": [e.request_id for e in entries],
                    "event_type": [e.event_type for e in entries],
                    "model_name": [e.model_name for e in entries],
                    "timestamp": [e.timestamp for e in entries],
                    "payload": [e.payload.decode("utf-8", "replace")
                                for e in entries],
                }
            )
            sink = io.BytesIO()
            pq.write_table(table, sink)
            return sink.getvalue()
        if self.fmt == "csv":
            return b"".join(self.marshal(e) for e in entries)
        # json lines
        return b"\n".join(e.payload for e in entries)

    @property
    def extension(self) -> str:
        return {"json": ".json", "csv": ".csv", "parquet": ".parquet"}[self.fmt]


class BlobStore:
    """Scheme-dispatched sink for marshalled payload files (reference
    pkg/logger/store.go:86-104 NewStoreForScheme): file paths write
    locally, s3:// uploads via the native S3 client."""

    def __init__(self, base_uri: str, s3_client=None):
        self.base_uri = base_uri
        self.scheme = (
            base_uri.split("://", 1)[0] if "://" in base_uri else "file"
        )
        if self.scheme == "s3":
            if s3_client is None:
                from kserve_amd.storage.http_providers import S3Client

                s3_client = S3Client()
            self._s3 = s3_client
            rest = base_uri[len("s3://"):]
            self._bucket, _, self._prefix = rest.partition("/")
        elif self.scheme == "file":
            self._dir = base_uri[len("file://"):] if "://" in base_uri else base_uri
        else:
            raise ValueError(f"unsupported log store scheme {self.scheme!r}")

    def put(self, name: str, data: bytes) -> None:
        if self.scheme == "s3":
            key = f"{self._prefix.rstrip('/')}/{name}".lstrip("/")
            self._s3.put_object(self._bucket, key, data)
        else:
            os.makedirs(self._dir, exist_ok=True)
            with open(os.path.join(self._dir, name), "wb") as f:
                f.write(data)


class BatchStrategy:
    """immediate | size | timed flush policies for the store sink
    (reference batch_{immediate,size,timed}.go)."""

    def __init__(self, kind: str = "immediate", size: int = 16,
                 interval_s: float = 5.0):
        assert kind in ("immediate", "size", "timed")
        self.kind = kind
        self.size = size
        self.interval_s = interval_s
        self._buf: List[LogEntry] = []
        self._last_flush = time.time()

    def add(self, entry: LogEntry) -> Optional[List[LogEntry]]:
        """Returns a batch to flush, or None to keep buffering."""
        if self.kind == "immediate":
            return [entry]
        self._buf.append(entry)
        if self.kind == "size" and len(self._buf) >= self.size:
            return self.drain()
        if self.kind == "timed" and time.time() - self._last_flush >= self.interval_s:
            return self.drain()
        return None

    def drain(self) -> Optional[List[LogEntry]]:
        if not self._buf:
            return None
        out, self._buf = self._buf, []
        self._last_flush = time.time()
        return out


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
        store: Optional[BlobStore] = None,
        batch: Optional[BatchStrategy] = None,
    ):
        self.url = url
        self.store_path = store_path
        self.mode = LogMode(mode)
        self.marshaller = Marshaller(marshaller)
        self.source = source
        self.num_workers = num_workers
        # blob-store sink: explicit store, or one derived from store_path
        # (s3:// uris route through the native S3 client)
        if store is None and store_path and "://" in store_path:
            store = BlobStore(store_path)
            self.store_path = None
        self.store = store
        self.batch = batch or BatchStrategy("immediate")
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
        if self.store is not None:
            ready = self.batch.add(entry)
            if ready:
                self._flush_batch(ready)

    def _flush_batch(self, entries: List[LogEntry]) -> None:
        data = self.marshaller.marshal_batch(entries)
        name = (
            f"{entries[0].model_name}-{int(entries[0].timestamp * 1000)}"
            f"-{uuid.uuid4().hex[:8]}-{len(entries)}{self.marshaller.extension}"
        )
        self.store.put(name, data)

    async def flush(self) -> None:
        """Drain the batch buffer (timed strategies / shutdown)."""
        ready = self.batch.drain()
        if ready:
            self._flush_batch(ready)
