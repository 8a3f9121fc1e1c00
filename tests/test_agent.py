"""Agent components: batcher, modelconfig watcher, payload logger."""

import asyncio
import json
import os

import httpx
import pytest
from fastapi.testclient import TestClient

from kserve_amd.agent.batcher import Batcher, create_batcher_proxy_app
from kserve_amd.agent.payload_logger import LogEntry, LogMode, PayloadLogger
from kserve_amd.agent.watcher import ModelConfigWatcher, ModelSpec


def run(coro):
    return asyncio.new_event_loop().run_until_complete(coro)


class TestBatcher:
    def test_flush_on_size(self):
        async def main():
            calls = []

            async def predict(instances):
                calls.append(list(instances))
                return {"predictions": [i * 2 for i in instances]}

            b = Batcher(predict, max_batch_size=4, max_latency_ms=60000)
            results = await asyncio.gather(
                b.predict([1, 2]), b.predict([3, 4])
            )
            assert len(calls) == 1  # one flush covered both
            assert calls[0] == [1, 2, 3, 4]
            assert results[0]["predictions"] == [2, 4]
            assert results[1]["predictions"] == [6, 8]
            assert results[0]["batchId"] == results[1]["batchId"]
            assert results[0]["instanceCount"] == 2

        run(main())

    def test_flush_on_latency(self):
        async def main():
            async def predict(instances):
                return {"predictions": [0 for _ in instances]}

            b = Batcher(predict, max_batch_size=100, max_latency_ms=50)
            t0 = asyncio.get_event_loop().time()
            out = await b.predict([1])
            dt = asyncio.get_event_loop().time() - t0
            assert out["predictions"] == [0]
            assert 0.03 < dt < 2.0

        run(main())

    def test_oversized_passthrough(self):
        async def main():
            calls = []

            async def predict(instances):
                calls.append(len(instances))
                return {"predictions": [0] * len(instances)}

            b = Batcher(predict, max_batch_size=2, max_latency_ms=60000)
            out = await b.predict([1, 2, 3])
            assert out["predictions"] == [0, 0, 0]
            assert calls == [3]

        run(main())

    def test_prediction_count_mismatch(self):
        async def main():
            async def predict(instances):
                return {"predictions": [0, 0, 0]}  # wrong count

            b = Batcher(predict, max_batch_size=4, max_latency_ms=50)
            with pytest.raises(RuntimeError):
                await b.predict([1, 2])

        run(main())

    def test_proxy_app(self):
        def backend(request: httpx.Request) -> httpx.Response:
            body = json.loads(request.content)
            return httpx.Response(
                200, json={"predictions": [x + 1 for x in body["instances"]]}
            )

        app = create_batcher_proxy_app(
            "http://backend",
            "m",
            max_batch_size=2,
            max_latency_ms=30,
            transport=httpx.MockTransport(backend),
        )
        with TestClient(app) as c:
            r = c.post("/v1/models/m:predict", json={"instances": [5]})
            assert r.status_code == 200
            assert r.json()["predictions"] == [6]


class TestWatcher:
    def test_load_and_unload(self, tmp_path):
        async def main():
            cfg_dir = tmp_path / "configs"
            cfg_dir.mkdir()
            model_dir = tmp_path / "models"
            model_dir.mkdir()
            src = tmp_path / "src_model"
            src.mkdir()
            (src / "model.joblib").write_bytes(b"fake")

            loaded, unloaded = [], []

            async def on_load(name, local, spec):
                loaded.append((name, local))

            async def on_unload(name):
                unloaded.append(name)

            w = ModelConfigWatcher(
                str(cfg_dir), str(model_dir), on_load, on_unload
            )
            cfg = [
                {
                    "modelName": "m1",
                    "modelSpec": {"storageUri": str(src), "framework": "sklearn"},
                }
            ]
            (cfg_dir / "modelconfig.json").write_text(json.dumps(cfg))
            assert await w.sync_once()
            assert loaded and loaded[0][0] == "m1"
            assert os.path.exists(model_dir / "m1" / "model.joblib")

            # remove the model -> unload
            (cfg_dir / "modelconfig.json").write_text("[]")
            os.utime(cfg_dir / "modelconfig.json", (1e9, 2e9))
            assert await w.sync_once()
            assert unloaded == ["m1"]
            assert not os.path.exists(model_dir / "m1")

        run(main())

    def test_changed_uri_reloads(self, tmp_path):
        async def main():
            cfg_dir = tmp_path / "c"
            cfg_dir.mkdir()
            mdir = tmp_path / "m"
            mdir.mkdir()
            s1 = tmp_path / "s1"
            s1.mkdir()
            (s1 / "f").write_bytes(b"1")
            s2 = tmp_path / "s2"
            s2.mkdir()
            (s2 / "f").write_bytes(b"2")
            events = []

            async def on_load(name, local, spec):
                events.append(("load", spec.storage_uri))

            async def on_unload(name):
                events.append(("unload", name))

            w = ModelConfigWatcher(str(cfg_dir), str(mdir), on_load, on_unload)
            p = cfg_dir / "modelconfig.json"
            p.write_text(json.dumps([{"modelName": "x", "modelSpec": {"storageUri": str(s1)}}]))
            await w.sync_once()
            p.write_text(json.dumps([{"modelName": "x", "modelSpec": {"storageUri": str(s2)}}]))
            os.utime(p, (1e9, 2e9))
            await w.sync_once()
            assert events == [
                ("load", str(s1)),
                ("unload", "x"),
                ("load", str(s2)),
            ]

        run(main())


class TestPayloadLogger:
    def test_cloudevent_http(self):
        async def main():
            received = []

            def sink(request: httpx.Request) -> httpx.Response:
                received.append((dict(request.headers), request.content))
                return httpx.Response(200)

            pl = PayloadLogger(
                url="http://sink/",
                transport=httpx.MockTransport(sink),
            )
            await pl.start()
            await pl.log(
                LogEntry(
                    request_id="r1",
                    event_type="org.kubeflow.serving.inference.request",
                    model_name="m",
                    payload=b'{"instances": [[1]]}',
                )
            )
            await pl.stop()
            assert len(received) == 1
            hdrs, body = received[0]
            assert hdrs["ce-specversion"] == "1.0"
            assert hdrs["ce-type"].endswith("request")
            assert hdrs["ce-requestid"] == "r1"
            assert body == b'{"instances": [[1]]}'

        run(main())

    def test_file_store_and_csv(self, tmp_path):
        async def main():
            pl = PayloadLogger(store_path=str(tmp_path), marshaller="csv")
            await pl.start()
            await pl.log(
                LogEntry(
                    request_id="r2",
                    event_type="org.kubeflow.serving.inference.response",
                    model_name="m",
                    payload=b'{"predictions": [[1, 2], [3, 4]]}',
                )
            )
            await pl.stop()
            path = tmp_path / "r2-response.csv"
            assert path.exists()
            assert path.read_text().strip().splitlines() == ["1,2", "3,4"]

        run(main())

    def test_mode_filter(self):
        async def main():
            pl = PayloadLogger(store_path="/tmp/unused", mode=LogMode.response)
            assert not pl.should_log("org.kubeflow.serving.inference.request")
            assert pl.should_log("org.kubeflow.serving.inference.response")

        run(main())


class TestAgentDrainer:
    def test_readyz_and_drain(self):
        import httpx
        from fastapi.testclient import TestClient

        from kserve_amd.agent.batcher import create_batcher_proxy_app

        def backend(request: httpx.Request) -> httpx.Response:
            if request.url.path == "/":
                return httpx.Response(200, json={"status": "alive"})
            body = json.loads(request.content)
            return httpx.Response(
                200, json={"predictions": [x for x in body["instances"]]}
            )

        app = create_batcher_proxy_app(
            "http://backend", "m",
            transport=httpx.MockTransport(backend),
        )
        with TestClient(app) as c:
            assert c.get("/readyz").status_code == 200
            r = c.post(
                "/v1/models/m:predict", json={"instances": [[1, 2]]}
            )
            assert r.status_code == 200
            # drain: probe flips to 503, in-flight drained
            d = c.post("/drain")
            assert d.json()["draining"] is True
            assert c.get("/readyz").status_code == 503


class TestLoggerStoreAndBatching:
    """Parquet marshaller + blob-store sinks + batch strategies (reference
    pkg/logger/store.go:64-104, marshaller_parquet.go, batch_*.go)."""

    def _entry(self, rid="r1", payload=b'{"instances": [[1, 2]]}'):
        from kserve_amd.agent.payload_logger import LogEntry

        return LogEntry(
            request_id=rid,
            event_type="org.kubeflow.serving.inference.request",
            model_name="m",
            payload=payload,
        )

    def test_parquet_marshaller_roundtrip(self):
        import io

        import pyarrow.parquet as pq

        from kserve_amd.agent.payload_logger import Marshaller

        m = Marshaller("parquet")
        data = m.marshal_batch([self._entry("a"), self._entry("b")])
        table = pq.read_table(io.BytesIO(data))
        assert table.num_rows == 2
        assert table.column("request_id").to_pylist() == ["a", "b"]
        assert "instances" in table.column("payload").to_pylist()[0]

    def test_file_blob_store(self, tmp_path):
        from kserve_amd.agent.payload_logger import BlobStore

        store = BlobStore(f"file://{tmp_path}/logs")
        store.put("x.json", b"{}")
        assert (tmp_path / "logs" / "x.json").read_bytes() == b"{}"

    def test_s3_blob_store_uploads(self):
        from kserve_amd.agent.payload_logger import BlobStore

        uploads = {}

        class FakeS3:
            def put_object(self, bucket, key, body):
                uploads[(bucket, key)] = body

        store = BlobStore("s3://logs-bucket/payloads", s3_client=FakeS3())
        store.put("a.parquet", b"PAR1")
        assert uploads == {("logs-bucket", "payloads/a.parquet"): b"PAR1"}

    def test_size_batch_flushes_at_threshold(self, tmp_path):
        import asyncio

        from kserve_amd.agent.payload_logger import (
            BatchStrategy,
            BlobStore,
            PayloadLogger,
        )

        logger = PayloadLogger(
            marshaller="parquet",
            store=BlobStore(f"file://{tmp_path}/batched"),
            batch=BatchStrategy("size", size=3),
        )

        async def run():
            await logger.start()
            for i in range(7):
                await logger.log(self._entry(f"r{i}"))
            await asyncio.sleep(0.2)
            await logger.flush()  # drain the 7th entry
            await logger.stop()

        asyncio.run(run())
        import glob

        files = sorted(glob.glob(str(tmp_path / "batched" / "*.parquet")))
        assert len(files) == 3  # 3 + 3 + 1(flush)
        import pyarrow.parquet as pq

        total = sum(pq.read_table(f).num_rows for f in files)
        assert total == 7


class TestAgentChainLogging:
    def test_proxy_logs_request_and_response_cloudevents(self, tmp_path):
        """Agent chain parity (cmd/agent/main.go:429-449): the proxy
        captures request AND response payloads through the PayloadLogger
        when one is attached."""
        from kserve_amd.agent.payload_logger import PayloadLogger

        def backend(request: httpx.Request) -> httpx.Response:
            body = json.loads(request.content)
            return httpx.Response(
                200, json={"predictions": [x * 2 for x in body["instances"]]}
            )

        pl = PayloadLogger(store_path=str(tmp_path), num_workers=1)
        app = create_batcher_proxy_app(
            "http://backend", "m", max_batch_size=4, max_latency_ms=10,
            transport=httpx.MockTransport(backend), payload_logger=pl,
        )
        with TestClient(app) as c:
            r = c.post(
                "/v1/models/m:predict",
                json={"instances": [3]},
                headers={"x-request-id": "req-7"},
            )
            assert r.status_code == 200
            # drain the async logger inside the app's loop
            c.portal.call(pl.stop)
        files = sorted(p.name for p in tmp_path.iterdir())
        assert any("request" in f for f in files), files
        assert any("response" in f for f in files), files
        # json marshaller stores the raw payload; file name carries the id
        req_file = [p for p in tmp_path.iterdir() if "request" in p.name][0]
        assert req_file.name.startswith("req-7-")
        logged = json.loads(req_file.read_text())
        assert logged["instances"] == [3]
        resp_file = [p for p in tmp_path.iterdir()
                     if "response" in p.name][0]
        assert json.loads(resp_file.read_text())["predictions"] == [6]
