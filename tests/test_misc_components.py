"""Timeseries protocol, multiprocess REST, LocalModelCache, OpenAI
embeddings adapter."""

import asyncio
import json
import os

import httpx
import pytest
from fastapi.testclient import TestClient

from kserve_amd.model import Model
from kserve_amd.model_repository import ModelRepository
from kserve_amd.protocol.dataplane import DataPlane
from kserve_amd.protocol.rest.server import create_app


def run(coro):
    return asyncio.new_event_loop().run_until_complete(coro)


class ForecastModel(Model):
    def __init__(self):
        super().__init__("ts")
        self.ready = True

    async def forecast(self, inputs, horizon, quantiles=None, parameters=None):
        out = []
        for series in inputs:
            vals = series["values"]
            mean = sum(vals) / len(vals)
            out.append({"forecast": [mean] * horizon})
        return out


class TestTimeSeries:
    def test_forecast(self):
        repo = ModelRepository()
        repo.update(ForecastModel())
        client = TestClient(create_app(DataPlane(repo)))
        r = client.post(
            "/timeseries/v1/forecast",
            json={
                "model": "ts",
                "inputs": [{"timestamps": [1, 2], "values": [2.0, 4.0]}],
                "horizon": 3,
            },
        )
        assert r.status_code == 200
        assert r.json()["outputs"][0]["forecast"] == [3.0, 3.0, 3.0]

    def test_unsupported_model(self):
        repo = ModelRepository()

        class Plain(Model):
            def __init__(self):
                super().__init__("p")
                self.ready = True

        repo.update(Plain())
        client = TestClient(create_app(DataPlane(repo)))
        r = client.post(
            "/timeseries/v1/forecast",
            json={"model": "p", "inputs": [{"values": [1]}], "horizon": 1},
        )
        assert r.status_code == 400


class TestMultiprocessREST:
    def test_workers_share_socket(self):
        import requests

        from kserve_amd.protocol.rest.multiprocess import RESTServerMultiProcess

        def app_factory():
            repo = ModelRepository()

            class Echo(Model):
                def __init__(self):
                    super().__init__("e")
                    self.ready = True

                def predict(self, payload, headers=None):
                    return {"predictions": payload["instances"], "pid": os.getpid()}

            repo.update(Echo())
            return create_app(DataPlane(repo))

        srv = RESTServerMultiProcess(app_factory, http_port=0, workers=2)
        srv.start()
        try:
            import time

            deadline = time.time() + 15
            ok = False
            while time.time() < deadline:
                try:
                    r = requests.post(
                        f"http://127.0.0.1:{srv.port}/v1/models/e:predict",
                        json={"instances": [1, 2]},
                        timeout=2,
                    )
                    if r.status_code == 200:
                        ok = True
                        break
                except Exception:
                    time.sleep(0.3)
            assert ok
            assert r.json()["predictions"] == [1, 2]
        finally:
            srv.stop()


class TestLocalModelCache:
    def test_cache_and_evict(self, tmp_path):
        from kserve_amd.controlplane.localmodel import (
            LocalModelCacheSpec,
            LocalModelNodeAgent,
            mount_for_isvc,
        )

        src = tmp_path / "src"
        src.mkdir()
        (src / "weights.bin").write_bytes(b"w" * 128)
        agent = LocalModelNodeAgent(str(tmp_path / "cache"), node_group="gpu")
        spec = LocalModelCacheSpec(
            name="llama", source_model_uri=str(src), node_groups=["gpu"]
        )
        agent.apply(spec)
        status = run(agent.reconcile_once())
        assert status["llama"].state == "Ready"
        assert agent.is_cached("llama")
        assert mount_for_isvc(agent, str(src)) == agent.model_path("llama")
        # other node group: not applied
        agent2 = LocalModelNodeAgent(str(tmp_path / "cache2"), node_group="cpu")
        agent2.apply(spec)
        assert "llama" not in agent2.desired
        # eviction
        agent.delete("llama")
        run(agent.reconcile_once())
        assert not agent.is_cached("llama")


class TestOpenAIEmbeddingsAdapter:
    def test_adapter_shape(self):
        from kserve_amd.protocol.rest.openai.types import EmbeddingRequest
        from kserve_amd.runtimes.encoder_model import OpenAIEmbeddingAdapter

        class FakeEncoder:
            name = "emb"

            def preprocess(self, payload, headers=None):
                return {"texts": payload["instances"], "_v2": None}

            def predict(self, payload, headers=None):
                return {
                    "output": None,
                    "ids_list": [[1, 2, 3]] * len(payload["texts"]),
                    "cu": None,
                    "_v2": None,
                }

            def postprocess(self, result, headers=None):
                return {"predictions": [[0.1, 0.2]] * len(result["ids_list"])}

        adapter = OpenAIEmbeddingAdapter(FakeEncoder())
        req = EmbeddingRequest(model="emb", input=["a", "b"])
        out = run(adapter.create_embedding(req))
        assert len(out.data) == 2
        assert out.data[0].embedding == [0.1, 0.2]
        assert out.usage.prompt_tokens == 6


class TestExplainer:
    def test_occlusion_explainer_end_to_end(self):
        import httpx

        from kserve_amd.model import PredictorConfig
        from kserve_amd.runtimes.explainer import ExplainerModel

        # fake predictor: prediction = 3*x0 + 1*x1 (feature 0 dominates)
        def predictor(request: httpx.Request) -> httpx.Response:
            body = json.loads(request.content)
            preds = [3 * r[0] + r[1] for r in body["instances"]]
            return httpx.Response(200, json={"predictions": preds})

        model = ExplainerModel(
            "exp", PredictorConfig(predictor_host="pred:80")
        )
        model._http_client = httpx.AsyncClient(
            transport=httpx.MockTransport(predictor)
        )
        out = run(model.explain({"instances": [[2.0, 5.0]]}))
        imps = out["explanations"]["importances"][0]
        assert imps[0] > imps[1]  # x0 matters more
        assert imps[0] == pytest.approx(6.0)
        assert imps[1] == pytest.approx(5.0)


    def test_square_attack_explainer_flips_label(self):
        import httpx

        from kserve_amd.model import PredictorConfig
        from kserve_amd.runtimes.explainer import (
            ExplainerModel,
            SquareAttackExplainer,
        )

        # black-box scorer: class-1 score rises with sum(pixels); decision
        # boundary at 8 (soft scores give the square search its signal)
        def predictor(request: httpx.Request) -> httpx.Response:
            body = json.loads(request.content)
            preds = []
            for inst in body["instances"]:
                s = sum(sum(row) for row in inst)
                preds.append([8.0 - s, s - 8.0])
            return httpx.Response(200, json={"predictions": preds})

        model = ExplainerModel(
            "art",
            PredictorConfig(predictor_host="pred:80"),
            explainer=SquareAttackExplainer(nb_classes=2, max_iter=60, eps=1.0),
        )
        model._http_client = httpx.AsyncClient(
            transport=httpx.MockTransport(predictor)
        )
        image = [[1.2, 1.2, 1.2], [1.2, 1.2, 1.2], [1.2, 1.2, 1.2]]  # sum 10.8
        out = run(model.explain({"instances": [image, 1]}))
        exp = out["explanations"]
        assert exp["prediction"] == 1
        assert exp["adversarial_prediction"] == 0  # attack flipped the label
        assert exp["L2 error"] > 0

    def test_aif_fairness_metrics(self):
        from kserve_amd.runtimes.aifserver import AIFFairnessModel

        model = AIFFairnessModel(
            "aif",
            feature_names=["age", "income"],
            label_names=["approved"],
            favorable_label=1.0,
            unfavorable_label=0.0,
            privileged_groups=[{"age": 1.0}],
            unprivileged_groups=[{"age": 0.0}],
        )
        # privileged (age=1): 3/4 favorable; unprivileged (age=0): 1/4
        instances = [
            [1, 10], [1, 20], [1, 30], [1, 40],
            [0, 10], [0, 20], [0, 30], [0, 40],
        ]
        outputs = [1, 1, 1, 0, 1, 0, 0, 0]
        out = run(model.explain({"instances": instances, "outputs": outputs}))
        m = out["metrics"]
        assert m["num_instances"] == 8
        assert m["num_positives"] == 4
        assert m["num_negatives"] == 4
        assert m["base_rate"] == pytest.approx(0.5)
        assert m["statistical_parity_difference"] == pytest.approx(0.25 - 0.75)
        assert m["disparate_impact"] == pytest.approx(0.25 / 0.75)
        assert 0.0 <= m["consistency"][0] <= 1.0

    def test_autogluon_gated(self):
        from kserve_amd.runtimes.autogluonserver import AutoGluonModel

        model = AutoGluonModel("ag", "/nonexistent")
        with pytest.raises(RuntimeError, match="autogluon"):
            model.load()


class TestRerankAdapter:
    def test_cosine_rerank(self):
        from kserve_amd.protocol.rest.openai.types import RerankRequest
        from kserve_amd.runtimes.encoder_model import OpenAIEmbeddingAdapter

        class FakeEncoder:
            name = "emb"

            def preprocess(self, payload, headers=None):
                return {"texts": payload["instances"], "_v2": None}

            def predict(self, payload, headers=None):
                vecs = {
                    "q": [1.0, 0.0],
                    "близко": [0.9, 0.1],
                    "far": [0.0, 1.0],
                }
                return {
                    "output": [vecs.get(t, [0.5, 0.5]) for t in payload["texts"]],
                    "ids_list": [[1]] * len(payload["texts"]),
                    "cu": None,
                    "_v2": None,
                }

            def postprocess(self, result, headers=None):
                return {"predictions": result["output"]}

        adapter = OpenAIEmbeddingAdapter(FakeEncoder())
        req = RerankRequest(model="emb", query="q", documents=["far", "близко"])
        out = run(adapter.create_rerank(req))
        assert out.results[0].index == 1  # близко ranks first
        assert out.results[0].relevance_score > out.results[1].relevance_score


class TestExampleSamples:
    """The examples/ directory mirrors the reference's custom_model /
    custom_transformer samples; keep them importable and serving-correct."""

    def test_custom_model_sample_end_to_end(self):
        import sys

        sys.path.insert(0, "examples")
        try:
            from custom_model import CustomModel
        finally:
            sys.path.pop(0)
        from fastapi.testclient import TestClient

        from kserve_amd.model_repository import ModelRepository
        from kserve_amd.protocol.dataplane import DataPlane
        from kserve_amd.protocol.rest.server import create_app

        model = CustomModel("custom-model")
        repo = ModelRepository()
        repo.update(model)
        app = create_app(DataPlane(repo))
        with TestClient(app) as c:
            r = c.post(
                "/v1/models/custom-model:predict",
                json={"instances": [[0.5] * 784, [0.1] * 784]},
            )
            assert r.status_code == 200
            assert len(r.json()["predictions"]) == 2

    def test_custom_transformer_sample(self):
        import sys

        import httpx

        sys.path.insert(0, "examples")
        try:
            from custom_transformer import ImageTransformer
        finally:
            sys.path.pop(0)

        def predictor(request: httpx.Request) -> httpx.Response:
            body = json.loads(request.content)
            # echo back the max of each (normalized) row
            return httpx.Response(
                200,
                json={"predictions": [max(r) for r in body["instances"]]},
            )

        t = ImageTransformer("t", predictor_host="pred:80")
        t._http_client = httpx.AsyncClient(transport=httpx.MockTransport(predictor))
        pre = t.preprocess({"instances": [[0, 128, 255]]})
        assert pre["instances"][0][2] == 1.0
        out = run(t.predict(pre))
        assert out["predictions"] == [1.0]


class TestStorage:
    def test_file_and_pvc_download(self, tmp_path):
        from kserve_amd.storage.storage import Storage

        src = tmp_path / "model"
        src.mkdir()
        (src / "weights.bin").write_bytes(b"abc")
        out = Storage.download(f"file://{src}", str(tmp_path / "o1"))
        assert (pathlib := __import__("pathlib")).Path(out, "weights.bin").read_bytes() == b"abc"
        # pvc:// maps under /mnt/pvc — simulate via plain local path form
        out2 = Storage.download(str(src), str(tmp_path / "o2"))
        assert pathlib.Path(out2, "weights.bin").exists()

    def test_http_tar_download(self, tmp_path):
        import http.server
        import io
        import socketserver
        import tarfile
        import threading

        from kserve_amd.storage.storage import Storage

        buf = io.BytesIO()
        with tarfile.open(fileobj=buf, mode="w:gz") as tf:
            data = b"hello-model"
            info = tarfile.TarInfo("m/weights.txt")
            info.size = len(data)
            tf.addfile(info, io.BytesIO(data))
        payload = buf.getvalue()

        class H(http.server.BaseHTTPRequestHandler):
            def do_GET(self):
                self.send_response(200)
                self.send_header("Content-Type", "application/gzip")
                self.end_headers()
                self.wfile.write(payload)

            def log_message(self, *a):
                pass

        with socketserver.TCPServer(("127.0.0.1", 0), H) as srv:
            port = srv.server_address[1]
            t = threading.Thread(target=srv.serve_forever, daemon=True)
            t.start()
            try:
                out = Storage.download(
                    f"http://127.0.0.1:{port}/model.tar.gz",
                    str(tmp_path / "o3"),
                )
            finally:
                srv.shutdown()
        import pathlib

        found = list(pathlib.Path(out).rglob("weights.txt"))
        assert found and found[0].read_bytes() == b"hello-model"

    def test_unknown_scheme_raises(self, tmp_path):
        from kserve_amd.storage.storage import Storage

        with pytest.raises(Exception):
            Storage.download("carrier-pigeon://model", str(tmp_path / "o4"))

    def test_git_clone_with_subdir(self, tmp_path):
        import pathlib
        import subprocess

        from kserve_amd.storage.storage import Storage

        repo = tmp_path / "src-repo"
        (repo / "models" / "a").mkdir(parents=True)
        (repo / "models" / "a" / "weights.bin").write_bytes(b"\x01\x02")
        (repo / "README.md").write_text("top")
        env = {
            "GIT_AUTHOR_NAME": "t", "GIT_AUTHOR_EMAIL": "t@t",
            "GIT_COMMITTER_NAME": "t", "GIT_COMMITTER_EMAIL": "t@t",
            "HOME": str(tmp_path), "PATH": os.environ["PATH"],
        }
        for cmd in (
            ["git", "init", "-q", "-b", "main"],
            ["git", "add", "-A"],
            ["git", "commit", "-q", "-m", "init"],
        ):
            subprocess.run(cmd, cwd=repo, check=True, env=env)

        out = Storage.download(
            f"git+file://{repo}#models/a", str(tmp_path / "git-out")
        )
        assert (pathlib.Path(out) / "weights.bin").read_bytes() == b"\x01\x02"
        # whole-repo clone (no subdir) copies everything but .git
        out2 = Storage.download(f"git+file://{repo}", str(tmp_path / "git-out2"))
        assert (pathlib.Path(out2) / "README.md").exists()
        assert not (pathlib.Path(out2) / ".git").exists()


class TestQpextMerging:
    """Prometheus exposition merging (reference qpext sanitizeMetrics
    :113 + scrape :198): one header per family, duplicate series
    disambiguated by source, noisy default collectors dropped."""

    def test_families_merge_with_single_header(self):
        from kserve_amd.agent.qpext import sanitize_metrics

        qp = (
            "# HELP requests_total Total requests\n"
            "# TYPE requests_total counter\n"
            'requests_total{code="200"} 5\n'
        )
        app = (
            "# HELP requests_total Total requests\n"
            "# TYPE requests_total counter\n"
            'requests_total{code="500"} 1\n'
            "# HELP model_load_seconds Load time\n"
            "# TYPE model_load_seconds gauge\n"
            "model_load_seconds 1.5\n"
        )
        merged = sanitize_metrics([("queue-proxy", qp), ("app", app)])
        assert merged.count("# TYPE requests_total counter") == 1
        assert 'requests_total{code="200"} 5' in merged
        assert 'requests_total{code="500"} 1' in merged
        assert "model_load_seconds 1.5" in merged

    def test_duplicate_series_get_source_label(self):
        from kserve_amd.agent.qpext import sanitize_metrics

        a = "# TYPE up gauge\nup 1\n"
        b = "# TYPE up gauge\nup 0\n"
        merged = sanitize_metrics([("queue-proxy", a), ("app", b)])
        assert "up 1" in merged
        assert 'up{source="app"} 0' in merged

    def test_noise_prefixes_dropped_and_histograms_kept(self):
        from kserve_amd.agent.qpext import sanitize_metrics

        text = (
            "# HELP python_gc_count GC\n# TYPE python_gc_count counter\n"
            "python_gc_count 3\n"
            "# HELP lat Latency\n# TYPE lat histogram\n"
            'lat_bucket{le="0.1"} 2\nlat_sum 0.3\nlat_count 2\n'
        )
        merged = sanitize_metrics([("app", text)])
        assert "python_gc_count" not in merged
        assert 'lat_bucket{le="0.1"} 2' in merged
        assert "lat_count 2" in merged

    def test_scrape_config_from_env(self):
        from kserve_amd.agent.qpext import ScrapeConfiguration

        cfg = ScrapeConfiguration.from_env(
            {
                "AGGREGATE_PROMETHEUS_METRICS_PORT": "8080",
                "APP_METRICS_PATH": "/stats",
                "QUEUE_PROXY_METRICS_PORT": "9091",
                "METRICS_SCRAPE_TIMEOUT_S": "2",
            }
        )
        urls = [t.url for t in cfg.targets]
        assert "http://127.0.0.1:9091/metrics" in urls
        assert "http://127.0.0.1:8080/stats" in urls
        assert all(t.timeout_s == 2.0 for t in cfg.targets)

    def test_app_merges_and_survives_dead_target(self):
        import httpx
        from fastapi.testclient import TestClient

        from kserve_amd.agent.qpext import ScrapeConfiguration, ScrapeTarget, create_qpext_app

        async def handler(request):
            if "good" in str(request.url):
                return httpx.Response(
                    200, text="# TYPE ok gauge\nok 1\n"
                )
            raise httpx.ConnectError("down")

        app = create_qpext_app(
            config=ScrapeConfiguration(
                targets=[
                    ScrapeTarget("http://good/metrics", source_label="app"),
                    ScrapeTarget("http://dead/metrics"),
                ]
            ),
            transport=httpx.MockTransport(handler),
        )
        client = TestClient(app)
        r = client.get("/metrics")
        assert r.status_code == 200
        assert "ok 1" in r.text
        assert client.get("/healthz").json()["status"] == "ok"


class TestLatencyLogging:
    def test_flag_gates_trace_lines(self, caplog):
        """--enable_latency_logging parity (reference ModelServer): the
        per-request latency trace line is emitted by default and
        suppressed when disabled."""
        import logging

        from kserve_amd import model as model_mod
        from kserve_amd.logging import TRACE_LOGGER_NAME

        class Echo(Model):
            def predict(self, payload, headers=None):
                return {"predictions": payload["instances"]}

        m = Echo("echo")
        m.ready = True
        loop = asyncio.new_event_loop()
        try:
            model_mod.set_latency_logging(True)
            with caplog.at_level(logging.INFO, logger=TRACE_LOGGER_NAME):
                loop.run_until_complete(m({"instances": [1]}))
            assert any("preprocess_ms" in r.message for r in caplog.records)
            caplog.clear()
            model_mod.set_latency_logging(False)
            with caplog.at_level(logging.INFO, logger=TRACE_LOGGER_NAME):
                loop.run_until_complete(m({"instances": [1]}))
            assert not any("preprocess_ms" in r.message
                           for r in caplog.records)
        finally:
            model_mod.set_latency_logging(True)
            loop.close()
