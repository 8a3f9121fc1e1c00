"""OpenAI endpoints end-to-end over the async engine (tiny CPU model)."""

import json

import pytest
import torch
from fastapi.testclient import TestClient

from kserve_amd.engine.config import (
    CacheConfig,
    EngineConfig,
    ModelConfig,
    SchedulerConfig,
)
from kserve_amd.model_repository import ModelRepository
from kserve_amd.protocol.dataplane import DataPlane
from kserve_amd.protocol.rest.openai.endpoints import register_openai_endpoints
from kserve_amd.protocol.rest.server import create_app
from kserve_amd.runtimes.llm_model import LLMModel


@pytest.fixture(scope="module")
def client():
    torch.manual_seed(0)
    cfg = EngineConfig(
        model=ModelConfig.tiny(vocab_size=128),
        cache=CacheConfig(block_size=4, num_gpu_blocks=128),
        scheduler=SchedulerConfig(
            max_num_seqs=8, max_num_batched_tokens=256, max_model_len=128
        ),
        device="cpu",
        eos_token_id=-1,
    )
    model = LLMModel("tiny", cfg)
    repo = ModelRepository()
    repo.update(model)
    dataplane = DataPlane(repo)
    app = create_app(dataplane)
    register_openai_endpoints(app, dataplane, [model])
    app.state.llm_model = model

    with TestClient(app) as c:
        # start engine manually (TestClient doesn't run ModelServer lifecycle)
        import asyncio

        loop = asyncio.new_event_loop()
        loop.run_until_complete(model.start_engine())
        yield c
        model.stop()


def test_models_list(client):
    r = client.get("/openai/v1/models")
    assert r.status_code == 200
    assert r.json()["data"][0]["id"] == "tiny"


def test_completion_token_ids(client):
    r = client.post(
        "/openai/v1/completions",
        json={
            "model": "tiny",
            "prompt": [1, 2, 3, 4],
            "max_tokens": 5,
            "temperature": 0.0,
        },
    )
    assert r.status_code == 200, r.text
    body = r.json()
    assert body["object"] == "text_completion"
    assert body["choices"][0]["finish_reason"] == "length"
    assert body["usage"]["prompt_tokens"] == 4
    assert body["usage"]["completion_tokens"] == 5
    # no tokenizer -> text is space-separated token ids
    assert len(body["choices"][0]["text"].split()) == 5


def test_completion_stream(client):
    with client.stream(
        "POST",
        "/openai/v1/completions",
        json={
            "model": "tiny",
            "prompt": [5, 6, 7],
            "max_tokens": 4,
            "temperature": 0.0,
            "stream": True,
        },
    ) as r:
        assert r.status_code == 200
        events = []
        for line in r.iter_lines():
            if line.startswith("data: "):
                events.append(line[len("data: ") :])
    assert events[-1] == "[DONE]"
    chunks = [json.loads(e) for e in events[:-1]]
    # multi-step windows may deliver several tokens per SSE chunk
    total_tokens = sum(
        len(c["choices"][0]["text"].split()) for c in chunks
    )
    assert total_tokens == 4
    assert chunks[-1]["choices"][0]["finish_reason"] == "length"


def test_chat_completion(client):
    r = client.post(
        "/openai/v1/chat/completions",
        json={
            "model": "tiny",
            "messages": [{"role": "user", "content": "hi"}],
            "max_tokens": 3,
            "temperature": 0.0,
        },
    )
    # no tokenizer -> chat template fallback requires tokenizer -> 400
    assert r.status_code == 400


def test_unknown_model(client):
    r = client.post(
        "/openai/v1/completions",
        json={"model": "nope", "prompt": [1], "max_tokens": 1},
    )
    assert r.status_code == 404


def test_bad_request(client):
    r = client.post(
        "/openai/v1/completions",
        json={"model": "tiny", "max_tokens": 1},
    )
    assert r.status_code == 400


def test_v1_alias(client):
    r = client.post(
        "/v1/completions",
        json={"model": "tiny", "prompt": [1, 2], "max_tokens": 2, "temperature": 0.0},
    )
    assert r.status_code == 200


def test_completion_logprobs_and_echo(client):
    r = client.post(
        "/openai/v1/completions",
        json={
            "model": "tiny",
            "prompt": [1, 2, 3],
            "max_tokens": 3,
            "temperature": 0.0,
            "logprobs": 2,
            "echo": True,
        },
    )
    assert r.status_code == 200, r.text
    choice = r.json()["choices"][0]
    lp = choice["logprobs"]
    assert lp is not None
    assert len(lp["token_logprobs"]) == 3
    assert all(v is not None and v <= 0 for v in lp["token_logprobs"])
    assert len(lp["top_logprobs"][0]) >= 2
    # echo: prompt token ids prefixed (no tokenizer -> id text)
    assert choice["text"].startswith("1 2 3")


def test_completion_n_choices(client):
    r = client.post(
        "/openai/v1/completions",
        json={
            "model": "tiny",
            "prompt": [4, 5],
            "max_tokens": 2,
            "n": 3,
            "temperature": 0.8,
            "seed": 1,
        },
    )
    assert r.status_code == 200, r.text
    assert len(r.json()["choices"]) == 3


def test_stream_disconnect_aborts(client):
    """Closing the SSE stream mid-generation aborts the engine request
    (AsyncLLMEngine submits an abort sentinel on generator close)."""
    import time

    from kserve_amd.runtimes.llm_model import LLMModel

    model = client.app.state.llm_model
    assert isinstance(model, LLMModel)
    engine = model.async_engine.engine

    with client.stream(
        "POST",
        "/openai/v1/completions",
        json={
            "model": "tiny",
            "prompt": [9, 9, 9],
            "max_tokens": 100,
            "temperature": 0.0,
            "stream": True,
        },
    ) as r:
        for i, line in enumerate(r.iter_lines()):
            if i >= 2:
                break  # client disconnects early

    deadline = time.time() + 10
    while time.time() < deadline and engine.has_unfinished():
        time.sleep(0.1)
    assert not engine.has_unfinished(), "aborted request still scheduled"


class TestOpenAIProxyModel:
    """Proxy model forwards completions to an upstream OpenAI server
    (reference openai_proxy_model.py), with hooks and SSE passthrough."""

    def _upstream_app(self):
        """In-process upstream: the tiny LLM engine behind OpenAI routes."""
        from kserve_amd.model_repository import ModelRepository
        from kserve_amd.protocol.dataplane import DataPlane
        from kserve_amd.protocol.rest.server import create_app
        from kserve_amd.runtimes.llm_model import LLMModel

        torch.manual_seed(0)
        cfg = EngineConfig(
            model=ModelConfig.tiny(vocab_size=128),
            cache=CacheConfig(block_size=4, num_gpu_blocks=128),
            scheduler=SchedulerConfig(
                max_num_seqs=4, max_num_batched_tokens=256, max_model_len=128
            ),
            device="cpu",
            eos_token_id=-1,
        )
        model = LLMModel("up", cfg)
        repo = ModelRepository()
        repo.update(model)
        dp = DataPlane(repo)
        app = create_app(dp)
        register_openai_endpoints(app, dp, [model])
        return app, model

    def test_proxy_completion_and_stream(self):
        import asyncio

        import httpx

        from kserve_amd.protocol.rest.openai.proxy_model import OpenAIProxyModel

        up_app, up_model = self._upstream_app()

        class CountingProxy(OpenAIProxyModel):
            pre = post = 0

            async def preprocess_completion_request(self, body):
                CountingProxy.pre += 1
                body["model"] = "up"  # route to the upstream model id
                return body

            async def postprocess_completion(self, body):
                CountingProxy.post += 1
                body["proxied"] = True
                return body

        client = httpx.AsyncClient(
            transport=httpx.ASGITransport(app=up_app), base_url="http://up"
        )
        proxy = CountingProxy("front", "http://up", http_client=client)
        from kserve_amd.model_repository import ModelRepository
        from kserve_amd.protocol.dataplane import DataPlane
        from kserve_amd.protocol.rest.server import create_app

        repo = ModelRepository()
        repo.update(proxy)
        dp = DataPlane(repo)
        front = create_app(dp)
        register_openai_endpoints(front, dp, [proxy])

        with TestClient(front) as c:
            asyncio.new_event_loop().run_until_complete(up_model.start_engine())
            r = c.post(
                "/openai/v1/completions",
                json={"model": "front", "prompt": [1, 2, 3], "max_tokens": 4,
                      "temperature": 0.0},
            )
            assert r.status_code == 200, r.text
            out = r.json()
            assert out["proxied"] is True
            assert out["usage"]["completion_tokens"] == 4
            assert CountingProxy.pre == 1 and CountingProxy.post == 1

            # streaming passthrough
            r = c.post(
                "/openai/v1/completions",
                json={"model": "front", "prompt": [1, 2, 3], "max_tokens": 4,
                      "temperature": 0.0, "stream": True},
            )
            assert r.status_code == 200
            assert r.headers["content-type"].startswith("text/event-stream")
            chunks = [
                json.loads(line[len("data: "):])
                for line in r.text.splitlines()
                if line.startswith("data: ") and "[DONE]" not in line
            ]
            total = sum(
                len(ch["choices"][0]["text"] or "") for ch in chunks if ch.get("choices")
            )
            assert total > 0
            up_model.stop()


def test_stream_disconnect_aborts_request(client):
    """Closing a stream mid-generation must abort the engine request
    (reference vLLM with_cancellation semantics)."""
    import time

    model = client.app.state.llm_model
    eng = model.async_engine

    with client.stream(
        "POST",
        "/openai/v1/completions",
        json={
            "model": "tiny",
            "prompt": [1, 2, 3],
            "max_tokens": 64,
            "temperature": 0.0,
            "stream": True,
        },
    ) as r:
        # read a couple of SSE lines then drop the connection
        it = r.iter_lines()
        got = 0
        for line in it:
            if line.startswith("data: "):
                got += 1
            if got >= 2:
                break
    # the abort sentinel is processed by the engine thread between steps
    for _ in range(100):
        if eng.engine is not None and not eng.engine.has_unfinished():
            break
        time.sleep(0.05)
    assert not eng.engine.has_unfinished(), "request not aborted after disconnect"


def test_stream_include_usage(client):
    r = client.post(
        "/openai/v1/completions",
        json={
            "model": "tiny",
            "prompt": [1, 2, 3],
            "max_tokens": 4,
            "temperature": 0.0,
            "stream": True,
            "stream_options": {"include_usage": True},
        },
    )
    assert r.status_code == 200
    chunks = [
        json.loads(line[len("data: "):])
        for line in r.text.splitlines()
        if line.startswith("data: ") and "[DONE]" not in line
    ]
    usage_chunks = [c for c in chunks if c.get("usage")]
    assert len(usage_chunks) == 1
    u = usage_chunks[-1]["usage"]
    assert u["completion_tokens"] == 4 and u["prompt_tokens"] == 3
    assert chunks[-1].get("usage")  # usage chunk is last


def test_overlong_prompt_is_400_not_hang(client):
    r = client.post(
        "/openai/v1/completions",
        json={"model": "tiny", "prompt": list(range(100)) * 3,
              "max_tokens": 2, "temperature": 0.0},
        )
    assert r.status_code == 400
    assert "max_model_len" in r.text


def test_external_abort_terminates_stream(client):
    """AsyncLLMEngine.abort() must end a consumer's stream with an abort
    output instead of hanging it."""
    import asyncio

    model = client.app.state.llm_model
    eng = model.async_engine

    async def run():
        from kserve_amd.engine.sampling_params import SamplingParams

        gen = eng.generate(
            [1, 2, 3], SamplingParams(temperature=0.0, max_tokens=4096),
            request_id="abort-me",
        )
        first = await asyncio.wait_for(gen.__anext__(), timeout=30)
        assert not first.finished
        await eng.abort("abort-me")
        # the stream must terminate promptly
        while True:
            out = await asyncio.wait_for(gen.__anext__(), timeout=30)
            if out.finished:
                assert out.finish_reason == "abort"
                break

    asyncio.new_event_loop().run_until_complete(run())


def test_response_format_json_object(client):
    r = client.post(
        "/openai/v1/completions",
        json={
            "model": "tiny",
            "prompt": [1, 2, 3],
            "max_tokens": 40,
            "temperature": 0.8,
            "seed": 3,
            "response_format": {"type": "json_object"},
        },
    )
    assert r.status_code == 200, r.text
    # tokenizer-less fallback streams token ids as text; re-check the raw
    # ids through the engine's own notion of validity instead
    from kserve_amd.engine.guided import JsonMachine

    model = client.app.state.llm_model
    # round-trip through a direct engine call for byte-level verification
    import asyncio

    from kserve_amd.engine.sampling_params import SamplingParams

    async def run():
        out = await model.async_engine.generate_full(
            [1, 2, 3],
            SamplingParams(
                temperature=0.8, seed=3, max_tokens=40,
                response_format="json_object",
            ),
        )
        return out

    out = asyncio.new_event_loop().run_until_complete(run())
    assert JsonMachine().accepts(bytes(out.output_token_ids))


class TestWithCancellation:
    """Client-disconnect cancellation (reference: vLLM's with_cancellation
    on every OpenAI route). The watcher listens on the ASGI receive
    channel; http.disconnect cancels the in-flight handler."""

    def _request(self, messages):
        import asyncio

        from fastapi import Request

        q = asyncio.Queue()
        for m in messages:
            q.put_nowait(m)

        async def receive():
            return await q.get()

        return Request(
            {"type": "http", "method": "POST", "headers": [],
             "path": "/", "query_string": b""},
            receive=receive,
        )

    def test_handler_completes_normally(self):
        import asyncio

        from kserve_amd.protocol.rest.openai.endpoints import (
            with_cancellation,
        )

        async def run():
            req = self._request([])  # no disconnect ever arrives

            async def handler():
                await asyncio.sleep(0.01)
                return "done"

            return await with_cancellation(handler(), req)

        assert asyncio.new_event_loop().run_until_complete(run()) == "done"

    def test_disconnect_cancels_handler(self):
        import asyncio

        from kserve_amd.protocol.rest.openai.endpoints import (
            with_cancellation,
        )

        cancelled = {}

        async def run():
            req = self._request([{"type": "http.disconnect"}])

            async def slow_handler():
                try:
                    await asyncio.sleep(30)
                except asyncio.CancelledError:
                    cancelled["yes"] = True
                    raise
                return "never"

            return await with_cancellation(slow_handler(), req)

        resp = asyncio.new_event_loop().run_until_complete(run())
        assert cancelled.get("yes") is True
        assert resp.status_code == 499

    def test_engine_abort_on_cancel(self, client):
        """Cancelling the handler mid-generation aborts the engine
        request: the scheduler drains back to empty."""
        import asyncio

        model = client.app.state.llm_model

        async def run():
            from kserve_amd.engine.sampling_params import SamplingParams

            gen = model.async_engine.generate(
                [1, 2, 3], SamplingParams(max_tokens=512, ignore_eos=True),
                request_id="cancel-me",
            )
            task = asyncio.ensure_future(gen.__anext__())
            await task  # first token arrives
            closer = asyncio.ensure_future(gen.aclose())
            await closer
            # abort sentinel drains through the engine thread
            for _ in range(100):
                eng = model.async_engine.engine
                if (eng.scheduler.num_waiting == 0
                        and not eng.scheduler.running):
                    return True
                await asyncio.sleep(0.05)
            return False

        assert asyncio.new_event_loop().run_until_complete(run())
