"""BASELINE config 5 end-to-end on CPU: tokenizer transformer -> LLM
predictor chained by the InferenceGraph router (Sequence node), all three
services running in-process as ASGI apps."""

import asyncio
import json

import httpx
import pytest
import torch

from kserve_amd.engine.config import (
    CacheConfig,
    EngineConfig,
    ModelConfig,
    SchedulerConfig,
)
from kserve_amd.graph.router import GraphRouter
from kserve_amd.graph.types import InferenceGraphSpec
from kserve_amd.model_repository import ModelRepository
from kserve_amd.protocol.dataplane import DataPlane
from kserve_amd.protocol.rest.openai.endpoints import register_openai_endpoints
from kserve_amd.protocol.rest.server import create_app
from kserve_amd.runtimes.llm_model import LLMModel
from kserve_amd.runtimes.tokenizer_transformer import TokenizerTransformer


class HostDispatchTransport(httpx.AsyncBaseTransport):
    """Route requests to per-host ASGI apps (multi-service in-process)."""

    def __init__(self, apps):
        self._transports = {
            host: httpx.ASGITransport(app=app) for host, app in apps.items()
        }

    async def handle_async_request(self, request):
        t = self._transports.get(request.url.host)
        if t is None:
            return httpx.Response(502, json={"error": f"no host {request.url.host}"})
        return await t.handle_async_request(request)


@pytest.mark.timeout(120)
def test_tokenizer_to_llm_pipeline():
    async def main():
        torch.manual_seed(0)
        # --- service 1: tokenizer transformer ---
        tok_repo = ModelRepository()
        tok_model = TokenizerTransformer(
            "tokenizer", predictor_model="tiny", max_tokens=5
        )
        tok_repo.update(tok_model)
        tok_app = create_app(DataPlane(tok_repo))

        # --- service 2: LLM predictor (native engine, tiny model) ---
        cfg = EngineConfig(
            model=ModelConfig.tiny(vocab_size=512),
            cache=CacheConfig(block_size=4, num_gpu_blocks=128),
            scheduler=SchedulerConfig(
                max_num_seqs=4, max_num_batched_tokens=256, max_model_len=256
            ),
            device="cpu",
            eos_token_id=-1,
        )
        llm = LLMModel("tiny", cfg)
        llm_repo = ModelRepository()
        llm_repo.update(llm)
        llm_dp = DataPlane(llm_repo)
        llm_app = create_app(llm_dp)
        register_openai_endpoints(llm_app, llm_dp, [llm])
        await llm.start_engine()

        # --- the router chaining them (Sequence; BASELINE config 5) ---
        spec = InferenceGraphSpec.from_dict(
            {
                "nodes": {
                    "root": {
                        "routerType": "Sequence",
                        "steps": [
                            {
                                "name": "tokenize",
                                "serviceUrl": "http://tokenizer/v1/models/tokenizer:predict",
                            },
                            {
                                "name": "generate",
                                "serviceUrl": "http://llm/v1/completions",
                            },
                        ],
                    }
                }
            }
        )
        transport = HostDispatchTransport(
            {"tokenizer": tok_app, "llm": llm_app}
        )
        router = GraphRouter(spec, transport=transport)
        code, out = await router.handle(
            {"instances": ["hello pipeline"]}, {"x-request-id": "e2e-1"}
        )
        assert code == 200, out
        assert out["object"] == "text_completion"
        assert out["usage"]["completion_tokens"] == 5
        # prompt was the byte-tokenized text
        assert out["usage"]["prompt_tokens"] == len("hello pipeline")
        await router.close()
        llm.stop()

    asyncio.new_event_loop().run_until_complete(main())
