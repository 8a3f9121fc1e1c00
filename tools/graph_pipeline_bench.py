#!/usr/bin/env python3
"""BASELINE config 5 measured: custom_tokenizer transformer → Llama-3-8B
predictor chained by the InferenceGraph router, end to end.

All three services run in-process as ASGI apps (the router's cross-pod
hops collapse to ASGI dispatch, so this measures the platform path —
router + transformer + OpenAI predictor on the native engine — on top of
the engine's own throughput). Reports pipeline requests/s + latency
percentiles and router overhead vs calling the predictor directly.

Run (GPU box): python tools/graph_pipeline_bench.py
"""

import argparse
import asyncio
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

_TUNE = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                     "profiles", "tunableop_gfx950.csv")
if os.path.exists(_TUNE.replace(".csv", "0.csv")):
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", _TUNE)


async def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-3-8b",
                    choices=["llama-3-8b", "tiny"])
    ap.add_argument("--requests", type=int, default=256)
    ap.add_argument("--concurrency", type=int, default=64)
    ap.add_argument("--max-tokens", type=int, default=32)
    args = ap.parse_args()

    import httpx
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
    from kserve_amd.protocol.rest.openai.endpoints import (
        register_openai_endpoints,
    )
    from kserve_amd.protocol.rest.server import create_app
    from kserve_amd.runtimes.llm_model import LLMModel
    from kserve_amd.runtimes.tokenizer_transformer import TokenizerTransformer

    use_gpu = torch.cuda.is_available()
    mcfg = (
        ModelConfig.llama3_8b() if args.model == "llama-3-8b"
        else ModelConfig.tiny(vocab_size=1024)
    )
    cfg = EngineConfig(
        model=mcfg,
        cache=CacheConfig(block_size=16,
                          num_gpu_blocks=None if use_gpu else 1024),
        scheduler=SchedulerConfig(
            max_num_seqs=args.concurrency,
            max_num_batched_tokens=16384,
            max_model_len=512,
        ),
        device="cuda" if use_gpu else "cpu",
        enforce_eager=not use_gpu,
        eos_token_id=-1,
    )

    tok_repo = ModelRepository()
    tok_model = TokenizerTransformer(
        "tokenizer", predictor_model=mcfg.model_name,
        max_tokens=args.max_tokens,
    )
    tok_repo.update(tok_model)
    tok_app = create_app(DataPlane(tok_repo))

    llm = LLMModel(mcfg.model_name, cfg)
    llm_repo = ModelRepository()
    llm_repo.update(llm)
    llm_dp = DataPlane(llm_repo)
    llm_app = create_app(llm_dp)
    register_openai_endpoints(llm_app, llm_dp, [llm])
    await llm.start_engine()

    class HostDispatchTransport(httpx.AsyncBaseTransport):
        def __init__(self, apps):
            self._t = {h: httpx.ASGITransport(app=a) for h, a in apps.items()}

        async def handle_async_request(self, request):
            return await self._t[request.url.host].handle_async_request(request)

    transport = HostDispatchTransport({"tok": tok_app, "llm": llm_app})
    spec = InferenceGraphSpec.from_dict(
        {
            "nodes": {
                "root": {
                    "routerType": "Sequence",
                    "steps": [
                        {
                            "serviceUrl": "http://tok/v1/models/tokenizer:predict",
                            "data": "$request",
                        },
                        {
                            "serviceUrl": "http://llm/openai/v1/completions",
                            "data": "$response",
                        },
                    ],
                }
            }
        }
    )
    router = GraphRouter(spec, transport=transport)

    sem = asyncio.Semaphore(args.concurrency)
    latencies = []
    tokens = [0]

    async def one(i):
        body = {
            "instances": [
                f"benchmark request {i} with a short prompt to tokenize"
            ]
        }
        async with sem:
            t0 = time.perf_counter()
            code, out = await router.handle(body, {})
            latencies.append(time.perf_counter() - t0)
        assert code == 200, out
        tokens[0] += out["usage"]["completion_tokens"]

    # warmup
    await asyncio.gather(*[one(i) for i in range(min(8, args.requests))])
    latencies.clear()
    tokens[0] = 0
    t0 = time.perf_counter()
    await asyncio.gather(*[one(i) for i in range(args.requests)])
    wall = time.perf_counter() - t0
    lat = sorted(latencies)
    print(
        {
            "metric": "graph pipeline (tokenizer->llm) requests/s",
            "model": mcfg.model_name,
            "requests": args.requests,
            "concurrency": args.concurrency,
            "requests_s": round(args.requests / wall, 1),
            "output_tok_s": round(tokens[0] / wall, 1),
            "latency_p50_ms": round(
                statistics.median(lat) * 1000, 1
            ),
            "latency_p99_ms": round(
                lat[max(0, int(len(lat) * 0.99) - 1)] * 1000, 1
            ),
            "max_tokens": args.max_tokens,
        }
    )
    llm.stop()


if __name__ == "__main__":
    asyncio.run(main())
