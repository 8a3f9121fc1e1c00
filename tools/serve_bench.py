#!/usr/bin/env python3
"""End-to-end serving load test: /v1/completions over HTTP against the full
stack (FastAPI + AsyncLLMEngine + native engine).

Measures whole-serving-path output tok/s + latency percentiles, vs the
engine-direct bench — the platform-overhead check (sidecar/
router overhead target ~1-3 ms/request).

Run (GPU box): python tools/serve_bench.py --concurrency 64 --max-tokens 64
"""

import argparse
import asyncio
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


async def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model", default="llama-3-8b", choices=["llama-3-8b", "tiny"])
    parser.add_argument("--concurrency", type=int, default=64)
    parser.add_argument("--requests", type=int, default=128)
    parser.add_argument("--prompt-len", type=int, default=512)
    parser.add_argument("--max-tokens", type=int, default=64)
    parser.add_argument("--stream", action="store_true")
    parser.add_argument("--profile", action="store_true",
                        help="cProfile the server event loop + engine thread")
    args = parser.parse_args()

    import torch
    import httpx

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

    use_gpu = torch.cuda.is_available()
    if args.model == "llama-3-8b":
        mcfg = ModelConfig.llama3_8b()
    else:
        mcfg = ModelConfig(
            vocab_size=1024, hidden_size=512, intermediate_size=1024,
            num_layers=2, num_heads=4, num_kv_heads=2, head_dim=128,
            max_position_embeddings=2048, model_name="tiny",
        )
    cfg = EngineConfig(
        model=mcfg,
        cache=CacheConfig(block_size=16),
        scheduler=SchedulerConfig(
            max_num_seqs=args.concurrency,
            max_num_batched_tokens=16384,
            max_model_len=args.prompt_len + args.max_tokens + 64,
        ),
        device="cuda" if use_gpu else "cpu",
        eos_token_id=-1,
        enforce_eager=not use_gpu,
    )
    if not use_gpu:
        cfg.cache.num_gpu_blocks = 512

    if args.profile:
        os.environ["KS_ENGINE_PROFILE"] = "1"  # before the engine thread starts
    model = LLMModel("bench", cfg)
    repo = ModelRepository()
    repo.update(model)
    dataplane = DataPlane(repo)
    app = create_app(dataplane)
    register_openai_endpoints(app, dataplane, [model])
    await model.start_engine()

    transport = httpx.ASGITransport(app=app)
    client = httpx.AsyncClient(
        transport=transport, base_url="http://srv", timeout=600
    )

    import random

    rng = random.Random(0)
    sem = asyncio.Semaphore(args.concurrency)
    latencies = []
    tokens_out = [0]

    async def one_request(i):
        prompt = [rng.randrange(mcfg.vocab_size) for _ in range(args.prompt_len)]
        async with sem:
            t0 = time.perf_counter()
            r = await client.post(
                "/v1/completions",
                json={
                    "model": "bench",
                    "prompt": prompt,
                    "max_tokens": args.max_tokens,
                    "temperature": 0.0,
                    "ignore_eos": True,
                },
            )
            dt = time.perf_counter() - t0
        assert r.status_code == 200, r.text[:300]
        body = r.json()
        tokens_out[0] += body["usage"]["completion_tokens"]
        latencies.append(dt)

    prof = None
    if args.profile:
        import cProfile

        prof = cProfile.Profile()
        prof.enable()
    t0 = time.perf_counter()
    await asyncio.gather(*[one_request(i) for i in range(args.requests)])
    elapsed = time.perf_counter() - t0
    if prof is not None:
        import pstats

        prof.disable()
        stats = pstats.Stats(prof, stream=sys.stderr)
        print("==== server event loop profile ====", file=sys.stderr)
        stats.sort_stats("cumulative").print_stats(30)
    lat_sorted = sorted(latencies)
    eng = model.async_engine.engine
    stats_extra = {}
    try:
        bm = eng.scheduler.block_manager
        stats_extra["kv_usage"] = round(bm.usage, 3)
    except Exception:
        pass
    print(
        {
            "serving_output_tok_s": round(tokens_out[0] / elapsed, 1),
            "prefill_tokens": args.requests * args.prompt_len,
            **stats_extra,
            "requests": args.requests,
            "concurrency": args.concurrency,
            "elapsed_s": round(elapsed, 2),
            "latency_p50_s": round(statistics.median(lat_sorted), 3),
            "latency_p99_s": round(lat_sorted[int(len(lat_sorted) * 0.99) - 1], 3),
            "max_tokens": args.max_tokens,
            "prompt_len": args.prompt_len,
        }
    )
    model.stop()


if __name__ == "__main__":
    asyncio.run(main())
