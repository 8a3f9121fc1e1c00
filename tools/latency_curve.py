#!/usr/bin/env python3
"""Latency/throughput tradeoff curve for the flagship model.

One engine, several offered concurrencies: for each point, submit C
requests (512-token prompts), run the prefill wave, then decode to
completion; report output tok/s, TTFT p50/p99, and TPOT (time per output
token at steady state). This is the curve the headline bench's single
full-load point cannot show (BASELINE metric is "output tok/s + p50
TTFT": the operating point is a choice, so publish the whole frontier).

Run (GPU box): python tools/latency_curve.py --model llama-3-8b
"""

import argparse
import json
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


def run_point(engine, SamplingParams, concurrency, prompt_len, max_tokens,
              vocab, rng):
    t_submit = time.perf_counter()
    sp = SamplingParams(temperature=0.0, max_tokens=max_tokens,
                        ignore_eos=True)
    for i in range(concurrency):
        prompt = [rng.randrange(vocab) for _ in range(prompt_len)]
        engine.add_request(prompt, sp, request_id=f"c{concurrency}-{i}")
    ttfts = {}
    done = 0
    decode_t0 = None
    decode_tokens = 0
    while engine.scheduler.has_unfinished():
        outs = engine.step()
        now = time.perf_counter()
        for o in outs:
            if o.finished:
                done += 1
        if decode_t0 is None:
            for r in list(engine.scheduler.running):
                if r.first_token_time is not None and r.request_id not in ttfts:
                    ttfts[r.request_id] = r.first_token_time - t_submit
            # decode starts once the whole wave has prefilled
            if len(ttfts) >= concurrency and engine.scheduler.num_waiting == 0:
                decode_t0 = now
        else:
            decode_tokens += sum(len(o.new_token_ids) for o in outs)
    ttfts = list(ttfts.values())
    total = time.perf_counter() - t_submit
    decode_time = time.perf_counter() - decode_t0 if decode_t0 else total
    out_tokens = concurrency * max_tokens
    return {
        "concurrency": concurrency,
        "output_tok_s": round(out_tokens / total, 1),
        "decode_tok_s": round(decode_tokens / decode_time, 1)
        if decode_tokens else None,
        "ttft_p50_ms": round(
            statistics.median(ttfts) * 1000, 1) if ttfts else None,
        "ttft_p99_ms": round(
            sorted(ttfts)[max(0, int(len(ttfts) * 0.99) - 1)] * 1000, 1
        ) if ttfts else None,
        "tpot_ms": round(decode_time / max_tokens * 1000, 2),
        "total_s": round(total, 2),
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-3-8b",
                    choices=["llama-3-8b", "llama-3-70b", "tiny"])
    ap.add_argument("--points", default="1,8,32,128,512,1536")
    ap.add_argument("--prompt-len", type=int, default=512)
    ap.add_argument("--max-tokens", type=int, default=128)
    ap.add_argument("--kv-cache-dtype", default="auto",
                    choices=["auto", "fp8"])
    args = ap.parse_args()

    import random

    import torch

    from kserve_amd.engine.config import (
        CacheConfig,
        EngineConfig,
        ModelConfig,
        SchedulerConfig,
    )
    from kserve_amd.engine.engine import LLMEngine
    from kserve_amd.engine.sampling_params import SamplingParams

    use_gpu = torch.cuda.is_available()
    points = [int(p) for p in args.points.split(",")]
    if args.model == "llama-3-8b":
        mcfg = ModelConfig.llama3_8b()
    elif args.model == "llama-3-70b":
        mcfg = ModelConfig.llama3_70b()
    else:
        mcfg = ModelConfig.tiny(vocab_size=1024)
    cfg = EngineConfig(
        model=mcfg,
        cache=CacheConfig(block_size=16,
                          kv_cache_dtype=args.kv_cache_dtype,
                          num_gpu_blocks=None if use_gpu else 4096),
        scheduler=SchedulerConfig(
            max_num_seqs=max(points),
            max_num_batched_tokens=16384,
            max_model_len=args.prompt_len + args.max_tokens + 64,
        ),
        device="cuda" if use_gpu else "cpu",
        seed=0,
        enforce_eager=not use_gpu,
        eos_token_id=-1,
    )
    engine = LLMEngine(cfg)
    rng = random.Random(0)
    # warmup
    run_point(engine, SamplingParams, min(points), args.prompt_len, 8,
              mcfg.vocab_size, rng)
    rows = []
    for c in points:
        rows.append(
            run_point(engine, SamplingParams, c, args.prompt_len,
                      args.max_tokens, mcfg.vocab_size, rng)
        )
        print(json.dumps({"model": mcfg.model_name, **rows[-1]}), flush=True)


if __name__ == "__main__":
    main()
