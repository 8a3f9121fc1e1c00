#!/usr/bin/env python3
"""Speculative decoding benefit on a repetitive workload (llama-3-8b,
batch 8, greedy): tok/s and accepted-draft share, spec off vs on."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from kserve_amd.engine.config import (
    CacheConfig,
    EngineConfig,
    ModelConfig,
    SchedulerConfig,
)
from kserve_amd.engine.engine import LLMEngine
from kserve_amd.engine.sampling_params import SamplingParams

N, OUT = 8, 256


def run(spec: int):
    torch.manual_seed(0)
    cfg = EngineConfig(
        model=ModelConfig.llama3_8b(),
        cache=CacheConfig(block_size=16, num_gpu_blocks=4096),
        scheduler=SchedulerConfig(
            max_num_seqs=N, max_num_batched_tokens=16384, max_model_len=2048,
            speculative_ngram=spec,
        ),
        device="cuda",
        seed=0,
        eos_token_id=-1,
    )
    engine = LLMEngine(cfg)
    g = torch.Generator().manual_seed(3)
    # highly repetitive prompts (code/doc style): 8-token motif repeated
    motif = torch.randint(10, 100000, (8,), generator=g).tolist()
    prompts = [
        (motif * 32) + [200 + i] for i in range(N)  # 257 tokens
    ]
    sp = SamplingParams(temperature=0.0, max_tokens=OUT, ignore_eos=True)
    for rid, p in enumerate(prompts):
        engine.add_request(p, sp, request_id=f"r{rid}")
    torch.cuda.synchronize()
    t0 = time.monotonic()
    toks = 0
    steps = 0
    while engine.has_unfinished():
        outs = engine.step()
        steps += 1
        toks += sum(len(o.new_token_ids) for o in outs)
    torch.cuda.synchronize()
    dt = time.monotonic() - t0
    del engine
    torch.cuda.empty_cache()
    return toks / dt, steps


base_tps, base_steps = run(0)
spec_tps, spec_steps = run(4)
print(f"spec off: {base_tps:7.0f} tok/s ({base_steps} steps)   "
      f"spec on (ngram k=4): {spec_tps:7.0f} tok/s ({spec_steps} steps)   "
      f"speedup {spec_tps/base_tps:.2f}x")
