#!/usr/bin/env python3
"""Prefix-cache TTFT benefit: 512-token prompts sharing a 448-token prefix
(system-prompt pattern), prefill wall time with caching off vs on."""

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

N, PREFIX, TAIL = 64, 448, 64


def run(prefix_caching: bool) -> float:
    torch.manual_seed(0)
    cfg = EngineConfig(
        model=ModelConfig.llama3_8b(),
        cache=CacheConfig(
            block_size=16, num_gpu_blocks=8192,
            enable_prefix_caching=prefix_caching,
        ),
        scheduler=SchedulerConfig(
            max_num_seqs=N, max_num_batched_tokens=16384, max_model_len=1024
        ),
        device="cuda",
        seed=0,
        eos_token_id=-1,
    )
    engine = LLMEngine(cfg)
    g = torch.Generator().manual_seed(7)
    shared = torch.randint(10, 100000, (PREFIX,), generator=g).tolist()
    prompts = [
        shared + torch.randint(10, 100000, (TAIL,), generator=g).tolist()
        for _ in range(N)
    ]
    sp = SamplingParams(temperature=0.0, max_tokens=2)
    # wave 1 populates the cache (or not); wave 2 is the measurement
    engine.generate(prompts, sp)
    torch.cuda.synchronize()
    t0 = time.monotonic()
    engine.generate([p + [1] for p in prompts], sp)  # new requests, same prefix
    torch.cuda.synchronize()
    dt = time.monotonic() - t0
    hit = engine.scheduler.block_manager.cache_hit_tokens
    del engine
    torch.cuda.empty_cache()
    return dt, hit


off, _ = run(False)
on, hits = run(True)
print(f"prefill+2tok wall, {N} reqs x {PREFIX + TAIL + 1} toks: "
      f"off={off*1000:.0f} ms  on={on*1000:.0f} ms  "
      f"speedup={off/on:.2f}x  cache_hit_tokens={hits}")
