"""Host-DRAM KV offload tier: swap-based preemption must preserve outputs
without recompute."""

import pytest
import torch

from kserve_amd.engine.config import (
    CacheConfig,
    EngineConfig,
    ModelConfig,
    SchedulerConfig,
)
from kserve_amd.engine.engine import LLMEngine
from kserve_amd.engine.sampling_params import SamplingParams


def make_engine(num_blocks, cpu_blocks):
    torch.manual_seed(0)
    cfg = EngineConfig(
        model=ModelConfig.tiny(vocab_size=128),
        cache=CacheConfig(
            block_size=4, num_gpu_blocks=num_blocks, num_cpu_blocks=cpu_blocks
        ),
        scheduler=SchedulerConfig(
            max_num_seqs=4, max_num_batched_tokens=256, max_model_len=128
        ),
        device="cpu",
        eos_token_id=-1,
    )
    return LLMEngine(cfg)


def test_swap_preemption_preserves_outputs():
    prompts = [[1, 2, 3, 4], [9, 8, 7, 6], [15, 16, 17, 18]]
    sp = SamplingParams(temperature=0.0, max_tokens=24)

    big = make_engine(num_blocks=64, cpu_blocks=0)
    ref = [o.output_token_ids for o in big.generate(prompts, sp).values()]

    small = make_engine(num_blocks=14, cpu_blocks=32)
    swapped_seen = {"n": 0}
    orig_schedule = small.scheduler.schedule

    def counting_schedule():
        b = orig_schedule()
        swapped_seen["n"] += len(b.swap_out)
        return b

    small.scheduler.schedule = counting_schedule
    got = [o.output_token_ids for o in small.generate(prompts, sp).values()]
    assert got == ref
    assert swapped_seen["n"] > 0, "expected the swap path to be exercised"


def test_swap_roundtrip_block_manager():
    from kserve_amd.engine.block_manager import BlockManager
    from kserve_amd.engine.request import Request

    bm = BlockManager(num_blocks=8, block_size=4, num_cpu_blocks=8)
    r = Request("a", list(range(10)), SamplingParams(max_tokens=4))
    bm.allocate(r)
    gpu_table = list(r.block_table)
    assert bm.can_swap_out(r)
    pairs = bm.swap_out(r)
    assert [g for g, _ in pairs] == gpu_table
    assert r.block_table == []
    assert bm.num_free_blocks == 7  # all back (block 0 reserved)
    assert bm.can_swap_in(r)
    pairs_in = bm.swap_in(r)
    assert len(pairs_in) == len(gpu_table)
    assert len(r.block_table) == len(gpu_table)
