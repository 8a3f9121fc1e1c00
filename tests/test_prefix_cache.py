"""Automatic prefix caching: content-addressed KV blocks shared across
requests; cache-hit prompts compute only their suffix via the
paged-context prefill path. Greedy outputs must be IDENTICAL with caching
on or off."""

import pytest
import torch

from kserve_amd.engine.block_manager import BlockManager
from kserve_amd.engine.config import (
    CacheConfig,
    EngineConfig,
    ModelConfig,
    SchedulerConfig,
)
from kserve_amd.engine.engine import LLMEngine
from kserve_amd.engine.request import Request
from kserve_amd.engine.sampling_params import SamplingParams


def mk_req(rid, tokens):
    return Request(rid, tokens, SamplingParams(max_tokens=4), eos_token_id=-1)


class TestBlockManagerPrefixCache:
    def test_reuse_after_publish(self):
        bm = BlockManager(32, 4, enable_prefix_caching=True)
        prompt = list(range(12))  # 3 full blocks; cap -> 2 matchable
        r1 = mk_req("a", prompt)
        bm.allocate(r1)
        assert r1.num_cached_tokens == 0
        r1.num_computed_tokens = 12
        bm.register_computed_blocks(r1)
        # same prompt again: the two capped full blocks hit
        r2 = mk_req("b", prompt)
        bm.allocate(r2)
        assert r2.num_cached_tokens == 8
        assert r2.block_table[:2] == r1.block_table[:2]  # shared
        assert r2.block_table[2] != r1.block_table[2]
        bm.free(r1)
        bm.free(r2)

    def test_shared_blocks_survive_owner_free(self):
        bm = BlockManager(32, 4, enable_prefix_caching=True)
        prompt = list(range(9))  # 2 full blocks, both matchable
        r1 = mk_req("a", prompt)
        bm.allocate(r1)
        r1.num_computed_tokens = 9
        bm.register_computed_blocks(r1)
        bm.free(r1)  # cached blocks become evictable, not freed
        r2 = mk_req("b", prompt)
        bm.allocate(r2)
        assert r2.num_cached_tokens == 8

    def test_eviction_under_pressure(self):
        bm = BlockManager(8, 4, enable_prefix_caching=True)  # 7 usable
        r1 = mk_req("a", list(range(9)))
        bm.allocate(r1)
        r1.num_computed_tokens = 9
        bm.register_computed_blocks(r1)
        bm.free(r1)
        # allocate everything: cached blocks must be evicted to satisfy it
        r2 = mk_req("b", list(range(100, 128)))  # 7 blocks
        bm.allocate(r2)
        assert len(r2.block_table) == 7
        # cache is gone now
        r3 = mk_req("c", list(range(9)))
        bm.free(r2)
        bm.allocate(r3)
        assert r3.num_cached_tokens == 0

    def test_divergent_suffix_shares_only_prefix(self):
        bm = BlockManager(32, 4, enable_prefix_caching=True)
        a = mk_req("a", [1, 2, 3, 4, 5, 6, 7, 8, 9, 10])
        bm.allocate(a)
        a.num_computed_tokens = 10
        bm.register_computed_blocks(a)
        b = mk_req("b", [1, 2, 3, 4, 99, 98, 97, 96, 95, 94])
        bm.allocate(b)
        assert b.num_cached_tokens == 4  # only the first block matches
        assert b.block_table[0] == a.block_table[0]
        assert b.block_table[1] != a.block_table[1]


def make_engine(prefix: bool, blocks=128):
    cfg = EngineConfig(
        model=ModelConfig.tiny(vocab_size=128),
        cache=CacheConfig(
            block_size=4, num_gpu_blocks=blocks, enable_prefix_caching=prefix
        ),
        scheduler=SchedulerConfig(
            max_num_seqs=8, max_num_batched_tokens=256, max_model_len=128
        ),
        device="cpu",
        eos_token_id=-1,
    )
    return LLMEngine(cfg)


def test_engine_outputs_identical_with_prefix_cache():
    torch.manual_seed(0)
    plain = make_engine(False)
    torch.manual_seed(0)
    cached = make_engine(True)
    sp = SamplingParams(temperature=0.0, max_tokens=8)
    shared = list(range(1, 21))  # 20-token shared system prompt
    prompts = [shared + [30 + i] for i in range(4)] + [shared]
    a = [o.output_token_ids for o in plain.generate(prompts, sp).values()]
    # run twice so the second wave hits the cache populated by the first
    b0 = [o.output_token_ids for o in cached.generate(prompts, sp).values()]
    b1 = [o.output_token_ids for o in cached.generate(prompts, sp).values()]
    assert a == b0 == b1
    bm = cached.scheduler.block_manager
    assert bm.cache_hit_tokens > 0, "second wave must hit the prefix cache"


def test_engine_prefix_cache_with_preemption():
    torch.manual_seed(0)
    plain = make_engine(False)
    sp = SamplingParams(temperature=0.0, max_tokens=8)
    shared = list(range(1, 17))
    prompts = [shared + [40 + i] for i in range(4)]
    a = [o.output_token_ids for o in plain.generate(prompts, sp).values()]
    torch.manual_seed(0)
    tiny = make_engine(True, blocks=28)  # pressure: eviction + preemption
    b = [o.output_token_ids for o in tiny.generate(prompts, sp).values()]
    assert a == b


def test_prefix_cache_with_chunked_prefill():
    """Both features on: cached prefix + chunked suffix compute."""
    torch.manual_seed(0)
    plain = make_engine(False)
    sp = SamplingParams(temperature=0.0, max_tokens=6)
    shared = list(range(1, 41))  # 40 tokens
    prompts = [shared + [60 + i] for i in range(3)]
    a = [o.output_token_ids for o in plain.generate(prompts, sp).values()]
    torch.manual_seed(0)
    cfg = EngineConfig(
        model=ModelConfig.tiny(vocab_size=128),
        cache=CacheConfig(
            block_size=4, num_gpu_blocks=128, enable_prefix_caching=True
        ),
        scheduler=SchedulerConfig(
            max_num_seqs=8,
            max_num_batched_tokens=16,  # forces chunking of the 41-token prompts
            max_model_len=128,
            enable_chunked_prefill=True,
        ),
        device="cpu",
        eos_token_id=-1,
    )
    both = LLMEngine(cfg)
    b0 = [o.output_token_ids for o in both.generate(prompts, sp).values()]
    b1 = [o.output_token_ids for o in both.generate(prompts, sp).values()]
    assert a == b0 == b1
    assert both.scheduler.block_manager.cache_hit_tokens > 0


def test_prefix_cache_isolates_lora_adapters(tmp_path):
    """LoRA changes K/V projections: identical prompts under different
    adapters must not share cached blocks (the hash chain is seeded with
    the adapter id)."""
    from tests.test_lora import make_adapter_dir

    torch.manual_seed(0)
    cached = make_engine(True)
    path, _ = make_adapter_dir(tmp_path, cached.config.model)
    cached.register_lora("adapt", path)
    torch.manual_seed(0)
    plain = make_engine(False)
    plain.register_lora("adapt", path)

    shared = list(range(1, 21))
    sp_base = SamplingParams(temperature=0.0, max_tokens=6)
    sp_lora = SamplingParams(temperature=0.0, max_tokens=6, lora_name="adapt")

    # populate the cache with BASE-model KV for the prompt
    a_base = list(cached.generate([shared], sp_base).values())[0]
    # the LoRA request must NOT hit those blocks
    a_lora = list(cached.generate([shared], sp_lora).values())[0]
    ref_lora = list(plain.generate([shared], sp_lora).values())[0]
    assert a_lora.output_token_ids == ref_lora.output_token_ids
    assert a_lora.output_token_ids != a_base.output_token_ids
    # same-adapter reuse still works
    b_lora = list(cached.generate([shared], sp_lora).values())[0]
    assert b_lora.output_token_ids == ref_lora.output_token_ids
