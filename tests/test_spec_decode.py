"""Speculative decoding (prompt-lookup/n-gram): exact greedy equivalence.

The verify forward corrects every rejected draft, so a spec-decode engine
must produce IDENTICAL greedy outputs to a plain engine — on any input.
(The reference's vLLM backend ships the same technique as
speculative_config ngram; ours reuses the paged-context prefill kernel.)
"""

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


def make_engine(spec: int = 0, **kw):
    cfg = EngineConfig(
        model=ModelConfig.tiny(vocab_size=128),
        cache=CacheConfig(block_size=4, num_gpu_blocks=256),
        scheduler=SchedulerConfig(
            max_num_seqs=8,
            max_num_batched_tokens=256,
            max_model_len=256,
            speculative_ngram=spec,
        ),
        device="cpu",
        eos_token_id=-1,
        **kw,
    )
    return LLMEngine(cfg)


PROMPTS = [
    [1, 2, 3, 4, 1, 2, 3, 4, 1, 2, 3],  # repetitive: drafts accept
    [9, 8, 7, 6, 5],                    # no repeats: drafts mostly reject
    [50, 51, 50, 51, 50, 51, 50],
    [17],
]


def test_spec_matches_plain_greedy():
    torch.manual_seed(0)
    plain = make_engine(spec=0)
    torch.manual_seed(0)
    spec = make_engine(spec=4)
    sp = SamplingParams(temperature=0.0, max_tokens=24)
    a = [o.output_token_ids for o in plain.generate(PROMPTS, sp).values()]
    b = [o.output_token_ids for o in spec.generate(PROMPTS, sp).values()]
    assert a == b


def test_spec_accepts_on_repetitive_input():
    """A model in a greedy loop must be accelerated: fewer engine steps than
    tokens generated."""
    torch.manual_seed(0)
    engine = make_engine(spec=4)
    sp = SamplingParams(temperature=0.0, max_tokens=40, ignore_eos=True)
    rid = engine.add_request(PROMPTS[0], sp)
    steps = 0
    toks = 0
    while engine.has_unfinished():
        outs = engine.step()
        steps += 1
        for o in outs:
            toks += len(o.new_token_ids)
        assert steps < 200
    assert toks == 40
    # random-weight tiny models loop quickly under greedy decoding, so the
    # n-gram drafter must land multi-token steps
    assert steps < 40, f"no acceleration: {steps} steps for {toks} tokens"


def test_spec_respects_stop_and_max_tokens():
    torch.manual_seed(0)
    plain = make_engine(spec=0)
    torch.manual_seed(0)
    spec = make_engine(spec=4)
    # run plain first to learn which token appears, use it as a stop token
    sp0 = SamplingParams(temperature=0.0, max_tokens=30)
    base = list(plain.generate([PROMPTS[0]], sp0).values())[0].output_token_ids
    stop_tok = base[len(base) // 2]
    sp_stop = SamplingParams(
        temperature=0.0, max_tokens=30, stop_token_ids=[stop_tok]
    )
    torch.manual_seed(0)
    plain2 = make_engine(spec=0)
    a = list(plain2.generate([PROMPTS[0]], sp_stop).values())[0]
    b = list(spec.generate([PROMPTS[0]], sp_stop).values())[0]
    assert a.output_token_ids == b.output_token_ids
    assert b.finish_reason == "stop"


def test_spec_with_preemption():
    """Spec decode under a tiny KV pool (preemption active) stays exact."""
    torch.manual_seed(0)
    plain = make_engine(spec=0)
    sp = SamplingParams(temperature=0.0, max_tokens=12)
    a = [o.output_token_ids for o in plain.generate(PROMPTS, sp).values()]
    torch.manual_seed(0)
    cfg = EngineConfig(
        model=ModelConfig.tiny(vocab_size=128),
        cache=CacheConfig(block_size=4, num_gpu_blocks=24),
        scheduler=SchedulerConfig(
            max_num_seqs=4,
            max_num_batched_tokens=256,
            max_model_len=256,
            speculative_ngram=4,
        ),
        device="cpu",
        eos_token_id=-1,
    )
    tiny = LLMEngine(cfg)
    b = [o.output_token_ids for o in tiny.generate(PROMPTS, sp).values()]
    assert a == b


def test_all_features_combined_matches_plain():
    """Spec decode + prefix caching + chunked prefill + tiny KV pool
    (preemption) all on together must still match plain greedy outputs."""
    torch.manual_seed(0)
    plain = make_engine(spec=0)
    sp = SamplingParams(temperature=0.0, max_tokens=16)
    shared = [5, 6, 7, 8] * 4  # 16-token shared, repetitive (spec accepts)
    prompts = [shared + [40 + i] for i in range(4)] + [PROMPTS[0]]
    a = [o.output_token_ids for o in plain.generate(prompts, sp).values()]
    torch.manual_seed(0)
    cfg = EngineConfig(
        model=ModelConfig.tiny(vocab_size=128),
        cache=CacheConfig(
            block_size=4, num_gpu_blocks=96, enable_prefix_caching=True
        ),
        scheduler=SchedulerConfig(
            max_num_seqs=8,
            max_num_batched_tokens=12,  # forces chunking
            max_model_len=256,
            speculative_ngram=4,
            enable_chunked_prefill=True,
        ),
        device="cpu",
        eos_token_id=-1,
    )
    full = LLMEngine(cfg)
    b0 = [o.output_token_ids for o in full.generate(prompts, sp).values()]
    b1 = [o.output_token_ids for o in full.generate(prompts, sp).values()]
    assert a == b0 == b1


class TestDraftModelSpeculation:
    """Draft-MODEL speculation (round 2): exact greedy outputs regardless
    of draft quality; a perfect draft (same weights) accepts everything."""

    def _engine(self, draft_seed_offset=1, k=3):
        import dataclasses

        from kserve_amd.engine.config import (
            CacheConfig,
            EngineConfig,
            ModelConfig,
            SchedulerConfig,
        )
        from kserve_amd.engine.engine import LLMEngine

        mcfg = ModelConfig.tiny(vocab_size=256)
        cfg = EngineConfig(
            model=mcfg,
            cache=CacheConfig(block_size=4, num_gpu_blocks=512),
            scheduler=SchedulerConfig(
                max_num_seqs=4, max_num_batched_tokens=256,
                max_model_len=128, multi_step=1, speculative_k=k,
            ),
            device="cpu",
            seed=11,
            eos_token_id=-1,
            draft_model=dataclasses.replace(mcfg) if k > 0 else None,
        )
        eng = LLMEngine(cfg)
        if eng.draft is not None and draft_seed_offset == 0:
            # perfect draft: copy the MAIN model's weights
            eng.draft.model.load_state_dict(
                eng.runner.model.state_dict()
            )
        return eng

    def _plain(self):
        from kserve_amd.engine.config import (
            CacheConfig,
            EngineConfig,
            ModelConfig,
            SchedulerConfig,
        )
        from kserve_amd.engine.engine import LLMEngine

        cfg = EngineConfig(
            model=ModelConfig.tiny(vocab_size=256),
            cache=CacheConfig(block_size=4, num_gpu_blocks=512),
            scheduler=SchedulerConfig(
                max_num_seqs=4, max_num_batched_tokens=256,
                max_model_len=128, multi_step=1,
            ),
            device="cpu",
            seed=11,
            eos_token_id=-1,
        )
        return LLMEngine(cfg)

    def _generate(self, eng, prompts, n=16):
        from kserve_amd.engine.sampling_params import SamplingParams

        sp = SamplingParams(temperature=0.0, max_tokens=n, ignore_eos=True)
        out = eng.generate(prompts, sp)
        steps = getattr(eng, "_dbg_steps", None)
        return [o.output_token_ids for o in out.values()]

    def test_exact_greedy_with_imperfect_draft(self):
        prompts = [[1, 2, 3, 4], [9, 8, 7]]
        ref = self._generate(self._plain(), prompts)
        got = self._generate(self._engine(draft_seed_offset=1), prompts)
        assert got == ref

    def test_perfect_draft_accepts_and_matches(self):
        import time

        prompts = [[5, 6, 7, 8]]
        ref = self._generate(self._plain(), prompts)
        eng = self._engine(draft_seed_offset=0)
        # count engine steps: a perfect draft emits k+1 tokens per step
        steps = 0
        from kserve_amd.engine.sampling_params import SamplingParams

        sp = SamplingParams(temperature=0.0, max_tokens=16, ignore_eos=True)
        eng.add_request(prompts[0], sp, request_id="p")
        toks = []
        while eng.scheduler.has_unfinished() and steps < 40:
            for o in eng.step():
                toks.extend(o.new_token_ids)
            steps += 1
        assert toks == ref[0]
        # 1 prefill + ceil(15/4)=4 spec rounds (k=3 -> up to 4/step)
        assert steps <= 6, f"perfect draft should compress steps, got {steps}"

    def test_draft_state_released_on_finish(self):
        eng = self._engine()
        prompts = [[1, 2, 3]]
        self._generate(eng, prompts, n=8)
        assert eng.draft._shadows == {}
        bm = eng.draft.block_manager
        # all blocks back on the free list (block 0 stays reserved)
        assert len(bm._free) == bm.num_blocks - 1
        assert bm._tables == {}
