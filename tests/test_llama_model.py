"""Numerics: our Llama forward vs HuggingFace transformers reference (CPU fp32),
and prefill+paged-decode consistency vs full-context forward."""

import numpy as np
import pytest
import torch

from kserve_amd.engine.config import CacheConfig, EngineConfig, ModelConfig, SchedulerConfig
from kserve_amd.models.llama import AttentionMetadata, LlamaForCausalLM


def tiny_config():
    return ModelConfig.tiny(vocab_size=128)


def build_model(cfg, seed=0):
    torch.manual_seed(seed)
    model = LlamaForCausalLM(cfg, dtype=torch.float32, device="cpu")
    model.random_init(seed=seed)
    return model


def full_forward_logits(model, token_ids):
    """Contiguous full-context prefill, no cache; returns [T, vocab]."""
    T = len(token_ids)
    meta = AttentionMetadata(
        is_prefill=True,
        slot_mapping=torch.zeros(T, dtype=torch.int32),
        cu_seqlens=torch.tensor([0, T], dtype=torch.int32),
        max_seqlen=T,
    )
    # empty caches (numel 0 -> skip reshape_and_cache)
    caches = [
        (torch.empty(0), torch.empty(0)) for _ in range(model.config.num_layers)
    ]
    ids = torch.tensor(token_ids, dtype=torch.int64)
    pos = torch.arange(T, dtype=torch.int64)
    hidden = model(ids, pos, caches, meta)
    return model.compute_logits(hidden)


class TestVsTransformers:
    def test_logits_match_hf(self):
        transformers = pytest.importorskip("transformers")
        cfg = tiny_config()
        hf_cfg = transformers.LlamaConfig(
            vocab_size=cfg.vocab_size,
            hidden_size=cfg.hidden_size,
            intermediate_size=cfg.intermediate_size,
            num_hidden_layers=cfg.num_layers,
            num_attention_heads=cfg.num_heads,
            num_key_value_heads=cfg.num_kv_heads,
            rms_norm_eps=cfg.rms_norm_eps,
            rope_theta=cfg.rope_theta,
            max_position_embeddings=cfg.max_position_embeddings,
            attention_bias=False,
            tie_word_embeddings=False,
        )
        torch.manual_seed(7)
        hf_model = transformers.LlamaForCausalLM(hf_cfg).eval().float()
        ours = LlamaForCausalLM(cfg, dtype=torch.float32, device="cpu")
        ours.load_hf_state_dict(dict(hf_model.state_dict()))

        token_ids = list(torch.randint(0, cfg.vocab_size, (24,)).tolist())
        with torch.no_grad():
            hf_logits = hf_model(
                torch.tensor([token_ids]), use_cache=False
            ).logits[0]
        our_logits = full_forward_logits(ours, token_ids)
        torch.testing.assert_close(our_logits, hf_logits, rtol=2e-4, atol=2e-4)


class TestPagedDecodeConsistency:
    def test_decode_matches_full_forward(self):
        """Greedy generation via prefill + paged decode must equal repeated
        full-context forwards."""
        cfg = tiny_config()
        model = build_model(cfg, seed=3)
        block_size = 4
        num_blocks = 32
        kv_heads = cfg.num_kv_heads
        caches = [
            (
                torch.zeros(num_blocks, kv_heads, block_size, cfg.head_dim),
                torch.zeros(num_blocks, kv_heads, block_size, cfg.head_dim),
            )
            for _ in range(cfg.num_layers)
        ]
        prompt = list(torch.randint(0, cfg.vocab_size, (9,)).tolist())

        # --- engine-style: prefill then paged decode ---
        from kserve_amd.engine.block_manager import BlockManager
        from kserve_amd.engine.request import Request
        from kserve_amd.engine.sampling_params import SamplingParams

        bm = BlockManager(num_blocks, block_size)
        req = Request("r", prompt, SamplingParams(max_tokens=6))
        bm.allocate(req)
        T = len(prompt)
        meta = AttentionMetadata(
            is_prefill=True,
            slot_mapping=torch.tensor(bm.slot_mapping(req, 0, T), dtype=torch.int32),
            cu_seqlens=torch.tensor([0, T], dtype=torch.int32),
            max_seqlen=T,
        )
        ids = torch.tensor(prompt, dtype=torch.int64)
        pos = torch.arange(T, dtype=torch.int64)
        hidden = model(ids, pos, caches, meta)
        logits = model.compute_logits(hidden[-1:])
        engine_tokens = [int(logits.argmax(-1))]
        req.append_output_token(engine_tokens[0])
        req.num_computed_tokens = T

        for _ in range(5):
            bm.append_slot(req)
            p = req.num_computed_tokens
            meta = AttentionMetadata(
                is_prefill=False,
                slot_mapping=torch.tensor(
                    bm.slot_mapping(req, p, p + 1), dtype=torch.int32
                ),
                block_tables=torch.tensor([req.block_table], dtype=torch.int32),
                context_lens=torch.tensor([p + 1], dtype=torch.int32),
            )
            ids = torch.tensor([req.all_token_ids[p]], dtype=torch.int64)
            pos = torch.tensor([p], dtype=torch.int64)
            hidden = model(ids, pos, caches, meta)
            logits = model.compute_logits(hidden)
            tok = int(logits.argmax(-1))
            engine_tokens.append(tok)
            req.append_output_token(tok)
            req.num_computed_tokens += 1

        # --- oracle: full forward each step ---
        oracle_tokens = []
        seq = list(prompt)
        for _ in range(6):
            logits = full_forward_logits(model, seq)
            tok = int(logits[-1].argmax(-1))
            oracle_tokens.append(tok)
            seq.append(tok)

        assert engine_tokens == oracle_tokens
