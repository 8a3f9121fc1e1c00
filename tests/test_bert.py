"""BERT encoder numerics vs transformers (CPU fp32) + encoder serving test."""

import json
import os

import numpy as np
import pytest
import torch

transformers = pytest.importorskip("transformers")

from kserve_amd.models.bert import (
    BertConfig,
    BertForMaskedLM,
    BertForSequenceClassification,
    BertModel,
)


def tiny_hf_bert_config():
    return transformers.BertConfig(
        vocab_size=512,
        hidden_size=128,
        num_hidden_layers=2,
        num_attention_heads=2,
        intermediate_size=256,
        max_position_embeddings=128,
        hidden_act="gelu",
    )


def tiny_cfg():
    return BertConfig.tiny()


class TestBertNumerics:
    def test_encoder_matches_hf(self):
        torch.manual_seed(0)
        hf = transformers.BertModel(tiny_hf_bert_config()).eval().float()
        ours = BertModel(tiny_cfg(), dtype=torch.float32)
        ours.load_hf_state_dict(dict(hf.state_dict()), prefix="")

        ids = torch.randint(0, 512, (20,))
        cu = torch.tensor([0, 8, 20], dtype=torch.int32)
        hidden = ours(ids, cu)
        with torch.no_grad():
            h1 = hf(ids[:8].unsqueeze(0)).last_hidden_state[0]
            h2 = hf(ids[8:].unsqueeze(0)).last_hidden_state[0]
        torch.testing.assert_close(hidden[:8], h1, atol=2e-4, rtol=2e-4)
        torch.testing.assert_close(hidden[8:], h2, atol=2e-4, rtol=2e-4)

    def test_masked_lm_matches_hf(self):
        torch.manual_seed(1)
        hf = transformers.BertForMaskedLM(tiny_hf_bert_config()).eval().float()
        ours = BertForMaskedLM(tiny_cfg(), dtype=torch.float32)
        ours.load_hf_state_dict(dict(hf.state_dict()))

        ids = torch.randint(0, 512, (12,))
        cu = torch.tensor([0, 12], dtype=torch.int32)
        logits = ours(ids, cu)
        with torch.no_grad():
            ref = hf(ids.unsqueeze(0)).logits[0]
        torch.testing.assert_close(logits, ref, atol=5e-4, rtol=5e-4)

    def test_sequence_classification_matches_hf(self):
        torch.manual_seed(2)
        hf_cfg = tiny_hf_bert_config()
        hf_cfg.num_labels = 3
        hf = (
            transformers.BertForSequenceClassification(hf_cfg).eval().float()
        )
        ours = BertForSequenceClassification(tiny_cfg(), 3, dtype=torch.float32)
        ours.load_hf_state_dict(dict(hf.state_dict()))
        ids = torch.randint(0, 512, (10,))
        cu = torch.tensor([0, 10], dtype=torch.int32)
        logits = ours(ids, cu)
        with torch.no_grad():
            ref = hf(ids.unsqueeze(0)).logits
        torch.testing.assert_close(logits, ref, atol=5e-4, rtol=5e-4)


class TestBackendDetection:
    def test_detect(self, tmp_path):
        from kserve_amd.runtimes.huggingfaceserver import detect_backend

        d = tmp_path / "m1"
        d.mkdir()
        (d / "config.json").write_text(
            json.dumps({"architectures": ["LlamaForCausalLM"]})
        )
        assert detect_backend(str(d)) == "engine"
        d2 = tmp_path / "m2"
        d2.mkdir()
        (d2 / "config.json").write_text(
            json.dumps({"architectures": ["BertForMaskedLM"]})
        )
        assert detect_backend(str(d2)) == "encoder"
