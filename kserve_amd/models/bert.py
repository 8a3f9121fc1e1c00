"""BERT-class encoder, MI355X-native (BASELINE config 2: BERT fill-mask bf16).

Fresh implementation (reference delegates to HF transformers,
encoder_model.py:71): varlen batching — sequences are concatenated and
attended bidirectionally with the MFMA flash kernel (no padding waste),
LayerNorm/GELU are the fused HIP kernels, GEMMs go through ops.linear.
"""

from __future__ import annotations

import json
import math
import os
from dataclasses import dataclass
from typing import Dict, List, Optional

import torch
import torch.nn as nn

from kserve_amd import ops


@dataclass
class BertConfig:
    vocab_size: int = 30522
    hidden_size: int = 768
    num_layers: int = 12
    num_heads: int = 12
    intermediate_size: int = 3072
    max_position_embeddings: int = 512
    type_vocab_size: int = 2
    layer_norm_eps: float = 1e-12
    pad_token_id: int = 0

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_heads

    @classmethod
    def from_hf_config(cls, path: str) -> "BertConfig":
        with open(path) as f:
            c = json.load(f)
        return cls(
            vocab_size=c["vocab_size"],
            hidden_size=c["hidden_size"],
            num_layers=c["num_hidden_layers"],
            num_heads=c["num_attention_heads"],
            intermediate_size=c["intermediate_size"],
            max_position_embeddings=c["max_position_embeddings"],
            type_vocab_size=c.get("type_vocab_size", 2),
            layer_norm_eps=c.get("layer_norm_eps", 1e-12),
            pad_token_id=c.get("pad_token_id", 0),
        )

    @classmethod
    def tiny(cls) -> "BertConfig":
        return cls(
            vocab_size=512,
            hidden_size=128,
            num_layers=2,
            num_heads=2,
            intermediate_size=256,
            max_position_embeddings=128,
        )


def _param(*shape, dtype):
    return nn.Parameter(torch.empty(*shape, dtype=dtype), requires_grad=False)


class BertLayer(nn.Module):
    def __init__(self, c: BertConfig, dtype):
        super().__init__()
        h = c.hidden_size
        self.qkv_w = _param(3 * h, h, dtype=dtype)
        self.qkv_b = _param(3 * h, dtype=dtype)
        self.attn_out_w = _param(h, h, dtype=dtype)
        self.attn_out_b = _param(h, dtype=dtype)
        self.attn_ln_w = _param(h, dtype=dtype)
        self.attn_ln_b = _param(h, dtype=dtype)
        self.inter_w = _param(c.intermediate_size, h, dtype=dtype)
        self.inter_b = _param(c.intermediate_size, dtype=dtype)
        self.out_w = _param(h, c.intermediate_size, dtype=dtype)
        self.out_b = _param(h, dtype=dtype)
        self.out_ln_w = _param(h, dtype=dtype)
        self.out_ln_b = _param(h, dtype=dtype)
        self.num_heads = c.num_heads
        self.head_dim = c.head_dim
        self.scale = 1.0 / math.sqrt(c.head_dim)
        self.eps = c.layer_norm_eps

    def forward(self, hidden: torch.Tensor, cu_seqlens: torch.Tensor, max_seqlen: int):
        T = hidden.shape[0]
        qkv = ops.linear(hidden, self.qkv_w, self.qkv_b)
        q, k, v = qkv.split(hidden.shape[-1], dim=-1)
        q = q.view(T, self.num_heads, self.head_dim)
        k = k.view(T, self.num_heads, self.head_dim)
        v = v.view(T, self.num_heads, self.head_dim)
        attn = ops.flash_attn_varlen(
            q, k, v, cu_seqlens, max_seqlen, self.scale, causal=False
        )
        attn = ops.linear(attn.reshape(T, -1), self.attn_out_w, self.attn_out_b)
        hidden = ops.fused_add_layer_norm(
            attn, hidden, self.attn_ln_w, self.attn_ln_b, self.eps
        )
        inter = ops.gelu(ops.linear(hidden, self.inter_w, self.inter_b))
        out = ops.linear(inter, self.out_w, self.out_b)
        return ops.fused_add_layer_norm(
            out, hidden, self.out_ln_w, self.out_ln_b, self.eps
        )


class BertModel(nn.Module):
    def __init__(self, config: BertConfig, dtype=torch.float32, device="cpu"):
        super().__init__()
        self.config = config
        self.dtype = dtype
        h = config.hidden_size
        self.word_embeddings = _param(config.vocab_size, h, dtype=dtype)
        self.position_embeddings = _param(
            config.max_position_embeddings, h, dtype=dtype
        )
        self.token_type_embeddings = _param(config.type_vocab_size, h, dtype=dtype)
        self.emb_ln_w = _param(h, dtype=dtype)
        self.emb_ln_b = _param(h, dtype=dtype)
        self.layers = nn.ModuleList(
            [BertLayer(config, dtype) for _ in range(config.num_layers)]
        )
        # pooler (classification)
        self.pooler_w = _param(h, h, dtype=dtype)
        self.pooler_b = _param(h, dtype=dtype)
        self.to(device)

    def forward(
        self,
        input_ids: torch.Tensor,      # [total_tokens] flat varlen
        cu_seqlens: torch.Tensor,     # [num_seqs + 1] int32
        token_type_ids: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        device = input_ids.device
        # per-sequence positions
        lens = cu_seqlens[1:] - cu_seqlens[:-1]
        positions = torch.cat(
            [torch.arange(int(n), device=device) for n in lens]
        )
        emb = self.word_embeddings[input_ids] + self.position_embeddings[positions]
        if token_type_ids is None:
            emb = emb + self.token_type_embeddings[0]
        else:
            emb = emb + self.token_type_embeddings[token_type_ids]
        hidden = ops.layer_norm(emb, self.emb_ln_w, self.emb_ln_b, self.config.layer_norm_eps)
        max_seqlen = int(lens.max())
        for layer in self.layers:
            hidden = layer(hidden, cu_seqlens, max_seqlen)
        return hidden

    def pool(self, hidden: torch.Tensor, cu_seqlens: torch.Tensor) -> torch.Tensor:
        """CLS pooling: tanh(W @ h[CLS])."""
        cls = hidden[cu_seqlens[:-1].long()]
        return torch.tanh(ops.linear(cls, self.pooler_w, self.pooler_b).float()).to(
            hidden.dtype
        )

    # -- HF weight loading ---------------------------------------------------
    @torch.no_grad()
    def load_hf_state_dict(self, sd: Dict[str, torch.Tensor], prefix: str = "bert."):
        def g(name):
            t = sd.get(prefix + name)
            if t is None:
                t = sd[name]
            return t() if callable(t) else t

        dt = self.dtype
        self.word_embeddings.data.copy_(g("embeddings.word_embeddings.weight").to(dt))
        self.position_embeddings.data.copy_(
            g("embeddings.position_embeddings.weight").to(dt)
        )
        self.token_type_embeddings.data.copy_(
            g("embeddings.token_type_embeddings.weight").to(dt)
        )
        self.emb_ln_w.data.copy_(g("embeddings.LayerNorm.weight").to(dt))
        self.emb_ln_b.data.copy_(g("embeddings.LayerNorm.bias").to(dt))
        for i, layer in enumerate(self.layers):
            p = f"encoder.layer.{i}."
            qw = g(p + "attention.self.query.weight")
            kw = g(p + "attention.self.key.weight")
            vw = g(p + "attention.self.value.weight")
            layer.qkv_w.data.copy_(torch.cat([qw, kw, vw], dim=0).to(dt))
            layer.qkv_b.data.copy_(
                torch.cat(
                    [
                        g(p + "attention.self.query.bias"),
                        g(p + "attention.self.key.bias"),
                        g(p + "attention.self.value.bias"),
                    ]
                ).to(dt)
            )
            layer.attn_out_w.data.copy_(g(p + "attention.output.dense.weight").to(dt))
            layer.attn_out_b.data.copy_(g(p + "attention.output.dense.bias").to(dt))
            layer.attn_ln_w.data.copy_(
                g(p + "attention.output.LayerNorm.weight").to(dt)
            )
            layer.attn_ln_b.data.copy_(g(p + "attention.output.LayerNorm.bias").to(dt))
            layer.inter_w.data.copy_(g(p + "intermediate.dense.weight").to(dt))
            layer.inter_b.data.copy_(g(p + "intermediate.dense.bias").to(dt))
            layer.out_w.data.copy_(g(p + "output.dense.weight").to(dt))
            layer.out_b.data.copy_(g(p + "output.dense.bias").to(dt))
            layer.out_ln_w.data.copy_(g(p + "output.LayerNorm.weight").to(dt))
            layer.out_ln_b.data.copy_(g(p + "output.LayerNorm.bias").to(dt))
        if prefix + "pooler.dense.weight" in sd or "pooler.dense.weight" in sd:
            try:
                self.pooler_w.data.copy_(g("pooler.dense.weight").to(dt))
                self.pooler_b.data.copy_(g("pooler.dense.bias").to(dt))
            except KeyError:
                self.pooler_w.data.zero_()
                self.pooler_b.data.zero_()
        else:
            self.pooler_w.data.zero_()
            self.pooler_b.data.zero_()


class BertForMaskedLM(nn.Module):
    """MLM head on top (fill-mask task)."""

    def __init__(self, config: BertConfig, dtype=torch.float32, device="cpu"):
        super().__init__()
        self.bert = BertModel(config, dtype=dtype, device="cpu")
        h = config.hidden_size
        self.transform_w = _param(h, h, dtype=dtype)
        self.transform_b = _param(h, dtype=dtype)
        self.transform_ln_w = _param(h, dtype=dtype)
        self.transform_ln_b = _param(h, dtype=dtype)
        self.decoder_bias = _param(config.vocab_size, dtype=dtype)
        self.eps = config.layer_norm_eps
        self.to(device)

    def forward(self, input_ids, cu_seqlens, token_type_ids=None):
        hidden = self.bert(input_ids, cu_seqlens, token_type_ids)
        t = ops.gelu(ops.linear(hidden, self.transform_w, self.transform_b))
        t = ops.layer_norm(t, self.transform_ln_w, self.transform_ln_b, self.eps)
        logits = ops.linear(t, self.bert.word_embeddings) + self.decoder_bias
        return logits

    @torch.no_grad()
    def load_hf_state_dict(self, sd: Dict[str, torch.Tensor]):
        self.bert.load_hf_state_dict(sd)
        dt = self.bert.dtype

        def g(name):
            t = sd[name]
            return t() if callable(t) else t

        self.transform_w.data.copy_(
            g("cls.predictions.transform.dense.weight").to(dt)
        )
        self.transform_b.data.copy_(g("cls.predictions.transform.dense.bias").to(dt))
        self.transform_ln_w.data.copy_(
            g("cls.predictions.transform.LayerNorm.weight").to(dt)
        )
        self.transform_ln_b.data.copy_(
            g("cls.predictions.transform.LayerNorm.bias").to(dt)
        )
        key = "cls.predictions.decoder.bias"
        if key not in sd:
            key = "cls.predictions.bias"
        self.decoder_bias.data.copy_(g(key).to(dt))


class BertForSequenceClassification(nn.Module):
    def __init__(self, config: BertConfig, num_labels: int, dtype=torch.float32, device="cpu"):
        super().__init__()
        self.bert = BertModel(config, dtype=dtype, device="cpu")
        self.classifier_w = _param(num_labels, config.hidden_size, dtype=dtype)
        self.classifier_b = _param(num_labels, dtype=dtype)
        self.to(device)

    def forward(self, input_ids, cu_seqlens, token_type_ids=None):
        hidden = self.bert(input_ids, cu_seqlens, token_type_ids)
        pooled = self.bert.pool(hidden, cu_seqlens)
        return ops.linear(pooled, self.classifier_w, self.classifier_b)

    @torch.no_grad()
    def load_hf_state_dict(self, sd):
        self.bert.load_hf_state_dict(sd)
        dt = self.bert.dtype
        self.classifier_w.data.copy_(sd["classifier.weight"].to(dt))
        self.classifier_b.data.copy_(sd["classifier.bias"].to(dt))
