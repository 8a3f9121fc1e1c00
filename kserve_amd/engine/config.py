"""Engine configuration.

The native LLM engine replaces the reference's delegation to vLLM
(reference: python/huggingfaceserver vllm/vllm_model.py:55-343). Config
surface mirrors the preset flags the control plane renders
(--tensor-parallel-size etc., config-llm-worker-data-parallel.yaml:188-199)
while the internals are MI355X-first: KV sizing against 288 GB HBM3E.
"""

from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from typing import Optional

from kserve_amd.constants import (
    DEFAULT_GPU_MEMORY_UTILIZATION,
    DEFAULT_KV_BLOCK_SIZE,
    HBM_BYTES_PER_GPU,
)


@dataclass
class ModelConfig:
    """Decoder-only transformer architecture description (Llama family)."""

    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    head_dim: Optional[int] = None
    rms_norm_eps: float = 1e-5
    rope_theta: float = 500000.0
    max_position_embeddings: int = 8192
    tie_word_embeddings: bool = False
    dtype: str = "bfloat16"
    model_name: str = "llama"
    # attention bias (Qwen2-style) / mlp bias
    attention_bias: bool = False
    mlp_bias: bool = False
    # mixture-of-experts (Mixtral): 0 experts = dense MLP
    num_local_experts: int = 0
    num_experts_per_tok: int = 2
    # EP: partition experts across the parallel group (full-width weights)
    # instead of TP-sharding every expert
    expert_parallel: bool = False
    # HF rope_scaling dict (Llama-3.1 'llama3' type supported)
    rope_scaling: Optional[dict] = None
    # Mistral-style sliding-window size from the HF config; enforced in
    # all three attention kernels (decode/flash/paged-context) as an
    # absolute-position bound since round 2.
    sliding_window: int = None

    def __post_init__(self):
        if self.head_dim is None:
            self.head_dim = self.hidden_size // self.num_heads

    @property
    def kv_bytes_per_token_per_layer(self) -> int:
        # K + V, bf16
        return 2 * self.num_kv_heads * self.head_dim * 2

    @classmethod
    def llama3_8b(cls) -> "ModelConfig":
        return cls(
            vocab_size=128256,
            hidden_size=4096,
            intermediate_size=14336,
            num_layers=32,
            num_heads=32,
            num_kv_heads=8,
            rope_theta=500000.0,
            max_position_embeddings=8192,
            model_name="llama-3-8b",
        )

    @classmethod
    def llama3_70b(cls) -> "ModelConfig":
        return cls(
            vocab_size=128256,
            hidden_size=8192,
            intermediate_size=28672,
            num_layers=80,
            num_heads=64,
            num_kv_heads=8,
            rope_theta=500000.0,
            max_position_embeddings=8192,
            model_name="llama-3-70b",
        )

    @classmethod
    def mistral_7b(cls) -> "ModelConfig":
        return cls(
            vocab_size=32000,
            hidden_size=4096,
            intermediate_size=14336,
            num_layers=32,
            num_heads=32,
            num_kv_heads=8,
            rope_theta=10000.0,
            max_position_embeddings=8192,
            sliding_window=4096,
            model_name="mistral-7b",
        )

    @classmethod
    def mixtral_8x7b(cls) -> "ModelConfig":
        return cls(
            vocab_size=32000,
            hidden_size=4096,
            intermediate_size=14336,
            num_layers=32,
            num_heads=32,
            num_kv_heads=8,
            rope_theta=1e6,
            max_position_embeddings=8192,
            num_local_experts=8,
            num_experts_per_tok=2,
            model_name="mixtral-8x7b",
        )

    @classmethod
    def tiny(cls, vocab_size: int = 256) -> "ModelConfig":
        """Small config for CPU tests."""
        return cls(
            vocab_size=vocab_size,
            hidden_size=64,
            intermediate_size=128,
            num_layers=2,
            num_heads=4,
            num_kv_heads=2,
            max_position_embeddings=512,
            rope_theta=10000.0,
            model_name="tiny-llama",
        )

    @classmethod
    def from_hf_config(cls, config_path: str) -> "ModelConfig":
        """Parse a HuggingFace config.json (model dir contract /mnt/models)."""
        with open(config_path) as f:
            cfg = json.load(f)
        return cls(
            vocab_size=cfg["vocab_size"],
            hidden_size=cfg["hidden_size"],
            intermediate_size=cfg["intermediate_size"],
            num_layers=cfg["num_hidden_layers"],
            num_heads=cfg["num_attention_heads"],
            num_kv_heads=cfg.get("num_key_value_heads", cfg["num_attention_heads"]),
            head_dim=cfg.get("head_dim"),
            rms_norm_eps=cfg.get("rms_norm_eps", 1e-5),
            rope_theta=cfg.get("rope_theta", 10000.0),
            max_position_embeddings=cfg.get("max_position_embeddings", 8192),
            tie_word_embeddings=cfg.get("tie_word_embeddings", False),
            # Qwen2 has qkv bias implicitly (no attention_bias key in HF)
            attention_bias=cfg.get(
                "attention_bias", cfg.get("model_type") == "qwen2"
            ),
            num_local_experts=cfg.get("num_local_experts", 0),
            num_experts_per_tok=cfg.get("num_experts_per_tok", 2),
            sliding_window=cfg.get("sliding_window"),
            rope_scaling=cfg.get("rope_scaling"),
            mlp_bias=cfg.get("mlp_bias", False),
            model_name=cfg.get("_name_or_path", os.path.dirname(config_path) or "model"),
        )


@dataclass
class CacheConfig:
    """Paged-KV cache sizing. Defaults size the pool from the 288 GB HBM3E
    budget after weights (MI355X-first: large pools, big batches)."""

    block_size: int = DEFAULT_KV_BLOCK_SIZE
    num_gpu_blocks: Optional[int] = None  # None -> derive from memory util
    gpu_memory_utilization: float = DEFAULT_GPU_MEMORY_UTILIZATION
    # host-DRAM offload tier (pinned memory, hipMemcpyAsync side stream)
    num_cpu_blocks: int = 0
    cpu_offload_bytes: int = 0
    # automatic prefix caching: content-addressed full prompt blocks are
    # shared across requests; cache-hit prompts only compute their suffix
    # through the paged-context prefill kernel
    enable_prefix_caching: bool = False
    # "auto" stores KV in the model dtype (bf16). "fp8" stores OCP E4M3
    # bytes (scale 1.0): half the KV bandwidth/capacity; attention math
    # stays fp32/bf16 after in-register conversion. GQA group <= 4, D=128.
    kv_cache_dtype: str = "auto"

    def cache_torch_dtype(self, model_dtype):
        if self.kv_cache_dtype in ("fp8", "fp8_e4m3"):
            import torch

            return torch.float8_e4m3fn
        return model_dtype

    def derive_num_gpu_blocks(
        self, model: ModelConfig, tp_size: int = 1, weight_bytes: Optional[int] = None,
        free_bytes: Optional[int] = None,
    ) -> int:
        if self.num_gpu_blocks is not None:
            return self.num_gpu_blocks
        per_block = (
            model.kv_bytes_per_token_per_layer
            * self.block_size
            * model.num_layers
            // tp_size
        )
        if free_bytes is None:
            total = HBM_BYTES_PER_GPU
            if weight_bytes is None:
                weight_bytes = 0
            free_bytes = int(total * self.gpu_memory_utilization) - weight_bytes
        return max(free_bytes // per_block, 16)


@dataclass
class SchedulerConfig:
    max_num_seqs: int = 256
    max_num_batched_tokens: int = 8192
    max_model_len: int = 8192
    # chunked prefill: cap on prompt tokens scheduled per step
    enable_chunked_prefill: bool = False
    # multi-step decode: consecutive greedy decode iterations run as pure
    # hipGraph replays (sampled token fed back on-GPU); stop conditions are
    # applied after the window. 1 disables.
    multi_step: int = 8
    # speculative decoding (prompt-lookup/n-gram): max draft tokens verified
    # per step through the paged-context prefill path. 0 disables. When >0 it
    # replaces multi-step windows for greedy batches (exact same outputs —
    # rejected drafts are corrected by the verify forward).
    speculative_ngram: int = 0
    speculative_ngram_min: int = 2
    speculative_ngram_max: int = 3
    # draft-MODEL speculation: tokens proposed per round by the draft
    # model configured in EngineConfig.draft_model. 0 disables.
    speculative_k: int = 0


@dataclass
class ParallelConfig:
    """Mirrors the CRD ParallelismSpec (reference v1alpha2 types :733-759)."""

    tensor_parallel_size: int = 1
    pipeline_parallel_size: int = 1
    data_parallel_size: int = 1
    expert_parallel: bool = False


@dataclass
class EngineConfig:
    model: ModelConfig = field(default_factory=ModelConfig.llama3_8b)
    cache: CacheConfig = field(default_factory=CacheConfig)
    scheduler: SchedulerConfig = field(default_factory=SchedulerConfig)
    parallel: ParallelConfig = field(default_factory=ParallelConfig)
    device: str = "cuda"
    seed: int = 0
    enforce_eager: bool = False  # True disables hipGraph capture
    # weights: path to model dir (safetensors) or None for random init
    model_path: Optional[str] = None
    # draft-model speculation: the small model's config (random-init unless
    # draft_model_path / engine.draft.load_hf_state_dict provides weights);
    # pairs with scheduler.speculative_k
    draft_model: Optional[ModelConfig] = None
    draft_model_path: Optional[str] = None
    eos_token_id: int = 128001
