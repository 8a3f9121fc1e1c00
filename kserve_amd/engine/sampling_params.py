"""Per-request sampling parameters (engine-side mirror of the OpenAI fields).

Semantics match vLLM's documented behavior (the engine the reference
delegates to — vllm_model.py:248-271); implementation is fresh.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Union


@dataclass
class SamplingParams:
    n: int = 1
    temperature: float = 1.0
    top_p: float = 1.0
    top_k: int = -1
    # keep tokens with prob >= min_p * max_prob (0 disables)
    min_p: float = 0.0
    # OpenAI logit_bias: token-id -> additive bias (-100..100)
    logit_bias: Optional[Dict[int, float]] = None
    # guided decoding: "json_object" constrains output to valid JSON;
    # "json_schema" additionally constrains to json_schema below
    response_format: Optional[str] = None
    # guided decoding: JSON Schema for response_format "json_schema"
    json_schema: Optional[Dict] = None
    # guided decoding: constrain output to one of these strings
    guided_choice: Optional[List[str]] = None
    max_tokens: int = 16
    min_tokens: int = 0
    stop: List[str] = field(default_factory=list)
    stop_token_ids: List[int] = field(default_factory=list)
    ignore_eos: bool = False
    seed: Optional[int] = None
    logprobs: Optional[int] = None
    presence_penalty: float = 0.0
    frequency_penalty: float = 0.0
    repetition_penalty: float = 1.0
    echo: bool = False
    # multi-LoRA: adapter name registered with the engine (None = base)
    lora_name: Optional[str] = None
    # scheduling priority (lower value runs first; same-priority FIFO)
    priority: int = 0

    def __post_init__(self):
        if self.temperature < 0:
            raise ValueError("temperature must be >= 0")
        if not 0.0 < self.top_p <= 1.0:
            raise ValueError("top_p must be in (0, 1]")
        if self.top_k == 0 or self.top_k < -1:
            raise ValueError("top_k must be -1 (disabled) or >= 1")
        if not 0.0 <= self.min_p <= 1.0:
            raise ValueError("min_p must be in [0, 1]")
        if self.max_tokens is not None and self.max_tokens < 1:
            raise ValueError("max_tokens must be >= 1")
        if self.n < 1:
            raise ValueError("n must be >= 1")

    @property
    def greedy(self) -> bool:
        return self.temperature == 0.0
