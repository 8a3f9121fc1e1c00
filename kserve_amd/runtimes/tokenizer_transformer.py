"""Tokenizer transformer: text -> token ids for a downstream LLM predictor.

Reference parity: python/custom_tokenizer (BERT tokenizer transformer used
in the InferenceGraph samples) — the transformer step of
BASELINE config 5: tokenizer transformer -> Llama predictor via the graph
router. Emits a ready-to-forward /v1/completions request body so a Sequence
node can chain it straight into the LLM service.
"""

from __future__ import annotations

from typing import Dict, List, Optional

from kserve_amd.errors import InvalidInput
from kserve_amd.model import Model


class TokenizerTransformer(Model):
    def __init__(
        self,
        name: str,
        tokenizer=None,
        predictor_model: str = "model",
        max_tokens: int = 64,
    ):
        super().__init__(name)
        self.tokenizer = tokenizer
        self.predictor_model = predictor_model
        self.max_tokens = max_tokens
        self.ready = True

    def _encode(self, text: str) -> List[int]:
        if self.tokenizer is not None:
            return self.tokenizer.encode(text)
        # tokenizer-less fallback for synthetic pipelines: bytes as ids
        return [b % 256 for b in text.encode("utf-8")]

    def predict(self, payload: Dict, headers=None) -> Dict:
        instances = payload.get("instances")
        if not isinstance(instances, list) or not instances:
            raise InvalidInput('Expected non-empty "instances"')
        text = str(instances[0])
        params = payload.get("parameters") or {}
        return {
            "model": params.get("model", self.predictor_model),
            "prompt": self._encode(text),
            "max_tokens": int(params.get("max_tokens", self.max_tokens)),
            "temperature": float(params.get("temperature", 0.0)),
        }
