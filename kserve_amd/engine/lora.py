"""Multi-LoRA adapter serving for the native engine.

Reference parity: huggingfaceserver __main__.py:334-337 (--enable-lora /
--lora-modules name=path register adapter names with the model server; the
vLLM backend applies adapters per request). MI355X-native design: adapters
stay unmerged; per batch the runner records contiguous row segments per
adapter and each parallel layer adds ``(x @ A^T) @ B^T * scale`` for its
segment — two skinny hipBLASLt GEMMs per adapter per module, no sort/gather
kernels, exact under tensor parallelism (column-parallel modules shard B
rows; row-parallel modules shard A columns and add the partial delta before
the xGMI all-reduce).
"""

from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import torch

# module keys the llama forward asks for
COLUMN_MODULES = ("q_proj", "k_proj", "v_proj", "gate_proj", "up_proj")
ROW_MODULES = ("o_proj", "down_proj")
ALL_MODULES = COLUMN_MODULES + ROW_MODULES


@dataclass
class LoRALayerWeights:
    a: torch.Tensor  # [r, in_local]
    b: torch.Tensor  # [out_local, r]
    scale: float

    def delta(self, x: torch.Tensor) -> torch.Tensor:
        return ((x @ self.a.t()) @ self.b.t()) * self.scale


@dataclass
class LoRAAdapter:
    adapter_id: int
    name: str
    rank: int
    # (layer_idx, module) -> weights
    weights: Dict[Tuple[int, str], LoRALayerWeights] = field(default_factory=dict)

    def get(self, layer_idx: int, module: str) -> Optional[LoRALayerWeights]:
        return self.weights.get((layer_idx, module))


@dataclass
class LoRABatchMeta:
    """Per-forward LoRA context: contiguous token-row segments per adapter."""

    # (adapter_id, start_row, end_row) — rows with adapter_id 0 are skipped
    segments: List[Tuple[int, int, int]]
    manager: "LoRAManager"

    def apply(
        self,
        layer_idx: int,
        module: str,
        x: torch.Tensor,   # [T, in_local] layer input
        y: torch.Tensor,   # [T, out] tensor to add the delta into (in place)
        out_offset: int = 0,
    ) -> None:
        for aid, s, e in self.segments:
            w = self.manager.layer_weights(aid, layer_idx, module)
            if w is None:
                continue
            d = w.delta(x[s:e])
            y[s:e, out_offset : out_offset + d.shape[1]] += d

    def delta_for(
        self, layer_idx: int, module: str, x: torch.Tensor
    ) -> Optional[torch.Tensor]:
        """Standalone delta tensor for row-parallel modules (added to the
        local partial before the all-reduce)."""
        out = None
        for aid, s, e in self.segments:
            w = self.manager.layer_weights(aid, layer_idx, module)
            if w is None:
                continue
            if out is None:
                out = torch.zeros(
                    x.shape[0], w.b.shape[0], dtype=x.dtype, device=x.device
                )
            out[s:e] = w.delta(x[s:e])
        return out


class LoRAManager:
    """Loads and holds adapters; resolves names to ids."""

    def __init__(self, device: str, dtype: torch.dtype, tp_rank: int = 0,
                 tp_size: int = 1):
        self.device = device
        self.dtype = dtype
        self.tp_rank = tp_rank
        self.tp_size = tp_size
        self._adapters: Dict[int, LoRAAdapter] = {}
        self._by_name: Dict[str, int] = {}
        self._next_id = 1

    # -- registration --------------------------------------------------------
    def register(self, name: str, path: str) -> int:
        if name in self._by_name:
            return self._by_name[name]
        adapter = self._load(name, path)
        self._adapters[adapter.adapter_id] = adapter
        self._by_name[name] = adapter.adapter_id
        return adapter.adapter_id

    def lookup(self, name: Optional[str]) -> int:
        if not name:
            return 0
        if name not in self._by_name:
            raise KeyError(f"unknown LoRA adapter: {name}")
        return self._by_name[name]

    def names(self) -> List[str]:
        return list(self._by_name)

    def layer_weights(
        self, adapter_id: int, layer_idx: int, module: str
    ) -> Optional[LoRALayerWeights]:
        a = self._adapters.get(adapter_id)
        return a.get(layer_idx, module) if a is not None else None

    # -- loading (HF PEFT format) -------------------------------------------
    def _load(self, name: str, path: str) -> LoRAAdapter:
        cfg_path = os.path.join(path, "adapter_config.json")
        with open(cfg_path) as f:
            cfg = json.load(f)
        r = int(cfg["r"])
        alpha = float(cfg.get("lora_alpha", r))
        scale = alpha / r
        if cfg.get("use_rslora"):
            scale = alpha / (r ** 0.5)
        tensors = self._read_tensors(path)
        adapter = LoRAAdapter(self._next_id, name, r)
        self._next_id += 1
        for key, t in tensors.items():
            parsed = self._parse_key(key)
            if parsed is None:
                continue
            layer_idx, module, which = parsed
            lw = adapter.weights.setdefault(
                (layer_idx, module), LoRALayerWeights(None, None, scale)
            )
            t = t.to(self.dtype)
            if which == "A":
                lw.a = self._shard_a(module, t)
            else:
                lw.b = self._shard_b(module, t)
        # drop incomplete pairs (e.g. modules outside the llama set)
        adapter.weights = {
            k: w
            for k, w in adapter.weights.items()
            if w.a is not None and w.b is not None
        }
        for w in adapter.weights.values():
            w.a = w.a.to(self.device).contiguous()
            w.b = w.b.to(self.device).contiguous()
        return adapter

    def _read_tensors(self, path: str) -> Dict[str, torch.Tensor]:
        st = os.path.join(path, "adapter_model.safetensors")
        if os.path.exists(st):
            from safetensors.torch import load_file

            return load_file(st)
        bin_path = os.path.join(path, "adapter_model.bin")
        if os.path.exists(bin_path):
            return torch.load(bin_path, map_location="cpu", weights_only=True)
        raise FileNotFoundError(f"no adapter weights under {path}")

    @staticmethod
    def _parse_key(key: str) -> Optional[Tuple[int, str, str]]:
        """'...model.layers.{i}.(self_attn|mlp).{mod}.lora_(A|B).weight'"""
        if "lora_A" in key:
            which = "A"
        elif "lora_B" in key:
            which = "B"
        else:
            return None
        parts = key.split(".")
        try:
            li = parts.index("layers")
            layer_idx = int(parts[li + 1])
        except (ValueError, IndexError):
            return None
        module = None
        for m in ALL_MODULES:
            if m in parts:
                module = m
                break
        if module is None:
            return None
        return layer_idx, module, which

    # -- TP sharding ---------------------------------------------------------
    def _shard_b(self, module: str, b: torch.Tensor) -> torch.Tensor:
        """Column-parallel modules shard the B rows (output dim)."""
        if self.tp_size == 1 or module in ROW_MODULES:
            return b
        out = b.shape[0]
        per = out // self.tp_size
        return b[self.tp_rank * per : (self.tp_rank + 1) * per]

    def _shard_a(self, module: str, a: torch.Tensor) -> torch.Tensor:
        """Row-parallel modules shard the A columns (input dim)."""
        if self.tp_size == 1 or module not in ROW_MODULES:
            return a
        inp = a.shape[1]
        per = inp // self.tp_size
        return a[:, self.tp_rank * per : (self.tp_rank + 1) * per]
