"""Tensor-parallel linear layers (Megatron-style sharding, fresh code).

Column-parallel: weight rows (output features) sharded; no comm on forward.
Row-parallel: weight cols (input features) sharded; sum-all-reduce on output.
The GEMMs themselves go through torch.nn.functional.linear — on ROCm that is
hipBLASLt, the sanctioned library path for plain GEMMs; fused hot ops are the
hand-written HIP kernels in kserve_amd/ops.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from kserve_amd import ops
from kserve_amd.parallel import comm


def _divide(a: int, b: int) -> int:
    assert a % b == 0, f"{a} not divisible by {b}"
    return a // b


class ColumnParallelLinear(nn.Module):
    """Y_shard = X @ W_shard^T; output features sharded across TP ranks."""

    def __init__(
        self,
        in_features: int,
        out_features: int,
        bias: bool = False,
        dtype: torch.dtype = torch.bfloat16,
        gather_output: bool = False,
    ):
        super().__init__()
        st = comm.get_state()
        self.tp_size = st.tp_size
        self.in_features = in_features
        self.out_features = out_features
        self.out_per_rank = _divide(out_features, self.tp_size)
        self.gather_output = gather_output
        self.weight = nn.Parameter(
            torch.empty(self.out_per_rank, in_features, dtype=dtype),
            requires_grad=False,
        )
        self.bias = (
            nn.Parameter(
                torch.empty(self.out_per_rank, dtype=dtype), requires_grad=False
            )
            if bias
            else None
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = ops.linear(x, self.weight, self.bias)
        if self.gather_output and self.tp_size > 1:
            y = comm.tp_all_gather(y, dim=-1)
        return y

    def load_shard(self, full_weight: torch.Tensor, full_bias: Optional[torch.Tensor] = None):
        """Slice the full tensor for this rank (rows = output features)."""
        r = comm.get_state().tp_rank
        lo, hi = r * self.out_per_rank, (r + 1) * self.out_per_rank
        self.weight.data.copy_(full_weight[lo:hi].to(self.weight.dtype))
        if self.bias is not None and full_bias is not None:
            self.bias.data.copy_(full_bias[lo:hi].to(self.bias.dtype))


class QKVParallelLinear(nn.Module):
    """Fused QKV projection, head-sharded. Output layout per rank:
    [q_heads_local*D | kv_heads_local*D | kv_heads_local*D]."""

    def __init__(
        self,
        hidden_size: int,
        head_dim: int,
        num_heads: int,
        num_kv_heads: int,
        bias: bool = False,
        dtype: torch.dtype = torch.bfloat16,
    ):
        super().__init__()
        st = comm.get_state()
        self.tp_size = st.tp_size
        self.head_dim = head_dim
        self.num_heads = num_heads
        self.num_kv_heads = num_kv_heads
        self.heads_local = _divide(num_heads, self.tp_size)
        # kv heads replicate when tp_size > num_kv_heads
        if num_kv_heads >= self.tp_size:
            self.kv_heads_local = _divide(num_kv_heads, self.tp_size)
            self.kv_replication = 1
        else:
            self.kv_heads_local = 1
            self.kv_replication = _divide(self.tp_size, num_kv_heads)
        out_local = (self.heads_local + 2 * self.kv_heads_local) * head_dim
        self.q_size = self.heads_local * head_dim
        self.kv_size = self.kv_heads_local * head_dim
        self.weight = nn.Parameter(
            torch.empty(out_local, hidden_size, dtype=dtype), requires_grad=False
        )
        self.bias = (
            nn.Parameter(torch.empty(out_local, dtype=dtype), requires_grad=False)
            if bias
            else None
        )

    def forward(self, x: torch.Tensor):
        qkv = ops.linear(x, self.weight, self.bias)
        return qkv.split([self.q_size, self.kv_size, self.kv_size], dim=-1)

    def load_shards(
        self,
        q_weight: torch.Tensor,
        k_weight: torch.Tensor,
        v_weight: torch.Tensor,
        q_bias: Optional[torch.Tensor] = None,
        k_bias: Optional[torch.Tensor] = None,
        v_bias: Optional[torch.Tensor] = None,
    ):
        r = comm.get_state().tp_rank
        D = self.head_dim
        q_lo = r * self.heads_local * D
        q_hi = q_lo + self.heads_local * D
        kv_rank = r // self.kv_replication
        kv_lo = kv_rank * self.kv_heads_local * D
        kv_hi = kv_lo + self.kv_heads_local * D
        w = torch.cat(
            [q_weight[q_lo:q_hi], k_weight[kv_lo:kv_hi], v_weight[kv_lo:kv_hi]], dim=0
        )
        self.weight.data.copy_(w.to(self.weight.dtype))
        if self.bias is not None and q_bias is not None:
            b = torch.cat(
                [q_bias[q_lo:q_hi], k_bias[kv_lo:kv_hi], v_bias[kv_lo:kv_hi]], dim=0
            )
            self.bias.data.copy_(b.to(self.bias.dtype))


class RowParallelLinear(nn.Module):
    """Y = sum_ranks(X_shard @ W_shard^T); input features sharded; output
    all-reduced (the per-layer RCCL call over xGMI).

    Comm/compute overlap (north star): with ``overlap_chunks`` > 1 the GEMM
    is split along output features and each chunk's all-reduce launches
    async as soon as its GEMM is queued — the next chunk's GEMM overlaps
    the previous chunk's xGMI time (per-link-bound ring). All handles are
    waited before the concatenated result is returned, so numerics are
    identical to the synchronous path.
    """

    # overlap engages above this many output elements (small decode
    # batches are latency-bound on the collective's launch, not its bytes)
    OVERLAP_MIN_NUMEL = 1 << 20

    def __init__(
        self,
        in_features: int,
        out_features: int,
        bias: bool = False,
        dtype: torch.dtype = torch.bfloat16,
        reduce_output: bool = True,
        overlap_chunks: int = 2,
    ):
        super().__init__()
        st = comm.get_state()
        self.tp_size = st.tp_size
        self.in_per_rank = _divide(in_features, self.tp_size)
        self.out_features = out_features
        self.reduce_output = reduce_output
        self.weight = nn.Parameter(
            torch.empty(out_features, self.in_per_rank, dtype=dtype),
            requires_grad=False,
        )
        # bias added once (rank 0 semantics handled by adding after reduce)
        self.bias = (
            nn.Parameter(torch.empty(out_features, dtype=dtype), requires_grad=False)
            if bias
            else None
        )
        self.overlap_chunks = max(1, overlap_chunks)

    def forward(
        self, x: torch.Tensor, delta: Optional[torch.Tensor] = None
    ) -> torch.Tensor:
        use_overlap = (
            self.reduce_output
            and self.tp_size > 1
            and delta is None
            and self.overlap_chunks > 1
            and x.shape[0] * self.out_features >= self.OVERLAP_MIN_NUMEL
        )
        if use_overlap:
            n = self.overlap_chunks
            step = -(-self.out_features // n)
            parts = []
            works = []
            for c in range(n):
                wc = self.weight[c * step : (c + 1) * step]
                if wc.shape[0] == 0:
                    break
                yc = ops.linear(x, wc)
                works.append(comm.tp_all_reduce_async(yc))
                parts.append(yc)
            for w in works:
                if w is not None:
                    w.wait()
            y = torch.cat(parts, dim=-1)
            if self.bias is not None:
                y = y + self.bias
            return y
        y = ops.linear(x, self.weight)
        if delta is not None:
            # per-rank partial (e.g. LoRA with A column-sharded): must be
            # summed by the same all-reduce as the main GEMM
            y = y + delta
        if self.reduce_output:
            y = comm.tp_all_reduce(y)
        if self.bias is not None:
            y = y + self.bias
        return y

    def load_shard(self, full_weight: torch.Tensor, full_bias: Optional[torch.Tensor] = None):
        r = comm.get_state().tp_rank
        lo, hi = r * self.in_per_rank, (r + 1) * self.in_per_rank
        self.weight.data.copy_(full_weight[:, lo:hi].to(self.weight.dtype))
        if self.bias is not None and full_bias is not None:
            self.bias.data.copy_(full_bias.to(self.bias.dtype))


class MergedColumnParallelLinear(nn.Module):
    """Fused gate+up projection (SwiGLU MLP), each half column-sharded.

    Per-rank output layout: [gate_local | up_local] so silu_and_mul can
    split it locally."""

    def __init__(
        self,
        in_features: int,
        out_features_each: int,
        bias: bool = False,
        dtype: torch.dtype = torch.bfloat16,
    ):
        super().__init__()
        st = comm.get_state()
        self.tp_size = st.tp_size
        self.each_per_rank = _divide(out_features_each, self.tp_size)
        self.weight = nn.Parameter(
            torch.empty(2 * self.each_per_rank, in_features, dtype=dtype),
            requires_grad=False,
        )
        self.bias = (
            nn.Parameter(
                torch.empty(2 * self.each_per_rank, dtype=dtype), requires_grad=False
            )
            if bias
            else None
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.linear(x, self.weight, self.bias)

    def load_shards(self, gate_weight: torch.Tensor, up_weight: torch.Tensor):
        r = comm.get_state().tp_rank
        lo, hi = r * self.each_per_rank, (r + 1) * self.each_per_rank
        w = torch.cat([gate_weight[lo:hi], up_weight[lo:hi]], dim=0)
        self.weight.data.copy_(w.to(self.weight.dtype))


class VocabParallelEmbedding(nn.Module):
    """Embedding table sharded over vocab; out-of-shard rows contribute 0 and
    the result is all-reduced."""

    def __init__(
        self, vocab_size: int, hidden_size: int, dtype: torch.dtype = torch.bfloat16
    ):
        super().__init__()
        st = comm.get_state()
        self.tp_size = st.tp_size
        self.vocab_per_rank = _divide(vocab_size, self.tp_size)
        self.vocab_start = st.tp_rank * self.vocab_per_rank
        self.weight = nn.Parameter(
            torch.empty(self.vocab_per_rank, hidden_size, dtype=dtype),
            requires_grad=False,
        )

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        if self.tp_size == 1:
            return F.embedding(ids, self.weight)
        local = ids - self.vocab_start
        mask = (local < 0) | (local >= self.vocab_per_rank)
        local = local.clamp(0, self.vocab_per_rank - 1)
        out = F.embedding(local, self.weight)
        out[mask] = 0
        return comm.tp_all_reduce(out)

    def load_shard(self, full_weight: torch.Tensor):
        lo = self.vocab_start
        hi = lo + self.vocab_per_rank
        self.weight.data.copy_(full_weight[lo:hi].to(self.weight.dtype))
