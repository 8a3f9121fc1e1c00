"""Llama-family decoder, MI355X-native.

Fresh implementation of the architecture (replaces the reference's delegation
to vLLM/HF — SURVEY.md §2.7 op inventory). Forward is built from:
- kserve_amd.ops HIP kernels: fused residual+RMSNorm, RoPE (host-precomputed
  cos/sin), flash prefill, paged decode attention, reshape_and_cache,
  silu_and_mul
- hipBLASLt GEMMs via F.linear inside the TP-sharded layers
- RCCL all-reduce over xGMI at the o_proj / down_proj boundaries (2/layer)
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import List, Optional, Tuple

import torch
import torch.nn as nn

from kserve_amd import ops
from kserve_amd.engine.config import ModelConfig
from kserve_amd.ops.torch_ref import make_cos_sin_cache
from kserve_amd.parallel import comm
from kserve_amd.parallel.layers import (
    ColumnParallelLinear,
    MergedColumnParallelLinear,
    QKVParallelLinear,
    RowParallelLinear,
    VocabParallelEmbedding,
)


@dataclass
class AttentionMetadata:
    """Per-step attention inputs prepared by the model runner."""

    is_prefill: bool
    slot_mapping: torch.Tensor  # [num_tokens] int32 — where new KV goes
    # prefill
    cu_seqlens: Optional[torch.Tensor] = None  # [num_seqs+1] int32
    max_seqlen: int = 0
    # decode — and chunked prefill, where block_tables/context_lens describe
    # the paged context the chunk attends to (context_lens includes the chunk)
    block_tables: Optional[torch.Tensor] = None  # [num_seqs, max_blocks] int32
    context_lens: Optional[torch.Tensor] = None  # [num_seqs] int32
    # multi-LoRA: per-batch adapter segments (engine/lora.LoRABatchMeta)
    lora: Optional[object] = None


class LlamaAttention(nn.Module):
    def __init__(self, config: ModelConfig, dtype: torch.dtype, layer_idx: int = 0):
        super().__init__()
        self.layer_idx = layer_idx
        st = comm.get_state()
        self.head_dim = config.head_dim
        self.scale = 1.0 / math.sqrt(config.head_dim)
        # Mistral-style sliding window (0/None = full attention); applied
        # in-kernel as a lower bound on attended positions
        self.sliding_window = int(getattr(config, "sliding_window", 0) or 0)
        self.qkv_proj = QKVParallelLinear(
            config.hidden_size,
            config.head_dim,
            config.num_heads,
            config.num_kv_heads,
            bias=config.attention_bias,
            dtype=dtype,
        )
        self.o_proj = RowParallelLinear(
            config.num_heads * config.head_dim,
            config.hidden_size,
            bias=False,
            dtype=dtype,
        )
        self.num_heads_local = self.qkv_proj.heads_local
        self.num_kv_heads_local = self.qkv_proj.kv_heads_local

    def forward(
        self,
        hidden: torch.Tensor,  # [T, H]
        positions: torch.Tensor,  # [T]
        cos_sin_cache: torch.Tensor,
        kv_cache: Tuple[torch.Tensor, torch.Tensor],
        meta: AttentionMetadata,
    ) -> torch.Tensor:
        T = hidden.shape[0]
        q, k, v = self.qkv_proj(hidden)
        if meta.lora is not None:
            meta.lora.apply(self.layer_idx, "q_proj", hidden, q)
            meta.lora.apply(self.layer_idx, "k_proj", hidden, k)
            meta.lora.apply(self.layer_idx, "v_proj", hidden, v)
        # strided views into the fused qkv output; kernels take row strides
        q = q.view(T, self.num_heads_local, self.head_dim)
        k = k.view(T, self.num_kv_heads_local, self.head_dim)
        v = v.view(T, self.num_kv_heads_local, self.head_dim)
        q, k = ops.rotary_embedding(positions, q, k, cos_sin_cache)
        k_cache, v_cache = kv_cache
        if k_cache.numel() > 0:
            ops.reshape_and_cache(k, v, k_cache, v_cache, meta.slot_mapping)
        if meta.is_prefill:
            if meta.context_lens is not None:
                # chunked prefill: attend to the whole paged context (past
                # chunks' KV + this chunk's, written to pages above)
                out = ops.context_attention_varlen(
                    q, k_cache, v_cache, meta.block_tables, meta.cu_seqlens,
                    meta.context_lens, meta.max_seqlen, self.scale,
                    window=self.sliding_window,
                )
            else:
                out = ops.flash_prefill_varlen(
                    q, k, v, meta.cu_seqlens, meta.max_seqlen, self.scale,
                    window=self.sliding_window,
                )
        else:
            out = ops.paged_attention_decode(
                q, k_cache, v_cache, meta.block_tables, meta.context_lens,
                self.scale, window=self.sliding_window,
            )
        o_in = out.reshape(T, -1)
        o_delta = (
            meta.lora.delta_for(self.layer_idx, "o_proj", o_in)
            if meta.lora is not None
            else None
        )
        return self.o_proj(o_in, o_delta)


class LlamaMLP(nn.Module):
    def __init__(self, config: ModelConfig, dtype: torch.dtype, layer_idx: int = 0):
        super().__init__()
        self.layer_idx = layer_idx
        self.gate_up_proj = MergedColumnParallelLinear(
            config.hidden_size,
            config.intermediate_size,
            bias=config.mlp_bias,
            dtype=dtype,
        )
        self.down_proj = RowParallelLinear(
            config.intermediate_size, config.hidden_size, bias=False, dtype=dtype
        )

    def forward(self, x: torch.Tensor, meta=None) -> torch.Tensor:
        gate_up = self.gate_up_proj(x)
        lora = meta.lora if meta is not None else None
        if lora is not None:
            half = gate_up.shape[-1] // 2
            lora.apply(self.layer_idx, "gate_proj", x, gate_up, 0)
            lora.apply(self.layer_idx, "up_proj", x, gate_up, half)
        act = ops.silu_and_mul(gate_up)
        d_delta = (
            lora.delta_for(self.layer_idx, "down_proj", act)
            if lora is not None
            else None
        )
        return self.down_proj(act, d_delta)


class MixtralMoE(nn.Module):
    """Sparse mixture-of-experts MLP block (Mixtral architecture).

    Reference parity: the Mixtral family the reference's vLLM backend
    serves. MI355X-native v2: per-expert weights live in STACKED tensors
    ([E_local, out, in]) so decode runs as two strided-batched GEMMs
    (torch.bmm -> hipBLASLt) over every local expert with the routing
    weights applied as a zero-masked mix — static shapes, no host sync,
    hipGraph-capturable. Prefill keeps the token-bucketed sparse loop
    (fewer FLOPs at large T). Under TP each expert is column/row-sharded
    like the dense MLP and ONE all-reduce covers the whole block; EP
    partitions full-width experts across the group instead.
    """

    # decode batches up to this size take the dense-bmm path (weight reads
    # dominate there, so the E/top_k FLOP overcompute is ~free); larger
    # batches use the sparse loop
    DENSE_MAX_TOKENS = 384

    def __init__(
        self,
        config: ModelConfig,
        dtype: torch.dtype,
        layer_idx: int = 0,
        expert_parallel: bool = False,
    ):
        super().__init__()
        st = comm.get_state()
        self.layer_idx = layer_idx
        self.num_experts = config.num_local_experts
        self.top_k = config.num_experts_per_tok
        H, inter = config.hidden_size, config.intermediate_size
        # expert parallelism: experts are PARTITIONED across the group
        # (full-width weights per expert) instead of every expert being
        # TP-sharded; the per-block all-reduce sums the partial outputs
        # either way (reference --enable-expert-parallel / ParallelismSpec
        # .Expert — vLLM EP there, RCCL all-reduce here).
        self.expert_parallel = expert_parallel and st.tp_size > 1
        if self.expert_parallel:
            assert self.num_experts % st.tp_size == 0, (
                "num_local_experts must divide the group size for EP"
            )
            self.experts_per_rank = self.num_experts // st.tp_size
            self.expert_lo = st.tp_rank * self.experts_per_rank
            self.inter_local = inter
        else:
            self.experts_per_rank = self.num_experts
            self.expert_lo = 0
            assert inter % st.tp_size == 0
            self.inter_local = inter // st.tp_size
        self.gate = nn.Parameter(
            torch.empty(self.num_experts, H, dtype=dtype),
            requires_grad=False,
        )
        # stacked per-expert weights: [E_local, 2*I_local, H] (gate|up) and
        # [E_local, H, I_local]
        self.w_gate_up = nn.Parameter(
            torch.empty(
                self.experts_per_rank, 2 * self.inter_local, H, dtype=dtype
            ),
            requires_grad=False,
        )
        self.w_down = nn.Parameter(
            torch.empty(
                self.experts_per_rank, H, self.inter_local, dtype=dtype
            ),
            requires_grad=False,
        )

    def load_expert(
        self, local_e: int, w1: torch.Tensor, w3: torch.Tensor,
        w2: torch.Tensor,
    ) -> None:
        """Load one expert from full-width HF tensors (w1=gate [I,H],
        w3=up [I,H], w2=down [H,I]), applying this rank's shard."""
        dt = self.w_gate_up.dtype
        if self.expert_parallel:
            self.w_gate_up.data[local_e].copy_(
                torch.cat([w1, w3], dim=0).to(dt)
            )
            self.w_down.data[local_e].copy_(w2.to(dt))
            return
        r = comm.get_state().tp_rank
        lo, hi = r * self.inter_local, (r + 1) * self.inter_local
        self.w_gate_up.data[local_e, : self.inter_local].copy_(w1[lo:hi].to(dt))
        self.w_gate_up.data[local_e, self.inter_local :].copy_(w3[lo:hi].to(dt))
        self.w_down.data[local_e].copy_(w2[:, lo:hi].to(dt))

    def _forward_dense(self, x, topi, topw):
        """Capture-safe grouped path: every local expert runs on every
        token (two bmm), the router weights zero out non-selected pairs.
        Exact (routing enters only through the final mix)."""
        E = self.experts_per_rank
        T, H = x.shape
        w_dense = torch.zeros(
            T, self.num_experts, dtype=torch.float32, device=x.device
        )
        w_dense.scatter_add_(1, topi, topw)
        w_loc = w_dense[:, self.expert_lo : self.expert_lo + E]  # [T, E]
        xb = x.unsqueeze(0).expand(E, T, H)
        gu = torch.bmm(xb, self.w_gate_up.transpose(1, 2))  # [E, T, 2I]
        act = ops.silu_and_mul(gu.reshape(E * T, -1)).reshape(E, T, -1)
        ye = torch.bmm(act, self.w_down.transpose(1, 2))  # [E, T, H]
        return torch.einsum(
            "eth,te->th", ye.float(), w_loc
        ).to(x.dtype)

    def _forward_sparse(self, x, topi, topw):
        """Token-bucketed loop (prefill): exact FLOPs, host-side gather."""
        T = x.shape[0]
        out = torch.zeros_like(x)
        flat_i = topi.flatten()
        flat_w = topw.flatten()
        token_idx = torch.arange(T, device=x.device).repeat_interleave(
            self.top_k
        )
        for le in range(self.experts_per_rank):
            e = self.expert_lo + le
            sel = (flat_i == e).nonzero(as_tuple=True)[0]
            if sel.numel() == 0:
                continue
            rows = token_idx[sel]
            xe = x.index_select(0, rows)
            gu = torch.nn.functional.linear(xe, self.w_gate_up[le])
            act = ops.silu_and_mul(gu)
            ye = torch.nn.functional.linear(act, self.w_down[le])
            out.index_add_(
                0, rows, ye * flat_w[sel].unsqueeze(1).to(ye.dtype)
            )
        return out

    def forward(self, x: torch.Tensor, meta=None) -> torch.Tensor:
        T = x.shape[0]
        router_logits = torch.nn.functional.linear(x.float(), self.gate.float())
        probs = torch.softmax(router_logits, dim=-1)
        topw, topi = probs.topk(self.top_k, dim=-1)
        topw = topw / topw.sum(dim=-1, keepdim=True)  # mixtral renorm
        capturing = (
            x.is_cuda and torch.cuda.is_current_stream_capturing()
        )
        if capturing or T <= self.DENSE_MAX_TOKENS:
            out = self._forward_dense(x, topi, topw)
        else:
            out = self._forward_sparse(x, topi, topw)
        if comm.get_state().tp_size > 1:
            out = comm.tp_all_reduce(out)
        return out


class LlamaDecoderLayer(nn.Module):
    def __init__(self, config: ModelConfig, dtype: torch.dtype, layer_idx: int = 0):
        super().__init__()
        self.self_attn = LlamaAttention(config, dtype, layer_idx)
        self.mlp = (
            MixtralMoE(
                config, dtype, layer_idx,
                expert_parallel=getattr(config, "expert_parallel", False),
            )
            if config.num_local_experts > 0
            else LlamaMLP(config, dtype, layer_idx)
        )
        self.input_layernorm = nn.Parameter(
            torch.empty(config.hidden_size, dtype=dtype), requires_grad=False
        )
        self.post_attention_layernorm = nn.Parameter(
            torch.empty(config.hidden_size, dtype=dtype), requires_grad=False
        )
        self.eps = config.rms_norm_eps

    def forward(self, hidden, residual, positions, cos_sin_cache, kv_cache, meta):
        if residual is None:
            residual = hidden
            hidden = ops.rms_norm(hidden, self.input_layernorm, self.eps)
        else:
            hidden, residual = ops.fused_add_rms_norm(
                hidden, residual, self.input_layernorm, self.eps
            )
        hidden = self.self_attn(hidden, positions, cos_sin_cache, kv_cache, meta)
        hidden, residual = ops.fused_add_rms_norm(
            hidden, residual, self.post_attention_layernorm, self.eps
        )
        hidden = self.mlp(hidden, meta)
        return hidden, residual


class LlamaForCausalLM(nn.Module):
    def __init__(
        self,
        config: ModelConfig,
        dtype: Optional[torch.dtype] = None,
        device: str = "cpu",
    ):
        super().__init__()
        self.config = config
        dtype = dtype or (
            torch.bfloat16 if config.dtype == "bfloat16" else torch.float32
        )
        self.dtype = dtype
        # pipeline parallelism: this rank holds layers [pp_first, pp_last);
        # embeddings live on the first stage, norm + lm_head on the last
        st = comm.get_state()
        per = -(-config.num_layers // st.pp_size)
        self.pp_first = st.pp_rank * per
        self.pp_last = min(config.num_layers, self.pp_first + per)
        self.is_first_stage = st.is_first_pp
        self.is_last_stage = st.is_last_pp
        self.embed_tokens = (
            VocabParallelEmbedding(
                config.vocab_size, config.hidden_size, dtype=dtype
            )
            if self.is_first_stage
            else None
        )
        self.layers = nn.ModuleList(
            [
                LlamaDecoderLayer(config, dtype, i)
                for i in range(self.pp_first, self.pp_last)
            ]
        )
        self.num_local_layers = len(self.layers)
        self.norm = (
            nn.Parameter(
                torch.empty(config.hidden_size, dtype=dtype), requires_grad=False
            )
            if self.is_last_stage
            else None
        )
        self.lm_head = (
            ColumnParallelLinear(
                config.hidden_size, config.vocab_size, dtype=dtype,
                gather_output=True,
            )
            if self.is_last_stage
            else None
        )
        self.register_buffer(
            "cos_sin_cache",
            make_cos_sin_cache(
                config.head_dim,
                config.max_position_embeddings,
                config.rope_theta,
                dtype=torch.float32,
                rope_scaling=getattr(config, "rope_scaling", None),
            ),
            persistent=False,
        )
        self.eps = config.rms_norm_eps
        self.to(device)

    # -- init ---------------------------------------------------------------
    @torch.no_grad()
    def random_init(self, std: float = 0.02, seed: int = 0):
        """Fast random init for the synthetic benchmark (BASELINE.json:
        'random-init weights')."""
        dev = next(self.parameters()).device
        gen = torch.Generator(device=dev)
        gen.manual_seed(seed + comm.get_state().tp_rank)
        for p in self.parameters():
            if p.dim() >= 2:
                p.normal_(0.0, std, generator=gen)
            else:
                p.fill_(1.0)
        for buf_name in ():
            pass
        return self

    # -- forward --------------------------------------------------------------
    def forward(
        self,
        input_ids: torch.Tensor,  # [T]
        positions: torch.Tensor,  # [T]
        kv_caches: List[Tuple[torch.Tensor, torch.Tensor]],
        meta: AttentionMetadata,
    ) -> torch.Tensor:
        st = comm.get_state()
        if self.is_first_stage:
            hidden = self.embed_tokens(input_ids)
            residual = None
        else:
            # synchronous pipeline: activations + residual arrive from the
            # previous stage over RCCL p2p (xGMI)
            T = input_ids.shape[0]
            H = self.config.hidden_size
            hidden = comm.pp_recv((T, H), self.dtype, input_ids.device)
            residual = comm.pp_recv((T, H), self.dtype, input_ids.device)
        for i, layer in enumerate(self.layers):
            hidden, residual = layer(
                hidden, residual, positions, self.cos_sin_cache, kv_caches[i], meta
            )
        if not self.is_last_stage:
            comm.pp_send(hidden)
            comm.pp_send(residual)
            return hidden
        hidden, _ = ops.fused_add_rms_norm(hidden, residual, self.norm, self.eps)
        return hidden

    def compute_logits(self, hidden: torch.Tensor):
        """hidden: [N, H] (already gathered to sampled positions).
        None on non-final pipeline stages."""
        if self.lm_head is None:
            return None
        return self.lm_head(hidden)

    # -- HF weight loading ------------------------------------------------------
    @torch.no_grad()
    def load_hf_state_dict(self, tensors) -> None:
        """Load HuggingFace-format llama weights.

        ``tensors``: mapping name -> tensor (may be lazily materialized).
        TP sharding applied per layer type.
        """
        def get(name):
            t = tensors[name]
            return t() if callable(t) else t

        if self.embed_tokens is not None:
            self.embed_tokens.load_shard(get("model.embed_tokens.weight"))
        if self.norm is not None:
            self.norm.data.copy_(get("model.norm.weight").to(self.dtype))
        if self.lm_head is not None:
            if self.config.tie_word_embeddings or "lm_head.weight" not in tensors:
                self.lm_head.load_shard(get("model.embed_tokens.weight"))
            else:
                self.lm_head.load_shard(get("lm_head.weight"))
        for i, layer in enumerate(self.layers):
            p = f"model.layers.{self.pp_first + i}."
            layer.input_layernorm.data.copy_(
                get(p + "input_layernorm.weight").to(self.dtype)
            )
            layer.post_attention_layernorm.data.copy_(
                get(p + "post_attention_layernorm.weight").to(self.dtype)
            )
            qb = kb = vb = None
            if self.config.attention_bias:
                qb = get(p + "self_attn.q_proj.bias")
                kb = get(p + "self_attn.k_proj.bias")
                vb = get(p + "self_attn.v_proj.bias")
            layer.self_attn.qkv_proj.load_shards(
                get(p + "self_attn.q_proj.weight"),
                get(p + "self_attn.k_proj.weight"),
                get(p + "self_attn.v_proj.weight"),
                qb,
                kb,
                vb,
            )
            layer.self_attn.o_proj.load_shard(get(p + "self_attn.o_proj.weight"))
            if isinstance(layer.mlp, MixtralMoE):
                moe = layer.mlp
                owned = range(
                    moe.expert_lo, moe.expert_lo + moe.experts_per_rank
                )
                if (p + "block_sparse_moe.gate.weight") in tensors:
                    # classic Mixtral checkpoint: experts.j.w1/w3/w2
                    moe.gate.data.copy_(
                        get(p + "block_sparse_moe.gate.weight").to(moe.gate.dtype)
                    )
                    for le, j in enumerate(owned):
                        ep = p + f"block_sparse_moe.experts.{j}."
                        moe.load_expert(
                            le,
                            get(ep + "w1.weight"),
                            get(ep + "w3.weight"),
                            get(ep + "w2.weight"),
                        )
                else:
                    # fused layout (transformers >= 4.5x): mlp.gate.weight,
                    # mlp.experts.gate_up_proj [E, 2I, H], .down_proj [E, H, I]
                    moe.gate.data.copy_(
                        get(p + "mlp.gate.weight").to(moe.gate.dtype)
                    )
                    gu = get(p + "mlp.experts.gate_up_proj")
                    dn = get(p + "mlp.experts.down_proj")
                    inter = gu.shape[1] // 2
                    for le, j in enumerate(owned):
                        moe.load_expert(
                            le, gu[j][:inter], gu[j][inter:], dn[j]
                        )
            else:
                layer.mlp.gate_up_proj.load_shards(
                    get(p + "mlp.gate_proj.weight"), get(p + "mlp.up_proj.weight")
                )
                layer.mlp.down_proj.load_shard(get(p + "mlp.down_proj.weight"))

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
