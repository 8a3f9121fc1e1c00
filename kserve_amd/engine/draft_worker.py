"""Draft-model speculative decoding: a small model proposes, the main
model verifies.

The prompt-lookup drafter (engine._draft_ngram) only accepts on repeated
text; a draft MODEL accepts on any predictable continuation — the vLLM
"draft model" speculation mode, MI355X-native here: the draft runs eagerly
(it is small; launch overhead is hidden under the main model's verify
forward budget), keeps its own paged KV, and each round

  1. CATCHES UP its KV on the tokens the main model accepted since the
     last round (a chunked paged-context forward — rejected draft
     positions are simply overwritten, so no rollback bookkeeping), then
  2. PROPOSES k tokens autoregressively (greedy argmax).

Outputs remain exactly the main model's greedy stream by construction
(the verify pass emits the main model's token at every position); the
draft only changes how many positions verify per forward.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch

from kserve_amd.engine.block_manager import BlockManager
from kserve_amd.engine.config import EngineConfig, ModelConfig
from kserve_amd.engine.request import Request
from kserve_amd.models.llama import AttentionMetadata, LlamaForCausalLM


class _Shadow:
    """BlockManager keys tables by request_id; the draft tracks its own
    tables with lightweight shadows of the live requests."""

    __slots__ = ("request_id", "num_tokens", "block_table",
                 "num_cached_tokens")

    def __init__(self, request_id: str):
        self.request_id = request_id
        self.num_tokens = 0
        self.block_table: List[int] = []
        self.num_cached_tokens = 0


class DraftModelWorker:
    def __init__(
        self,
        config: EngineConfig,
        draft_config: ModelConfig,
        num_blocks: int = 0,
    ):
        self.config = config
        self.device = torch.device(config.device)
        dtype = torch.bfloat16 if self.device.type == "cuda" else torch.float32
        self.model = LlamaForCausalLM(
            draft_config, dtype=dtype, device=config.device
        ).eval()
        self.model.random_init(seed=config.seed + 1)
        bs = config.cache.block_size
        blocks = num_blocks or max(
            64,
            (config.scheduler.max_model_len // bs + 2)
            * config.scheduler.max_num_seqs
            // 8,
        )
        self.block_manager = BlockManager(blocks, bs)
        kv_heads = draft_config.num_kv_heads
        self.kv_caches = [
            (
                torch.zeros(blocks, kv_heads, bs, draft_config.head_dim,
                            dtype=dtype, device=self.device),
                torch.zeros(blocks, kv_heads, bs, draft_config.head_dim,
                            dtype=dtype, device=self.device),
            )
            for _ in range(draft_config.num_layers)
        ]
        self._shadows: Dict[str, _Shadow] = {}
        self._computed: Dict[str, int] = {}

    def load_hf_state_dict(self, tensors) -> None:
        self.model.load_hf_state_dict(tensors)

    # -- lifecycle ---------------------------------------------------------
    def release(self, request_id: str) -> None:
        sh = self._shadows.pop(request_id, None)
        if sh is not None:
            self.block_manager.free(sh)
        self._computed.pop(request_id, None)

    def _shadow(self, req: Request) -> _Shadow:
        sh = self._shadows.get(req.request_id)
        if sh is None:
            sh = _Shadow(req.request_id)
            self._shadows[req.request_id] = sh
            self._computed[req.request_id] = 0
        return sh

    def _ensure_capacity(self, sh: _Shadow, num_tokens: int) -> bool:
        bs = self.block_manager.block_size
        sh.num_tokens = num_tokens
        table = self.block_manager._tables.get(sh.request_id)
        if table is None:
            if not self.block_manager.can_allocate(sh, num_tokens):
                return False
            self.block_manager.allocate(sh, num_tokens)
            return True
        need = self.block_manager.blocks_needed(num_tokens) - len(table)
        if need > 0:
            if need > self.block_manager.num_free_blocks:
                return False
            table.extend(self.block_manager.take_blocks(need))
        return True

    # -- forward helpers ---------------------------------------------------
    @torch.no_grad()
    def _forward_chunk(self, reqs, chunks) -> torch.Tensor:
        """Varlen paged-context forward over per-request token chunks
        (the catch-up pass). Returns last-row logits per request."""
        import numpy as np

        bs = self.block_manager.block_size
        widths = [len(c) for c in chunks]
        total = sum(widths)
        tokens = np.empty(total, dtype=np.int64)
        positions = np.empty(total, dtype=np.int64)
        slots = np.empty(total, dtype=np.int32)
        cu = [0]
        ctx = np.empty(len(reqs), dtype=np.int32)
        nb_per = []
        off = 0
        for i, (req, chunk) in enumerate(zip(reqs, chunks)):
            sh = self._shadows[req.request_id]
            pos0 = self._computed[req.request_id]
            w = len(chunk)
            tokens[off : off + w] = chunk
            pr = np.arange(pos0, pos0 + w, dtype=np.int64)
            positions[off : off + w] = pr
            bt = np.asarray(
                self.block_manager.get_block_table(sh), dtype=np.int32
            )
            slots[off : off + w] = bt[pr // bs] * bs + (pr % bs).astype(
                np.int32
            )
            ctx[i] = pos0 + w
            nb_per.append(-(-(pos0 + w) // bs))
            off += w
            cu.append(off)
        max_nb = max(nb_per)
        btab = np.zeros((len(reqs), max_nb), dtype=np.int32)
        for i, req in enumerate(reqs):
            sh = self._shadows[req.request_id]
            bt = self.block_manager.get_block_table(sh)
            btab[i, : nb_per[i]] = bt[: nb_per[i]]
        dev = self.device
        meta = AttentionMetadata(
            is_prefill=True,
            slot_mapping=torch.from_numpy(slots).to(dev),
            cu_seqlens=torch.tensor(cu, dtype=torch.int32, device=dev),
            max_seqlen=max(widths),
            block_tables=torch.from_numpy(btab).to(dev),
            context_lens=torch.from_numpy(ctx).to(dev),
        )
        hidden = self.model(
            torch.from_numpy(tokens).to(dev),
            torch.from_numpy(positions).to(dev),
            self.kv_caches,
            meta,
        )
        logits = self.model.compute_logits(hidden)
        last_rows = torch.tensor(
            [cu[i + 1] - 1 for i in range(len(reqs))], device=logits.device
        )
        return logits.index_select(0, last_rows)

    @torch.no_grad()
    def _forward_decode(self, reqs, input_tokens: List[int]) -> torch.Tensor:
        import numpy as np

        bs = self.block_manager.block_size
        n = len(reqs)
        positions = np.empty(n, dtype=np.int64)
        slots = np.empty(n, dtype=np.int32)
        ctx = np.empty(n, dtype=np.int32)
        nb = []
        for i, req in enumerate(reqs):
            sh = self._shadows[req.request_id]
            pos = self._computed[req.request_id]
            positions[i] = pos
            bt = self.block_manager.get_block_table(sh)
            slots[i] = bt[pos // bs] * bs + pos % bs
            ctx[i] = pos + 1
            nb.append(len(bt))
        max_nb = max(nb)
        btab = np.zeros((n, max_nb), dtype=np.int32)
        for i, req in enumerate(reqs):
            bt = self.block_manager.get_block_table(
                self._shadows[req.request_id]
            )
            btab[i, : len(bt)] = bt
        dev = self.device
        meta = AttentionMetadata(
            is_prefill=False,
            slot_mapping=torch.from_numpy(slots).to(dev),
            block_tables=torch.from_numpy(btab).to(dev),
            context_lens=torch.from_numpy(ctx).to(dev),
        )
        hidden = self.model(
            torch.tensor(input_tokens, dtype=torch.int64, device=dev),
            torch.from_numpy(positions).to(dev),
            self.kv_caches,
            meta,
        )
        return self.model.compute_logits(hidden)

    # -- the round ---------------------------------------------------------
    @torch.no_grad()
    def propose(self, requests: List[Request], k: int) -> List[List[int]]:
        """Draft up to k tokens per request. Returns one draft list per
        request ([] when the draft cannot participate, e.g. KV full)."""
        if k <= 0:
            return [[] for _ in requests]
        live = []
        for req in requests:
            sh = self._shadow(req)
            # capacity: accepted context + k drafts
            if not self._ensure_capacity(sh, req.num_computed_tokens + 1 + k):
                continue
            live.append(req)
        if not live:
            return [[] for _ in requests]

        # 1. catch-up: tokens [computed, num_computed_tokens) — the prompt
        #    on round one, the accepted tokens afterwards. Leave the LAST
        #    accepted token for the decode loop's first input.
        chunks = []
        for req in live:
            lo = self._computed[req.request_id]
            hi = req.num_computed_tokens  # last accepted position + ...
            chunks.append(list(req.all_token_ids[lo:hi]))
        catch = [i for i, c in enumerate(chunks) if len(c) > 0]
        if catch:
            reqs_c = [live[i] for i in catch]
            self._forward_chunk(reqs_c, [chunks[i] for i in catch])
            for i in catch:
                self._computed[live[i].request_id] = live[i].num_computed_tokens

        # 2. autoregressive proposal: feed the current token (position
        #    num_computed_tokens), take argmax, repeat
        drafts: Dict[str, List[int]] = {r.request_id: [] for r in live}
        inputs = [r.all_token_ids[r.num_computed_tokens] for r in live]
        for _ in range(k):
            logits = self._forward_decode(live, inputs)
            nxt = logits.argmax(dim=-1).tolist()
            for r, t in zip(live, nxt):
                drafts[r.request_id].append(int(t))
                self._computed[r.request_id] += 1
            inputs = nxt
        # the decode loop advanced _computed past the accepted boundary by
        # writing draft KV; roll the counter back so the next round's
        # catch-up recomputes from the true accepted position (stale draft
        # KV beyond it is overwritten then)
        for r in live:
            self._computed[r.request_id] = r.num_computed_tokens
        return [drafts.get(r.request_id, []) for r in requests]
