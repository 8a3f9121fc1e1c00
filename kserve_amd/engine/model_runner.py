"""Model runner: owns KV caches, prepares step inputs, executes forwards.

MI355X notes:
- KV layout [num_blocks, num_kv_heads, block_size, head_dim] keeps each
  (block, head) tile contiguous (4 KB bf16 at 16x128) for LDS staging in the
  decode kernel.
- Decode steps are hipGraph-captured per batch-size bucket (static buffers)
  to eliminate per-step launch overhead (SURVEY.md hard part #1).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch

from kserve_amd.engine.block_manager import BlockManager
from kserve_amd.engine.config import EngineConfig
from kserve_amd.engine.request import Request
from kserve_amd.engine.scheduler import ScheduledBatch
from kserve_amd.logging import logger
from kserve_amd.models.llama import AttentionMetadata, LlamaForCausalLM

_DEFAULT_GRAPH_BATCH_SIZES = [1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128, 192, 256, 384, 512, 768, 1024, 1280, 1536, 2048]


class ModelRunner:
    def __init__(
        self,
        config: EngineConfig,
        model: LlamaForCausalLM,
        device: str,
    ):
        self.config = config
        self.model = model
        self.device = torch.device(device)
        self.is_cuda = self.device.type == "cuda"
        self.kv_caches: List[Tuple[torch.Tensor, torch.Tensor]] = []
        self.num_gpu_blocks = 0
        self.max_blocks_per_seq = (
            config.scheduler.max_model_len + config.cache.block_size - 1
        ) // config.cache.block_size
        # hipGraph state
        self._graphs: Dict[int, torch.cuda.CUDAGraph] = {}
        # sampled-decode variants: same buckets, sampler inside the graph
        # (captured lazily on first temperature>0 window; shares the pool)
        self._sampled_graphs: Dict[int, torch.cuda.CUDAGraph] = {}
        self._graph_pool = None
        self._graph_buffers: Optional[Dict[str, torch.Tensor]] = None
        self._graph_batch_sizes: List[int] = []
        self._window_host: Optional[torch.Tensor] = None
        self._sample_pin: Optional[Dict[str, torch.Tensor]] = None
        # monotonically advancing RNG step for windowed sampling (seeds are
        # base + step*PRIME; the graph advances seeds in-replay, this mirrors
        # the count on the host between windows)
        self._rng_step = 0
        # pinned staging + per-request numpy block-table cache
        self._pin: Optional[Dict[str, torch.Tensor]] = None
        self._pin_np = None
        self._bt_cache: Dict[str, object] = {}
        # host-DRAM KV offload tier (pinned) + dedicated copy stream
        self.cpu_kv_caches: List[Tuple[torch.Tensor, torch.Tensor]] = []
        self.num_cpu_blocks = 0
        self._swap_stream = (
            torch.cuda.Stream() if self.is_cuda else None
        )
        # multi-LoRA (set by LLMEngine.register_lora)
        self.lora_manager = None

    # -- KV cache -----------------------------------------------------------
    def profile_and_allocate_kv(self) -> int:
        m = self.config.model
        tp = max(1, self._tp_size())
        dtype = self.model.dtype
        if self.is_cuda:
            torch.cuda.synchronize()
            free_bytes, _total = torch.cuda.mem_get_info(self.device)
            # leave headroom for activations/graphs
            budget = int(
                free_bytes * self.config.cache.gpu_memory_utilization
            ) - (2 << 30)
            kv_itemsize = self.config.cache.cache_torch_dtype(dtype).itemsize
            # kv_bytes_per_token_per_layer assumes bf16 (2 B); rescale
            per_block_bytes = (
                m.kv_bytes_per_token_per_layer
                * kv_itemsize
                * self.config.cache.block_size
                * self._num_local_layers()
                // (tp * 2)
            )
            num_blocks = self.config.cache.num_gpu_blocks or max(
                budget // per_block_bytes,
                16,
            )
        else:
            num_blocks = self.config.cache.num_gpu_blocks or 512
        self.allocate_kv_cache(num_blocks)
        return num_blocks

    def _num_local_layers(self) -> int:
        return getattr(self.model, "num_local_layers", self.config.model.num_layers)

    def allocate_kv_cache(self, num_blocks: int):
        m = self.config.model
        kv_heads_local = max(1, m.num_kv_heads // max(1, self._tp_size()))
        shape = (
            num_blocks,
            kv_heads_local,
            self.config.cache.block_size,
            m.head_dim,
        )
        dtype = self.config.cache.cache_torch_dtype(self.model.dtype)
        # empty, not zeros: context_lens bound all reads, and zeroing a
        # 250 GB pool costs seconds of startup
        alloc = torch.empty if self.is_cuda else torch.zeros
        self.kv_caches = [
            (
                alloc(shape, dtype=dtype, device=self.device),
                alloc(shape, dtype=dtype, device=self.device),
            )
            for _ in range(self._num_local_layers())
        ]
        self.num_gpu_blocks = num_blocks
        self._allocate_cpu_kv()
        kv_gib = (
            2 * num_blocks * kv_heads_local * self.config.cache.block_size
            * m.head_dim * self._num_local_layers() * dtype.itemsize
        ) / (1 << 30)
        if dtype.itemsize == 1:
            logger.info("KV cache dtype: fp8_e4m3 (scale 1.0)")
        logger.info(
            "KV cache: %d blocks x %d tokens (%.1f GiB, layout [blocks, kv_heads, block, head_dim])",
            num_blocks,
            self.config.cache.block_size,
            kv_gib,
        )

    def _allocate_cpu_kv(self):
        """Pinned host-DRAM KV tier (reference KVCacheOffloadingSpec
        cpu_bytes_to_use -> vLLM OffloadingConnector; ours is native:
        hipMemcpyAsync block copies on a side stream)."""
        m = self.config.model
        cache = self.config.cache
        kv_heads_local = max(1, m.num_kv_heads // max(1, self._tp_size()))
        kv_dtype = cache.cache_torch_dtype(self.model.dtype)
        per_block = (
            2 * kv_heads_local * cache.block_size * m.head_dim
            * self._num_local_layers() * kv_dtype.itemsize
        )
        n = cache.num_cpu_blocks
        if not n and cache.cpu_offload_bytes:
            n = int(cache.cpu_offload_bytes // per_block)
        if not n:
            return
        shape = (n, kv_heads_local, cache.block_size, m.head_dim)
        pin = self.is_cuda
        self.cpu_kv_caches = [
            (
                torch.empty(shape, dtype=kv_dtype, pin_memory=pin),
                torch.empty(shape, dtype=kv_dtype, pin_memory=pin),
            )
            for _ in range(self._num_local_layers())
        ]
        self.num_cpu_blocks = n
        logger.info(
            "KV host-offload tier: %d pinned blocks (%.2f GiB)",
            n,
            n * per_block / (1 << 30),
        )

    def swap_blocks(self, pairs, to_gpu: bool):
        """Copy KV pages between tiers on the side stream; the compute
        stream waits on the copies before the next kernel touches them."""
        if not pairs:
            return
        def do_copies():
            for layer, (gk, gv) in enumerate(self.kv_caches):
                ck, cv = self.cpu_kv_caches[layer]
                for a, b in pairs:
                    if to_gpu:  # (cpu, gpu)
                        gk[b].copy_(ck[a], non_blocking=True)
                        gv[b].copy_(cv[a], non_blocking=True)
                    else:       # (gpu, cpu)
                        ck[b].copy_(gk[a], non_blocking=True)
                        cv[b].copy_(gv[a], non_blocking=True)
        if self.is_cuda:
            ev_before = torch.cuda.Event()
            ev_before.record(torch.cuda.current_stream())
            with torch.cuda.stream(self._swap_stream):
                # copies must not race kernels still using the pages
                self._swap_stream.wait_event(ev_before)
                do_copies()
                ev = torch.cuda.Event()
                ev.record(self._swap_stream)
            torch.cuda.current_stream().wait_event(ev)
        else:
            do_copies()

    def _tp_size(self) -> int:
        from kserve_amd.parallel import comm

        return comm.get_state().tp_size

    # -- input prep -----------------------------------------------------------
    def prepare_prefill(
        self, batch: ScheduledBatch, block_manager: BlockManager
    ):
        import numpy as np

        total = sum(batch.num_scheduled_tokens)
        tokens = np.empty(total, dtype=np.int64)
        positions = np.empty(total, dtype=np.int64)
        slots = np.empty(total, dtype=np.int32)
        cu = [0]
        max_seqlen = 0
        last_token_idx: List[int] = []
        off = 0
        bs = self.config.cache.block_size
        for req, n in zip(batch.requests, batch.num_scheduled_tokens):
            start = req.num_computed_tokens
            end = start + n
            tokens[off : off + n] = req.all_token_ids[start:end]
            positions[off : off + n] = np.arange(start, end, dtype=np.int64)
            # vectorized slot mapping from the block table
            bt = np.asarray(
                block_manager.get_block_table(req), dtype=np.int32
            )
            pos_range = np.arange(start, end, dtype=np.int64)
            slots[off : off + n] = bt[pos_range // bs] * bs + (
                pos_range % bs
            ).astype(np.int32)
            off += n
            cu.append(cu[-1] + n)
            max_seqlen = max(max_seqlen, n)
            last_token_idx.append(cu[-1] - 1)
        dev = self.device
        meta = AttentionMetadata(
            is_prefill=True,
            slot_mapping=torch.from_numpy(slots).to(dev, non_blocking=True),
            cu_seqlens=torch.tensor(cu, dtype=torch.int32, device=dev),
            max_seqlen=max_seqlen,
        )
        # chunked-prefill resume: any request with context already in pages
        # must attend to it — switch to the paged-context attention path
        if any(r.num_computed_tokens > 0 for r in batch.requests) and (
            self.kv_caches and self.kv_caches[0][0].numel() > 0
        ):
            nb_per = [
                -(-(r.num_computed_tokens + n) // bs)
                for r, n in zip(batch.requests, batch.num_scheduled_tokens)
            ]
            max_nb = max(nb_per)
            btab = np.zeros((len(batch.requests), max_nb), dtype=np.int32)
            ctx = np.empty(len(batch.requests), dtype=np.int32)
            for i, (req, n) in enumerate(
                zip(batch.requests, batch.num_scheduled_tokens)
            ):
                t = block_manager.get_block_table(req)
                btab[i, : nb_per[i]] = t[: nb_per[i]]
                ctx[i] = req.num_computed_tokens + n
            meta.block_tables = torch.from_numpy(btab).to(dev, non_blocking=True)
            meta.context_lens = torch.from_numpy(ctx).to(dev, non_blocking=True)
        input_ids = torch.from_numpy(tokens).to(dev, non_blocking=True)
        pos = torch.from_numpy(positions).to(dev, non_blocking=True)
        sel = torch.tensor(last_token_idx, dtype=torch.int64, device=dev)
        return input_ids, pos, meta, sel

    def _ensure_decode_buffers(self, n: int, max_blocks: int):
        """Pinned host staging buffers + numpy views (fast CPU fill, one
        async H2D per tensor per step)."""
        import numpy as np

        need = self._pin is None or self._pin["input_ids"].shape[0] < n or self._pin[
            "block_tables"
        ].shape[1] < max_blocks
        if need:
            cap = max(n, self.config.scheduler.max_num_seqs)
            mb = max(max_blocks, self.max_blocks_per_seq)
            pin = self.is_cuda
            self._pin = {
                "input_ids": torch.empty(cap, dtype=torch.int64, pin_memory=pin),
                "positions": torch.empty(cap, dtype=torch.int64, pin_memory=pin),
                "slot_mapping": torch.empty(cap, dtype=torch.int32, pin_memory=pin),
                "context_lens": torch.empty(cap, dtype=torch.int32, pin_memory=pin),
                "block_tables": torch.zeros(
                    (cap, mb), dtype=torch.int32, pin_memory=pin
                ),
            }
            self._pin_np = {k: v.numpy() for k, v in self._pin.items()}
        return self._pin, self._pin_np

    def _fill_pinned(self, batch: ScheduledBatch, block_manager: BlockManager):
        """Fill the pinned staging buffers for a decode batch; returns
        (n, nb) = batch size and max block-table width used."""
        import numpy as np

        n = len(batch.requests)
        bs = self.config.cache.block_size
        max_blocks = 1
        tables = []
        for req in batch.requests:
            bt = block_manager.get_block_table(req)
            tables.append(bt)
            if len(bt) > max_blocks:
                max_blocks = len(bt)
        pin, pnp = self._ensure_decode_buffers(n, max_blocks)
        ii = pnp["input_ids"]
        pp = pnp["positions"]
        sm = pnp["slot_mapping"]
        ctx = pnp["context_lens"]
        btab = pnp["block_tables"]
        for i, req in enumerate(batch.requests):
            pos = req.num_computed_tokens
            ii[i] = req.all_token_ids[pos]
            pp[i] = pos
            t = tables[i]
            sm[i] = t[pos // bs] * bs + pos % bs
            ctx[i] = pos + 1
            # cached numpy row per request; refreshed on growth AND on table
            # identity change (swap_in / recompute preemption reassign block
            # ids at the same length — block_manager.table_seq catches those)
            seq = block_manager.table_seq(req.request_id)
            cache = self._bt_cache.get(req.request_id)
            if cache is None or cache[0] != seq or cache[1].shape[0] != len(t):
                cache = (seq, np.asarray(t, dtype=np.int32))
                self._bt_cache[req.request_id] = cache
            btab[i, : cache[1].shape[0]] = cache[1]
        return n, max_blocks

    def release_request(self, request_id: str):
        self._bt_cache.pop(request_id, None)

    def prepare_decode(
        self, batch: ScheduledBatch, block_manager: BlockManager
    ):
        n, nb = self._fill_pinned(batch, block_manager)
        pin = self._pin
        dev = self.device
        meta = AttentionMetadata(
            is_prefill=False,
            slot_mapping=pin["slot_mapping"][:n].to(dev, non_blocking=True),
            block_tables=pin["block_tables"][:n, :nb].to(dev, non_blocking=True)
            .contiguous(),
            context_lens=pin["context_lens"][:n].to(dev, non_blocking=True),
        )
        input_ids = pin["input_ids"][:n].to(dev, non_blocking=True)
        pos = pin["positions"][:n].to(dev, non_blocking=True)
        return input_ids, pos, meta

    def _lora_meta(self, batch: ScheduledBatch, per_request_rows: bool):
        """Contiguous (adapter_id, start_row, end_row) segments for the batch.
        Prefill rows span each request's scheduled tokens; decode rows are one
        per request."""
        from kserve_amd.engine.lora import LoRABatchMeta

        segments = []
        row = 0
        cur_id, cur_start = None, 0
        widths = (
            [1] * len(batch.requests)
            if per_request_rows
            else batch.num_scheduled_tokens
        )
        for req, w in zip(batch.requests, widths):
            if req.lora_id != cur_id:
                if cur_id:
                    segments.append((cur_id, cur_start, row))
                cur_id, cur_start = req.lora_id, row
            row += w
        if cur_id:
            segments.append((cur_id, cur_start, row))
        return LoRABatchMeta(segments, self.lora_manager)

    # -- execution ----------------------------------------------------------
    @torch.no_grad()
    def execute_prefill(self, batch, block_manager) -> torch.Tensor:
        input_ids, positions, meta, sel = self.prepare_prefill(batch, block_manager)
        if self.lora_manager is not None and any(
            r.lora_id for r in batch.requests
        ):
            meta.lora = self._lora_meta(batch, per_request_rows=False)
        hidden = self.model(input_ids, positions, self.kv_caches, meta)
        return self.model.compute_logits(hidden[sel])

    @torch.no_grad()
    def execute_decode(self, batch, block_manager) -> torch.Tensor:
        n = len(batch.requests)
        has_lora = any(r.lora_id for r in batch.requests)
        if self._graphs and not self.config.enforce_eager and not has_lora:
            bucket = self._graph_bucket(n)
            if bucket is not None:
                return self._run_graph(bucket, batch, block_manager)
        input_ids, positions, meta = self.prepare_decode(batch, block_manager)
        if has_lora:
            meta.lora = self._lora_meta(batch, per_request_rows=True)
        hidden = self.model(input_ids, positions, self.kv_caches, meta)
        return self.model.compute_logits(hidden)

    @torch.no_grad()
    def execute_verify(self, batch, drafts, block_manager):
        """Speculative-decode verify forward: per request feed
        [last_token, draft_1..draft_k] as a chunk attending to the paged
        context (same math as chunked prefill), returning logits for EVERY
        row plus the per-request row offsets (cu)."""
        import numpy as np

        widths = [1 + len(d) for d in drafts]
        total = sum(widths)
        bs = self.config.cache.block_size
        tokens = np.empty(total, dtype=np.int64)
        positions = np.empty(total, dtype=np.int64)
        slots = np.empty(total, dtype=np.int32)
        cu = [0]
        nb_per = []
        for req, d, w in zip(batch.requests, drafts, widths):
            pos0 = req.num_computed_tokens
            nb_per.append(-(-(pos0 + w) // bs))
        max_nb = max(nb_per)
        btab = np.zeros((len(batch.requests), max_nb), dtype=np.int32)
        ctx = np.empty(len(batch.requests), dtype=np.int32)
        off = 0
        for i, (req, d, w) in enumerate(zip(batch.requests, drafts, widths)):
            pos0 = req.num_computed_tokens
            tokens[off] = req.all_token_ids[pos0]
            tokens[off + 1 : off + w] = d
            pr = np.arange(pos0, pos0 + w, dtype=np.int64)
            positions[off : off + w] = pr
            t = block_manager.get_block_table(req)
            bt = np.asarray(t, dtype=np.int32)
            slots[off : off + w] = bt[pr // bs] * bs + (pr % bs).astype(np.int32)
            btab[i, : nb_per[i]] = bt[: nb_per[i]]
            ctx[i] = pos0 + w
            off += w
            cu.append(off)
        dev = self.device
        from kserve_amd.models.llama import AttentionMetadata

        meta = AttentionMetadata(
            is_prefill=True,
            slot_mapping=torch.from_numpy(slots).to(dev, non_blocking=True),
            cu_seqlens=torch.tensor(cu, dtype=torch.int32, device=dev),
            max_seqlen=max(widths),
            block_tables=torch.from_numpy(btab).to(dev, non_blocking=True),
            context_lens=torch.from_numpy(ctx).to(dev, non_blocking=True),
        )
        input_ids = torch.from_numpy(tokens).to(dev, non_blocking=True)
        pos = torch.from_numpy(positions).to(dev, non_blocking=True)
        hidden = self.model(input_ids, pos, self.kv_caches, meta)
        return self.model.compute_logits(hidden), cu

    # -- hipGraph capture ------------------------------------------------------
    def capture_decode_graphs(self, batch_sizes: Optional[List[int]] = None):
        """Capture the decode forward into hipGraphs per batch-size bucket.

        Static input buffers are sized for the largest bucket; smaller
        batches replay the nearest bucket >= n with padded slots pointing at
        a scratch block (block 0 is reserved as scratch by convention when
        graphs are on — the allocator hands out blocks from the top).
        """
        if not self.is_cuda:
            return
        sizes = [
            b
            for b in (batch_sizes or _DEFAULT_GRAPH_BATCH_SIZES)
            if b <= self.config.scheduler.max_num_seqs
        ]
        max_bs = max(sizes)
        dev = self.device
        mb = self.max_blocks_per_seq
        self._graph_buffers = {
            "input_ids": torch.zeros(max_bs, dtype=torch.int64, device=dev),
            "positions": torch.zeros(max_bs, dtype=torch.int64, device=dev),
            "block_tables": torch.zeros(
                (max_bs, mb), dtype=torch.int32, device=dev
            ),
            "context_lens": torch.ones(max_bs, dtype=torch.int32, device=dev),
            # per-request sampling params (sampled-window variants)
            "temps": torch.ones(max_bs, dtype=torch.float32, device=dev),
            "top_p": torch.ones(max_bs, dtype=torch.float32, device=dev),
            "top_k": torch.zeros(max_bs, dtype=torch.int32, device=dev),
            "seeds": torch.zeros(max_bs, dtype=torch.int64, device=dev),
            "greedy_mask": torch.zeros(max_bs, dtype=torch.bool, device=dev),
            "logits": {},
            "sampled": {},
            "sampled_rand": {},
        }
        torch.cuda.synchronize()
        pool = None
        from kserve_amd import ops as _ops

        bsz = self.config.cache.block_size
        for bs in sorted(sizes, reverse=True):
            buf = self._graph_buffers

            def body(bs=bs, buf=buf):
                # slot mapping derived IN-GRAPH from (block_tables, positions)
                # so multi-step windows advance with zero host involvement
                pos = buf["positions"][:bs]
                bt = buf["block_tables"][:bs]
                blk = torch.gather(bt, 1, (pos // bsz).unsqueeze(1)).squeeze(1)
                slot = (blk * bsz + (pos % bsz).to(torch.int32)).to(torch.int32)
                meta = AttentionMetadata(
                    is_prefill=False,
                    slot_mapping=slot,
                    block_tables=bt,
                    context_lens=buf["context_lens"][:bs],
                )
                hidden = self.model(
                    buf["input_ids"][:bs], pos, self.kv_caches, meta
                )
                logits = self.model.compute_logits(hidden)
                # greedy sample in-graph; feed back as the next step's input
                sampled = _ops.greedy_sample(logits)
                buf["input_ids"][:bs].copy_(sampled)
                buf["positions"][:bs].add_(1)
                buf["context_lens"][:bs].add_(1)
                return logits, sampled

            # warmup run (allocator state, rccl comms)
            body()
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g, pool=pool):
                logits, sampled = body()
            if pool is None:
                pool = g.pool()
            self._graphs[bs] = g
            buf["logits"][bs] = logits
            buf["sampled"][bs] = sampled
        self._graph_pool = pool
        self._graph_batch_sizes = sorted(self._graphs.keys())
        torch.cuda.synchronize()
        logger.info("Captured decode hipGraphs for batch sizes %s", self._graph_batch_sizes)

    _SEED_PRIME = 0x9E3779B1

    def _ensure_sampled_graph(self, bs: int) -> torch.cuda.CUDAGraph:
        """Capture (once, lazily) the sampled-decode variant for a bucket:
        the fused temperature/top-k/top-p Gumbel sampler runs IN-GRAPH, the
        seed buffer advances in-replay, and greedy rows are overridden by
        argmax — so multi-step windows work for temperature>0 batches
        (round-1 limitation: windows were greedy-only)."""
        g = self._sampled_graphs.get(bs)
        if g is not None:
            return g
        from kserve_amd import ops as _ops

        buf = self._graph_buffers
        bsz = self.config.cache.block_size

        def body():
            pos = buf["positions"][:bs]
            bt = buf["block_tables"][:bs]
            blk = torch.gather(bt, 1, (pos // bsz).unsqueeze(1)).squeeze(1)
            slot = (blk * bsz + (pos % bsz).to(torch.int32)).to(torch.int32)
            meta = AttentionMetadata(
                is_prefill=False,
                slot_mapping=slot,
                block_tables=bt,
                context_lens=buf["context_lens"][:bs],
            )
            hidden = self.model(
                buf["input_ids"][:bs], pos, self.kv_caches, meta
            )
            logits = self.model.compute_logits(hidden)
            greedy = _ops.greedy_sample(logits)
            rand = torch.empty(bs, dtype=torch.int64, device=logits.device)
            _ops.topk_topp_sample_into(
                rand, logits, buf["temps"][:bs], buf["top_p"][:bs],
                buf["top_k"][:bs], buf["seeds"][:bs],
            )
            sampled = torch.where(buf["greedy_mask"][:bs], greedy, rand)
            buf["input_ids"][:bs].copy_(sampled)
            buf["positions"][:bs].add_(1)
            buf["context_lens"][:bs].add_(1)
            # fresh Gumbel noise next replay, zero host involvement
            buf["seeds"][:bs].add_(self._SEED_PRIME)
            return sampled

        body()
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g, pool=self._graph_pool):
            sampled = body()
        self._sampled_graphs[bs] = g
        buf["sampled_rand"][bs] = sampled
        logger.info("Captured sampled-decode hipGraph for batch size %d", bs)
        return g

    def _graph_bucket(self, n: int) -> Optional[int]:
        for b in self._graph_batch_sizes:
            if b >= n:
                return b
        return None

    def _run_graph(self, bucket: int, batch, block_manager) -> torch.Tensor:
        n, nb = self._fill_pinned(batch, block_manager)
        pin = self._pin
        pnp = self._pin_np
        if n < bucket:
            # padded rows: write scratch KV to block 0 (reserved), ctx len 1
            pnp["input_ids"][n:bucket] = 0
            pnp["positions"][n:bucket] = 0
            pnp["context_lens"][n:bucket] = 1
            pnp["block_tables"][n:bucket, :nb] = 0
        buf = self._graph_buffers
        # single async H2D per tensor, pinned -> static graph buffers
        # (slot mapping is derived in-graph from block_tables + positions)
        buf["input_ids"][:bucket].copy_(pin["input_ids"][:bucket], non_blocking=True)
        buf["positions"][:bucket].copy_(pin["positions"][:bucket], non_blocking=True)
        buf["context_lens"][:bucket].copy_(
            pin["context_lens"][:bucket], non_blocking=True
        )
        buf["block_tables"][:bucket, :nb].copy_(
            pin["block_tables"][:bucket, :nb], non_blocking=True
        )
        self._graphs[bucket].replay()
        return buf["logits"][bucket][:n]

    def _stage_graph_inputs(self, bucket: int, batch, block_manager) -> int:
        n, nb = self._fill_pinned(batch, block_manager)
        pin = self._pin
        pnp = self._pin_np
        if n < bucket:
            pnp["input_ids"][n:bucket] = 0
            pnp["positions"][n:bucket] = 0
            pnp["context_lens"][n:bucket] = 1
            pnp["block_tables"][n:bucket, :nb] = 0
        buf = self._graph_buffers
        buf["input_ids"][:bucket].copy_(pin["input_ids"][:bucket], non_blocking=True)
        buf["positions"][:bucket].copy_(pin["positions"][:bucket], non_blocking=True)
        buf["context_lens"][:bucket].copy_(
            pin["context_lens"][:bucket], non_blocking=True
        )
        buf["block_tables"][:bucket, :nb].copy_(
            pin["block_tables"][:bucket, :nb], non_blocking=True
        )
        return n

    def _stage_sampling_params(self, bucket: int, requests) -> None:
        """Stage per-request sampling params into the static graph buffers
        (pinned → device, one async copy per tensor)."""
        import numpy as np

        if self._sample_pin is None or self._sample_pin["temps"].shape[0] < bucket:
            max_bs = self._graph_buffers["temps"].shape[0]
            self._sample_pin = {
                "temps": torch.ones(max_bs, dtype=torch.float32, pin_memory=True),
                "top_p": torch.ones(max_bs, dtype=torch.float32, pin_memory=True),
                "top_k": torch.zeros(max_bs, dtype=torch.int32, pin_memory=True),
                "seeds": torch.zeros(max_bs, dtype=torch.int64, pin_memory=True),
                "greedy_mask": torch.zeros(max_bs, dtype=torch.bool, pin_memory=True),
            }
        pin = self._sample_pin
        tn = pin["temps"].numpy()
        pn = pin["top_p"].numpy()
        kn = pin["top_k"].numpy()
        sn = pin["seeds"].numpy()
        gn = pin["greedy_mask"].numpy()
        step_off = self._rng_step * self._SEED_PRIME
        for i, r in enumerate(requests):
            sp = r.sampling_params
            g = sp.greedy
            gn[i] = g
            tn[i] = max(sp.temperature, 1e-5)
            pn[i] = sp.top_p
            kn[i] = sp.top_k
            base = (
                sp.seed
                if sp.seed is not None
                else hash(r.request_id) & 0x7FFFFFFF
            )
            sn[i] = (base + step_off) & 0x7FFFFFFFFFFFFFFF
        n = len(requests)
        if n < bucket:
            # padded rows: greedy (cheap, deterministic)
            gn[n:bucket] = True
            tn[n:bucket] = 1.0
            pn[n:bucket] = 1.0
            kn[n:bucket] = 0
            sn[n:bucket] = 1
        buf = self._graph_buffers
        for key in ("temps", "top_p", "top_k", "seeds", "greedy_mask"):
            buf[key][:bucket].copy_(pin[key][:bucket], non_blocking=True)

    @torch.no_grad()
    def multi_step_decode(
        self, batch, block_manager, k_steps: int, sampled: bool = False
    ) -> torch.Tensor:
        """Run ``k_steps`` decode iterations with the sampled token fed back
        on-GPU (graph path: pure replays, zero host work in the loop).
        ``sampled=True`` uses the in-graph temperature/top-k/top-p variant;
        default is the greedy graph. Caller must have reserved KV capacity
        for the whole window. Returns host int64 [k_steps, n]."""
        n = len(batch.requests)
        bucket = None
        if self.is_cuda and self._graphs and not self.config.enforce_eager:
            bucket = self._graph_bucket(n)
        if bucket is not None:
            if sampled:
                g = self._ensure_sampled_graph(bucket)
                self._stage_sampling_params(bucket, batch.requests)
                out_dev = self._graph_buffers["sampled_rand"][bucket]
            else:
                g = self._graphs[bucket]
                out_dev = self._graph_buffers["sampled"][bucket]
            self._stage_graph_inputs(bucket, batch, block_manager)
            if (
                self._window_host is None
                or self._window_host.shape[0] < k_steps
                or self._window_host.shape[1] < n
            ):
                self._window_host = torch.empty(
                    (max(k_steps, 16), self.config.scheduler.max_num_seqs),
                    dtype=torch.int64,
                    pin_memory=True,
                )
            out_host = self._window_host[:k_steps, :n]
            for k in range(k_steps):
                g.replay()
                out_host[k].copy_(out_dev[:n], non_blocking=True)
            torch.cuda.synchronize()
            if sampled:
                self._rng_step += k_steps
            return out_host
        # eager fallback (CPU tests / enforce_eager): same semantics
        from kserve_amd import ops as _ops

        input_ids, positions, meta = self.prepare_decode(batch, block_manager)
        bsz = self.config.cache.block_size
        bt = meta.block_tables
        ctx = meta.context_lens
        out = torch.empty((k_steps, n), dtype=torch.int64)
        dev = input_ids.device
        if sampled:
            reqs = batch.requests
            temps = torch.tensor(
                [max(r.sampling_params.temperature, 1e-5) for r in reqs],
                dtype=torch.float32, device=dev,
            )
            top_p = torch.tensor(
                [r.sampling_params.top_p for r in reqs],
                dtype=torch.float32, device=dev,
            )
            top_k = torch.tensor(
                [r.sampling_params.top_k for r in reqs],
                dtype=torch.int32, device=dev,
            )
            gmask = torch.tensor(
                [r.sampling_params.greedy for r in reqs],
                dtype=torch.bool, device=dev,
            )
            gen = torch.Generator()
            gen.manual_seed(self.config.seed + self._rng_step)
            seed_base = torch.tensor(
                [
                    (r.sampling_params.seed
                     if r.sampling_params.seed is not None
                     else hash(r.request_id) & 0x7FFFFFFF)
                    for r in reqs
                ],
                dtype=torch.int64, device=dev,
            )
        for k in range(k_steps):
            blk = torch.gather(bt, 1, (positions // bsz).unsqueeze(1)).squeeze(1)
            slot = (blk * bsz + (positions % bsz).to(torch.int32)).to(torch.int32)
            meta_k = AttentionMetadata(
                is_prefill=False,
                slot_mapping=slot,
                block_tables=bt,
                context_lens=ctx,
            )
            hidden = self.model(input_ids, positions, self.kv_caches, meta_k)
            logits = self.model.compute_logits(hidden)
            if sampled:
                seeds = seed_base + (self._rng_step + k) * self._SEED_PRIME
                rand = _ops.random_sample(
                    logits, temps, top_p, top_k, seeds=seeds, generator=gen
                )
                greedy = _ops.greedy_sample(logits)
                tok = torch.where(gmask, greedy, rand)
            else:
                tok = _ops.greedy_sample(logits)
            out[k] = tok.cpu()
            input_ids = tok.to(dev)
            positions = positions + 1
            ctx = ctx + 1
        if sampled:
            self._rng_step += k_steps
        return out
