"""Paged KV-cache block allocator.

Fresh design for 288 GB HBM3E: a flat free-list of fixed-size blocks
(block_size tokens each) with per-request block tables. Supports
allocate-on-prefill, append-on-decode, free, and full preemption
(recompute-style: blocks are released and the request re-prefills later).

Replaces the paged-KV machinery the reference delegates to vLLM
(SURVEY.md §2.7 "Paged attention" row).
"""

from __future__ import annotations

import hashlib

from collections import OrderedDict
from typing import Dict, List, Optional, Tuple

from kserve_amd.engine.request import Request


def cdiv(a: int, b: int) -> int:
    return -(-a // b)


class BlockManager:
    def __init__(
        self,
        num_blocks: int,
        block_size: int,
        watermark: float = 0.01,
        num_cpu_blocks: int = 0,
        enable_prefix_caching: bool = False,
    ):
        self.num_blocks = num_blocks
        self.block_size = block_size
        # watermark: headroom kept free so decodes can always append
        self.watermark_blocks = max(1, int(num_blocks * watermark))
        # block 0 is reserved as the scratch target for hipGraph padding
        # (padded decode lanes write their KV there); never handed out.
        self._free: List[int] = list(range(1, num_blocks))
        self._tables: Dict[str, List[int]] = {}
        # monotonic per-request table identity: a new value every time the
        # request's block ids are (re)assigned (allocate / swap_in). The
        # runner's pinned block-table cache compares this, so a preempted or
        # swapped request with a same-LENGTH but different-CONTENT table is
        # never served stale block ids.
        self._seq = 0
        self._table_seq: Dict[str, int] = {}
        # host-DRAM offload tier (pinned; hipMemcpyAsync side stream)
        self.num_cpu_blocks = num_cpu_blocks
        self._cpu_free: List[int] = list(range(num_cpu_blocks))
        self._cpu_tables: Dict[str, List[int]] = {}
        # -- automatic prefix caching (content-addressed full prompt blocks,
        # reference: vLLM APC; ours reuses the paged-context prefill kernel
        # for the uncached suffix) --
        self.enable_prefix_caching = enable_prefix_caching
        self._refcount: Dict[int, int] = {}
        self._hash_to_block: Dict[bytes, int] = {}  # READY (computed) blocks
        self._block_to_hash: Dict[int, bytes] = {}
        # refcount-0 cached blocks, insertion order = LRU
        self._evictable: "OrderedDict[int, None]" = OrderedDict()
        self.cache_hit_tokens = 0
        self.cache_query_tokens = 0

    # -- capacity ----------------------------------------------------------
    @property
    def num_free_blocks(self) -> int:
        # evictable cached blocks are allocatable (cache is best-effort)
        return len(self._free) + len(self._evictable)

    @property
    def usage(self) -> float:
        return 1.0 - len(self._free) / self.num_blocks

    def blocks_needed(self, num_tokens: int) -> int:
        return cdiv(num_tokens, self.block_size)

    def can_allocate(self, request: Request, num_tokens: Optional[int] = None) -> bool:
        n = self.blocks_needed(num_tokens or request.num_tokens)
        return self.num_free_blocks - n >= self.watermark_blocks

    def can_append(self, request: Request) -> bool:
        """True if one more token can be placed (possibly needing a new block)."""
        if self._slots_free_in_last_block(request) > 0:
            return True
        return self.num_free_blocks >= 1

    # -- prefix-cache internals ---------------------------------------------
    def _prompt_block_hashes(self, request: Request, limit_tokens: int) -> List[bytes]:
        """Chained hashes of the FULL prompt blocks within limit_tokens,
        capped so the final prompt token is always computed fresh (its
        logits are needed for the first sampled token)."""
        bs = self.block_size
        n_prompt = request.num_prompt_tokens
        max_full = min(limit_tokens, n_prompt - 1) // bs
        toks = request.prompt_token_ids
        hashes: List[bytes] = []
        # chained sha256 over packed token ids: collision-proof against both
        # accidental and adversarially crafted prompts (a 64-bit hash()
        # collision would alias different content to the same KV block and
        # leak context across requests). Seeded with the LoRA adapter id:
        # adapters change the K/V projections, so identical tokens under
        # different adapters must NOT share blocks.
        h = hashlib.sha256(
            b"lora:%d" % getattr(request, "lora_id", 0)
        ).digest()
        for i in range(max_full):
            chunk = toks[i * bs : (i + 1) * bs]
            h = hashlib.sha256(
                h + b"".join(t.to_bytes(8, "little") for t in chunk)
            ).digest()
            hashes.append(h)
        return hashes

    def _take_block(self) -> int:
        if self._free:
            return self._free.pop()
        if self._evictable:
            # evict the least-recently-freed cached block
            blk, _ = self._evictable.popitem(last=False)
            h = self._block_to_hash.pop(blk, None)
            if h is not None:
                self._hash_to_block.pop(h, None)
            return blk
        raise RuntimeError("Out of KV blocks")

    def _release_block(self, blk: int) -> None:
        rc = self._refcount.get(blk, 1) - 1
        if rc > 0:
            self._refcount[blk] = rc
            return
        self._refcount.pop(blk, None)
        if blk in self._block_to_hash:
            self._evictable[blk] = None  # stays in cache, LRU-evictable
        else:
            self._free.append(blk)

    def query_cached_prefix(self, request: Request) -> int:
        """Tokens of the request's prompt currently resident in the prefix
        cache (pure lookup, no refcounting) — lets the scheduler size the
        chunk before allocate() claims the blocks."""
        if not self.enable_prefix_caching:
            return 0
        hashes = self._prompt_block_hashes(request, request.num_prompt_tokens)
        n = 0
        for h in hashes:
            if h not in self._hash_to_block:
                break
            n += 1
        return n * self.block_size

    def register_computed_blocks(self, request: Request) -> None:
        """Publish the request's fully-computed full prompt blocks into the
        prefix cache (called by the engine AFTER their KV exists)."""
        if not self.enable_prefix_caching:
            return
        table = self._tables.get(request.request_id)
        if not table:
            return
        hashes = getattr(request, "prompt_block_hashes", None)
        if not hashes:
            return
        bs = self.block_size
        n_ready = min(len(hashes), request.num_computed_tokens // bs, len(table))
        for i in range(n_ready):
            blk = table[i]
            if blk in self._block_to_hash:
                continue
            h = hashes[i]
            if h in self._hash_to_block:
                continue  # identical content already published; keep first
            self._hash_to_block[h] = blk
            self._block_to_hash[blk] = h

    # -- operations ----------------------------------------------------------
    def allocate(self, request: Request, num_tokens: Optional[int] = None) -> List[int]:
        """Allocate blocks to hold ``num_tokens`` (default: all request
        tokens). With prefix caching on, full prompt blocks whose content is
        already resident are shared (refcounted) and reported via
        ``request.num_cached_tokens``."""
        assert request.request_id not in self._tables, "already allocated"
        target = num_tokens or request.num_tokens
        n = self.blocks_needed(target)
        blocks: List[int] = []
        cached_tokens = 0
        if self.enable_prefix_caching:
            hashes = self._prompt_block_hashes(request, target)
            request.prompt_block_hashes = hashes
            self.cache_query_tokens += len(hashes) * self.block_size
            for h in hashes:
                blk = self._hash_to_block.get(h)
                if blk is None:
                    break
                blocks.append(blk)
                self._refcount[blk] = self._refcount.get(blk, 0) + 1
                self._evictable.pop(blk, None)
                cached_tokens += self.block_size
            self.cache_hit_tokens += cached_tokens
        if n - len(blocks) > self.num_free_blocks:
            # roll back the shared refs before failing
            for blk in blocks:
                self._release_block(blk)
            raise RuntimeError("Out of KV blocks")
        for _ in range(n - len(blocks)):
            blk = self._take_block()
            self._refcount[blk] = 1
            blocks.append(blk)
        self._tables[request.request_id] = blocks
        self._seq += 1
        self._table_seq[request.request_id] = self._seq
        request.block_table = blocks
        request.num_cached_tokens = cached_tokens
        return blocks

    def table_seq(self, request_id: str) -> int:
        """Identity of the request's current block table (changes whenever
        the block ids are reassigned, not merely appended to)."""
        return self._table_seq.get(request_id, -1)

    def take_blocks(self, n: int) -> List[int]:
        """Pop n fresh blocks (refcount 1), evicting cached blocks if the
        free list is short."""
        out = []
        for _ in range(n):
            blk = self._take_block()
            self._refcount[blk] = 1
            out.append(blk)
        return out

    def append_slot(self, request: Request) -> None:
        """Ensure capacity for one more token in the request's table."""
        table = self._tables[request.request_id]
        while len(table) * self.block_size < request.num_tokens:
            blk = self._take_block()
            self._refcount[blk] = 1
            table.append(blk)
        request.block_table = table

    def _slots_free_in_last_block(self, request: Request) -> int:
        table = self._tables.get(request.request_id)
        if not table:
            return 0
        capacity = len(table) * self.block_size
        return capacity - request.num_tokens

    def free(self, request: Request) -> None:
        table = self._tables.pop(request.request_id, None)
        self._table_seq.pop(request.request_id, None)
        if table:
            for blk in reversed(table):
                self._release_block(blk)
        request.block_table = []

    def get_block_table(self, request: Request) -> List[int]:
        return self._tables.get(request.request_id, [])

    def slot_mapping(self, request: Request, start: int, end: int) -> List[int]:
        """Physical slot index (block_id*block_size + offset) for token
        positions [start, end)."""
        table = self._tables[request.request_id]
        out = []
        for pos in range(start, end):
            b = table[pos // self.block_size]
            out.append(b * self.block_size + pos % self.block_size)
        return out


    # -- host-DRAM offload tier (north-star KV offload) ------
    def can_swap_out(self, request: Request) -> bool:
        table = self._tables.get(request.request_id)
        if not table or len(self._cpu_free) < len(table):
            return False
        # shared (prefix-cached) blocks cannot leave the GPU
        return all(self._refcount.get(b, 1) == 1 for b in table)

    def swap_out(self, request: Request) -> List[tuple]:
        """Move the request's blocks to the CPU tier. Returns (gpu, cpu)
        pairs; caller must COPY gpu->cpu before any kernel reuses the gpu
        blocks (the runner orders this via a side-stream event)."""
        table = self._tables.pop(request.request_id)
        self._table_seq.pop(request.request_id, None)
        cpu_blocks = [self._cpu_free.pop() for _ in table]
        self._cpu_tables[request.request_id] = cpu_blocks
        pairs = list(zip(table, cpu_blocks))
        for blk in reversed(table):
            # content leaves the GPU: unpublish from the prefix cache
            h = self._block_to_hash.pop(blk, None)
            if h is not None:
                self._hash_to_block.pop(h, None)
            self._refcount.pop(blk, None)
            self._free.append(blk)
        request.block_table = []
        return pairs

    def can_swap_in(self, request: Request) -> bool:
        cpu_table = self._cpu_tables.get(request.request_id)
        return bool(cpu_table) and self.num_free_blocks - len(cpu_table) >= self.watermark_blocks

    def swap_in(self, request: Request) -> List[tuple]:
        """Restore CPU-tier blocks to GPU. Returns (cpu, gpu) pairs."""
        cpu_table = self._cpu_tables.pop(request.request_id)
        gpu_blocks = []
        for _ in cpu_table:
            blk = self._take_block()
            self._refcount[blk] = 1
            gpu_blocks.append(blk)
        self._tables[request.request_id] = gpu_blocks
        self._seq += 1
        self._table_seq[request.request_id] = self._seq
        request.block_table = gpu_blocks
        pairs = list(zip(cpu_table, gpu_blocks))
        self._cpu_free.extend(reversed(cpu_table))
        return pairs

    def free_cpu(self, request: Request) -> None:
        cpu_table = self._cpu_tables.pop(request.request_id, None)
        if cpu_table:
            self._cpu_free.extend(reversed(cpu_table))
