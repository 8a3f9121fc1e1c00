"""Paged KV-cache block allocator.

Fresh design for 288 GB HBM3E: a flat free-list of fixed-size blocks
(block_size tokens each) with per-request block tables. Supports
allocate-on-prefill, append-on-decode, free, and full preemption
(recompute-style: blocks are released and the request re-prefills later).

Replaces the paged-KV machinery the reference delegates to vLLM
(SURVEY.md §2.7 "Paged attention" row).
"""

from __future__ import annotations

from typing import Dict, List, Optional

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
    ):
        self.num_blocks = num_blocks
        self.block_size = block_size
        # watermark: headroom kept free so decodes can always append
        self.watermark_blocks = max(1, int(num_blocks * watermark))
        # block 0 is reserved as the scratch target for hipGraph padding
        # (padded decode lanes write their KV there); never handed out.
        self._free: List[int] = list(range(1, num_blocks))
        self._tables: Dict[str, List[int]] = {}
        # host-DRAM offload tier (pinned; hipMemcpyAsync side stream)
        self.num_cpu_blocks = num_cpu_blocks
        self._cpu_free: List[int] = list(range(num_cpu_blocks))
        self._cpu_tables: Dict[str, List[int]] = {}

    # -- capacity ----------------------------------------------------------
    @property
    def num_free_blocks(self) -> int:
        return len(self._free)

    @property
    def usage(self) -> float:
        return 1.0 - len(self._free) / self.num_blocks

    def blocks_needed(self, num_tokens: int) -> int:
        return cdiv(num_tokens, self.block_size)

    def can_allocate(self, request: Request, num_tokens: Optional[int] = None) -> bool:
        n = self.blocks_needed(num_tokens or request.num_tokens)
        return len(self._free) - n >= self.watermark_blocks

    def can_append(self, request: Request) -> bool:
        """True if one more token can be placed (possibly needing a new block)."""
        if self._slots_free_in_last_block(request) > 0:
            return True
        return len(self._free) >= 1

    # -- operations ----------------------------------------------------------
    def allocate(self, request: Request, num_tokens: Optional[int] = None) -> List[int]:
        """Allocate blocks to hold ``num_tokens`` (default: all request tokens)."""
        assert request.request_id not in self._tables, "already allocated"
        n = self.blocks_needed(num_tokens or request.num_tokens)
        if n > len(self._free):
            raise RuntimeError("Out of KV blocks")
        blocks = [self._free.pop() for _ in range(n)]
        self._tables[request.request_id] = blocks
        request.block_table = blocks
        return blocks

    def append_slot(self, request: Request) -> None:
        """Ensure capacity for one more token in the request's table."""
        table = self._tables[request.request_id]
        while len(table) * self.block_size < request.num_tokens:
            if not self._free:
                raise RuntimeError("Out of KV blocks on append")
            table.append(self._free.pop())
        request.block_table = table

    def _slots_free_in_last_block(self, request: Request) -> int:
        table = self._tables.get(request.request_id)
        if not table:
            return 0
        capacity = len(table) * self.block_size
        return capacity - request.num_tokens

    def free(self, request: Request) -> None:
        table = self._tables.pop(request.request_id, None)
        if table:
            self._free.extend(reversed(table))
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
        return bool(table) and len(self._cpu_free) >= len(table)

    def swap_out(self, request: Request) -> List[tuple]:
        """Move the request's blocks to the CPU tier. Returns (gpu, cpu)
        pairs; caller must COPY gpu->cpu before any kernel reuses the gpu
        blocks (the runner orders this via a side-stream event)."""
        table = self._tables.pop(request.request_id)
        cpu_blocks = [self._cpu_free.pop() for _ in table]
        self._cpu_tables[request.request_id] = cpu_blocks
        pairs = list(zip(table, cpu_blocks))
        self._free.extend(reversed(table))
        request.block_table = []
        return pairs

    def can_swap_in(self, request: Request) -> bool:
        cpu_table = self._cpu_tables.get(request.request_id)
        return bool(cpu_table) and len(self._free) - len(cpu_table) >= self.watermark_blocks

    def swap_in(self, request: Request) -> List[tuple]:
        """Restore CPU-tier blocks to GPU. Returns (cpu, gpu) pairs."""
        cpu_table = self._cpu_tables.pop(request.request_id)
        gpu_blocks = [self._free.pop() for _ in cpu_table]
        self._tables[request.request_id] = gpu_blocks
        request.block_table = gpu_blocks
        pairs = list(zip(cpu_table, gpu_blocks))
        self._cpu_free.extend(reversed(cpu_table))
        return pairs

    def free_cpu(self, request: Request) -> None:
        cpu_table = self._cpu_tables.pop(request.request_id, None)
        if cpu_table:
            self._cpu_free.extend(reversed(cpu_table))
