"""Continuous-batching scheduler.

Fresh design (vLLM *semantics* per SURVEY.md §7 stage 4, no code studied):
prefill-priority batching — each step schedules either a prefill batch
(waiting requests, FIFO, bounded by token/seq budgets and KV headroom) or a
decode batch (all running requests, one token each). Out-of-blocks on decode
preempts the newest running request (recompute-style: its blocks are freed
and it returns to the head of the waiting queue).

This is the piece the reference's HF fallback lacks (request-serial,
generative_model.py:372) and its vLLM path outsources.
"""

from __future__ import annotations

from collections import deque
from dataclasses import dataclass, field
from typing import Deque, List, Optional

from kserve_amd.engine.block_manager import BlockManager
from kserve_amd.engine.config import CacheConfig, SchedulerConfig
from kserve_amd.engine.request import Request, RequestStatus
from kserve_amd.metrics import LLM_NUM_RUNNING, LLM_NUM_WAITING, LLM_PREEMPTIONS


@dataclass
class ScheduledBatch:
    is_prefill: bool
    requests: List[Request] = field(default_factory=list)
    # per-request number of tokens to run this step (prefill: prompt chunk;
    # decode: 1)
    num_scheduled_tokens: List[int] = field(default_factory=list)
    preempted: List[Request] = field(default_factory=list)
    # KV offload tier: (gpu_block, cpu_block) / (cpu_block, gpu_block) pairs
    # the runner must copy before this batch executes
    swap_out: List[tuple] = field(default_factory=list)
    swap_in: List[tuple] = field(default_factory=list)

    @property
    def total_tokens(self) -> int:
        return sum(self.num_scheduled_tokens)

    def __bool__(self):
        return bool(self.requests)


class Scheduler:
    def __init__(
        self,
        scheduler_config: SchedulerConfig,
        cache_config: CacheConfig,
        num_gpu_blocks: int,
        num_cpu_blocks: int = 0,
    ):
        self.config = scheduler_config
        self.block_manager = BlockManager(
            num_gpu_blocks, cache_config.block_size,
            num_cpu_blocks=num_cpu_blocks or cache_config.num_cpu_blocks,
            enable_prefix_caching=cache_config.enable_prefix_caching,
        )
        # waiting queue ordered by (priority, arrival seq): lower priority
        # value first, FIFO within a priority class. A preempted or
        # partially-prefilled request re-enters at its ORIGINAL seq, so it
        # returns to the head of its class.
        self.waiting: List[Request] = []
        self._arrival_seq = 0
        self.running: List[Request] = []
        # requests whose KV lives in the host-DRAM tier
        self.swapped: List[Request] = []

    # -- queue ops -----------------------------------------------------------
    def _queue_key(self, request: Request):
        return (request.sampling_params.priority, request.arrival_seq)

    def _enqueue(self, request: Request) -> None:
        key = self._queue_key(request)
        for i, r in enumerate(self.waiting):
            if self._queue_key(r) > key:
                self.waiting.insert(i, request)
                return
        self.waiting.append(request)

    def add_request(self, request: Request) -> None:
        if request.num_prompt_tokens > self.config.max_model_len:
            request.status = RequestStatus.FINISHED_LENGTH
            request.is_finished = True
            return
        request.arrival_seq = self._arrival_seq
        self._arrival_seq += 1
        self._enqueue(request)

    def abort_request(self, request_id: str) -> Optional[Request]:
        for q in (self.waiting, self.running, self.swapped):
            for r in list(q):
                if r.request_id == request_id:
                    r.status = RequestStatus.FINISHED_ABORTED
                    r.is_finished = True
                    q.remove(r)
                    self.block_manager.free(r)
                    self.block_manager.free_cpu(r)
                    return r
        return None

    @property
    def num_waiting(self) -> int:
        return len(self.waiting)

    @property
    def num_running(self) -> int:
        return len(self.running)

    def has_unfinished(self) -> bool:
        return bool(self.waiting or self.running or self.swapped)

    # -- scheduling ----------------------------------------------------------
    def schedule(self) -> ScheduledBatch:
        batch = self._schedule_prefill()
        if not batch:
            batch = self._schedule_decode()
        LLM_NUM_RUNNING.set(len(self.running))
        LLM_NUM_WAITING.set(len(self.waiting))
        return batch

    def _schedule_prefill(self) -> ScheduledBatch:
        batch = ScheduledBatch(is_prefill=True)
        token_budget = self.config.max_num_batched_tokens
        while self.waiting:
            req = self.waiting[0]
            # prefix cache: size the chunk from the cached-prefix boundary
            # so budget/allocation cover exactly the tokens that will run
            cached_hint = 0
            if not req.block_table and req.num_computed_tokens == 0:
                cached_hint = self.block_manager.query_cached_prefix(req)
            base_computed = max(req.num_computed_tokens, cached_hint)
            # num_tokens (not num_prompt_tokens): a preempted request
            # re-prefills its generated tokens as context too
            n_new = req.num_tokens - base_computed
            if len(self.running) + len(batch.requests) + 1 > self.config.max_num_seqs:
                break
            if n_new > token_budget:
                if self.config.enable_chunked_prefill and token_budget > 0 and batch.total_tokens == 0:
                    n_new = token_budget
                else:
                    break
            if req.block_table:
                # resumed chunked prefill: blocks may partially exist
                need = self.block_manager.blocks_needed(
                    req.num_computed_tokens + n_new
                ) - len(req.block_table)
                if need > self.block_manager.num_free_blocks:
                    break
                req.block_table.extend(self.block_manager.take_blocks(need))
            else:
                if not self.block_manager.can_allocate(
                    req, base_computed + n_new
                ):
                    break
                self.block_manager.allocate(req, base_computed + n_new)
                cached = getattr(req, "num_cached_tokens", 0)
                if cached > req.num_computed_tokens:
                    # prefix-cache hit: KV for the prefix already resident;
                    # only the suffix runs (paged-context prefill path).
                    # cached == cached_hint (same cache state as the query
                    # above), so the allocation already covers the chunk;
                    # the capacity clamp is a safety net only.
                    req.num_computed_tokens = cached
                    capacity = (
                        len(req.block_table) * self.block_manager.block_size
                    )
                    n_new = min(req.num_tokens - cached, capacity - cached)
            self.waiting.pop(0)
            req.status = RequestStatus.RUNNING
            batch.requests.append(req)
            batch.num_scheduled_tokens.append(n_new)
            token_budget -= n_new
        for r in batch.requests:
            self.running.append(r)
        return batch

    def _schedule_decode(self) -> ScheduledBatch:
        batch = ScheduledBatch(is_prefill=False)
        # resume swapped requests first (KV restored from the host tier,
        # no recompute) while capacity allows
        while (
            self.swapped
            and len(self.running) < self.config.max_num_seqs
            and self.block_manager.can_swap_in(self.swapped[0])
        ):
            req = self.swapped.pop(0)
            batch.swap_in.extend(self.block_manager.swap_in(req))
            req.status = RequestStatus.RUNNING
            self.running.append(req)
        if not self.running:
            return batch
        # ensure every running request can take one more token; preempt the
        # newest-arrived requests when blocks run out (recompute preemption)
        scheduled: List[Request] = list(self.running)

        def blocks_needed_now(req: Request) -> int:
            # blocks to extend capacity to num_tokens (the token being decoded)
            have = len(self.block_manager.get_block_table(req))
            need = self.block_manager.blocks_needed(req.num_tokens)
            return max(0, need - have)

        while scheduled:
            # fast path: each decode step needs at most one new block per
            # request, so enough free blocks means no preemption check
            if self.block_manager.num_free_blocks >= len(scheduled):
                break
            total_needed = sum(blocks_needed_now(r) for r in scheduled)
            if total_needed <= self.block_manager.num_free_blocks:
                break
            victim = scheduled.pop()  # newest in running order
            if self.block_manager.can_swap_out(victim):
                batch.swap_out.extend(self._preempt_swap(victim))
            else:
                self._preempt(victim)
            batch.preempted.append(victim)
        for req in scheduled:
            self.block_manager.append_slot(req)
            batch.requests.append(req)
            batch.num_scheduled_tokens.append(1)
        self.running = scheduled
        return batch

    def _preempt(self, req: Request) -> None:
        """Recompute-style preemption: free blocks, re-prefill later.

        Generated tokens stay in ``output_token_ids`` (accounting is
        unchanged); ``num_computed_tokens=0`` makes the next prefill
        reprocess ``all_token_ids``.
        """
        LLM_PREEMPTIONS.inc()
        self.block_manager.free(req)
        req.num_computed_tokens = 0
        req.status = RequestStatus.PREEMPTED
        self._enqueue(req)

    def requeue_partial_prefill(self, req: Request) -> None:
        """A chunked prefill finished its chunk but not the prompt: return it
        to the head of the waiting queue (KV blocks kept) so the next step
        schedules the following chunk instead of decoding it."""
        self.running.remove(req)
        req.status = RequestStatus.WAITING
        self._enqueue(req)

    def _preempt_swap(self, req: Request):
        """Offload preemption: KV pages move to pinned host DRAM; the request
        resumes later via swap_in with no recompute."""
        LLM_PREEMPTIONS.inc()
        pairs = self.block_manager.swap_out(req)
        req.status = RequestStatus.PREEMPTED
        self.swapped.append(req)
        return pairs

    # -- post-step ------------------------------------------------------------
    def finish_requests(self, finished: List[Request]) -> None:
        for req in finished:
            if req in self.running:
                self.running.remove(req)
            self.block_manager.free(req)
            self.block_manager.free_cpu(req)

    # -- multi-step decode windows -------------------------------------------
    # window length while admission is blocked: a waiting request was
    # already unschedulable THIS step (seats or KV), so a short window
    # cannot starve it — it bounds the extra admission delay instead of
    # dropping to one-token steps under steady trickle-in traffic
    BLOCKED_ADMISSION_WINDOW = 4

    def reserve_decode_window(self, batch: ScheduledBatch, max_k: int) -> int:
        """Largest k such that all scheduled requests can decode k steps with
        no scheduling events (no swaps, no preemption, no per-request
        finish before step k except stop-token/EOS which is handled
        post-hoc). Reserves the KV blocks for the window.

        A non-empty waiting queue here means _schedule_prefill just failed
        to admit its head (this method is only reached from the decode
        branch), so decoding cannot make that request MORE blocked; the
        window is merely capped to bound the admission latency."""
        if max_k <= 1 or self.swapped or batch.preempted:
            return 1
        if self.waiting:
            max_k = min(max_k, self.BLOCKED_ADMISSION_WINDOW)
            if max_k <= 1:
                return 1
        reqs = batch.requests
        if not reqs:
            return 1
        k = max_k
        for r in reqs:
            sp = r.sampling_params
            if sp.max_tokens is not None:
                k = min(k, sp.max_tokens - r.num_output_tokens)
            k = min(k, self.config.max_model_len - r.num_tokens)
        if k <= 1:
            return 1
        bm = self.block_manager
        bs = bm.block_size
        while k > 1:
            needed = 0
            for r in reqs:
                have = len(bm.get_block_table(r)) * bs
                want = r.num_tokens + k - 1  # max position + 1 in the window
                if want > have:
                    needed += (want - have + bs - 1) // bs
            if needed <= bm.num_free_blocks - bm.watermark_blocks:
                break
            k -= max(1, k // 2)
        if k <= 1:
            return 1
        for r in reqs:
            want = r.num_tokens + k - 1
            table = bm._tables[r.request_id]
            while len(table) * bs < want:
                table.extend(bm.take_blocks(1))
            r.block_table = table
        return k
