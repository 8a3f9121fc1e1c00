"""Property-based invariants for the continuous-batching scheduler: a
random engine-loop simulation (arrivals, chunked prefill, decode,
preemption via KV exhaustion, finishes) must never lose a request,
overrun seats or the token budget, or leak KV blocks."""

import hypothesis.strategies as st
from hypothesis import settings
from hypothesis.stateful import (
    RuleBasedStateMachine,
    invariant,
    precondition,
    rule,
)

from kserve_amd.engine.config import CacheConfig, SchedulerConfig
from kserve_amd.engine.request import Request, RequestStatus
from kserve_amd.engine.sampling_params import SamplingParams
from kserve_amd.engine.scheduler import Scheduler

MAX_SEQS = 6
TOKEN_BUDGET = 32
NUM_BLOCKS = 24
BLOCK_SIZE = 4


class SchedulerMachine(RuleBasedStateMachine):
    def __init__(self):
        super().__init__()
        self.sched = Scheduler(
            SchedulerConfig(
                max_num_seqs=MAX_SEQS,
                max_num_batched_tokens=TOKEN_BUDGET,
                max_model_len=64,
                enable_chunked_prefill=True,
            ),
            CacheConfig(block_size=BLOCK_SIZE),
            num_gpu_blocks=NUM_BLOCKS,
            num_cpu_blocks=8,
        )
        self.all = {}
        self.finished = set()
        self.counter = 0

    @rule(prompt_len=st.integers(min_value=1, max_value=40))
    def arrive(self, prompt_len):
        rid = f"r{self.counter}"
        self.counter += 1
        req = Request(rid, list(range(prompt_len)), SamplingParams())
        self.sched.add_request(req)
        self.all[rid] = req

    @rule()
    def step(self):
        """One engine iteration: schedule, then emulate the model step."""
        batch = self.sched.schedule()
        if not batch:
            return
        if batch.is_prefill:
            assert batch.total_tokens <= TOKEN_BUDGET, (
                f"prefill budget overrun: {batch.total_tokens}"
            )
            for req, n in zip(batch.requests, batch.num_scheduled_tokens):
                assert n > 0
                req.num_computed_tokens += n
                if req.num_computed_tokens >= req.num_prompt_tokens:
                    # prompt done: first output token materializes
                    req.output_token_ids.append(7)
                else:
                    # chunked: back to waiting for the next chunk
                    self.sched.requeue_partial_prefill(req)
        else:
            for req, n in zip(batch.requests, batch.num_scheduled_tokens):
                assert n == 1
                req.output_token_ids.append(7)
                req.num_computed_tokens += 1

    @precondition(lambda self: any(
        r.status == RequestStatus.RUNNING for r in self.all.values()))
    @rule(data=st.data())
    def finish_one(self, data):
        running = [r for r in self.all.values()
                   if r.status == RequestStatus.RUNNING]
        req = data.draw(st.sampled_from(sorted(running,
                                               key=lambda r: r.request_id)))
        req.status = RequestStatus.FINISHED_STOPPED
        self.sched.finish_requests([req])
        self.finished.add(req.request_id)

    # -- invariants --------------------------------------------------------
    @invariant()
    def seats_bounded(self):
        assert len(self.sched.running) <= MAX_SEQS

    @invariant()
    def nothing_lost(self):
        tracked = (
            {r.request_id for r in self.sched.waiting}
            | {r.request_id for r in self.sched.running}
            | {r.request_id for r in self.sched.swapped}
            | self.finished
        )
        assert tracked == set(self.all), (
            f"lost: {set(self.all) - tracked}"
        )
        # and each request sits in exactly one place
        n = (len(self.sched.waiting) + len(self.sched.running)
             + len(self.sched.swapped) + len(self.finished))
        assert n == len(self.all)

    @invariant()
    def finished_requests_hold_no_kv(self):
        bm = self.sched.block_manager
        for rid in self.finished:
            assert rid not in bm._tables
            assert rid not in bm._cpu_tables

    @invariant()
    def running_have_tables_waiting_usually_dont(self):
        bm = self.sched.block_manager
        for r in self.sched.running:
            assert r.request_id in bm._tables
        # swapped requests keep only CPU-tier tables
        for r in self.sched.swapped:
            assert r.request_id not in bm._tables
            assert r.request_id in bm._cpu_tables


TestSchedulerProperties = SchedulerMachine.TestCase
TestSchedulerProperties.settings = settings(
    max_examples=60, stateful_step_count=50, deadline=None
)
