from kserve_amd.engine.config import CacheConfig, SchedulerConfig
from kserve_amd.engine.request import Request, RequestStatus
from kserve_amd.engine.sampling_params import SamplingParams
from kserve_amd.engine.scheduler import Scheduler


def make_scheduler(num_blocks=64, block_size=4, max_seqs=8, max_tokens=64):
    return Scheduler(
        SchedulerConfig(
            max_num_seqs=max_seqs, max_num_batched_tokens=max_tokens, max_model_len=64
        ),
        CacheConfig(block_size=block_size),
        num_blocks,
    )


def make_req(rid, n_prompt, max_tokens=8):
    return Request(rid, list(range(n_prompt)), SamplingParams(max_tokens=max_tokens))


def test_prefill_first():
    s = make_scheduler()
    r1, r2 = make_req("a", 8), make_req("b", 8)
    s.add_request(r1)
    s.add_request(r2)
    batch = s.schedule()
    assert batch.is_prefill
    assert [r.request_id for r in batch.requests] == ["a", "b"]
    assert batch.num_scheduled_tokens == [8, 8]
    assert s.num_running == 2


def test_token_budget_limits_prefill():
    s = make_scheduler(max_tokens=10)
    s.add_request(make_req("a", 8))
    s.add_request(make_req("b", 8))
    batch = s.schedule()
    assert [r.request_id for r in batch.requests] == ["a"]
    # b scheduled next round
    for r in batch.requests:
        r.num_computed_tokens = r.num_prompt_tokens
        r.append_output_token(1)
    batch2 = s.schedule()
    assert batch2.is_prefill
    assert [r.request_id for r in batch2.requests] == ["b"]


def test_decode_after_prefill():
    s = make_scheduler()
    r = make_req("a", 8)
    s.add_request(r)
    batch = s.schedule()
    assert batch.is_prefill
    r.num_computed_tokens = 8
    r.append_output_token(1)
    batch2 = s.schedule()
    assert not batch2.is_prefill
    assert batch2.requests == [r]
    assert batch2.num_scheduled_tokens == [1]


def test_preemption_on_oom():
    # tiny pool: 7 usable blocks (block 0 reserved), block_size 4
    s = make_scheduler(num_blocks=8, block_size=4, max_tokens=128)
    r1, r2 = make_req("a", 12, max_tokens=32), make_req("b", 12, max_tokens=32)
    s.add_request(r1)
    s.add_request(r2)
    b = s.schedule()
    # both fit initially: 7 usable blocks, 3 each, watermark 1
    assert len(b.requests) == 2
    for r in (r1, r2):
        r.num_computed_tokens = 12
        r.append_output_token(1)
    # decode until the pool is exhausted -> newest request (b) preempted
    preempted = []
    for _ in range(16):
        batch = s.schedule()
        preempted.extend(batch.preempted)
        if preempted:
            break
        for req in batch.requests:
            req.num_computed_tokens += 1
            req.append_output_token(1)
    assert preempted and preempted[0] is r2
    assert s.num_waiting == 1  # preempted request re-queued
    assert s.num_running == 1


def test_finish_frees_blocks():
    s = make_scheduler()
    r = make_req("a", 8, max_tokens=1)
    s.add_request(r)
    s.schedule()
    free_before = s.block_manager.num_free_blocks
    r.num_computed_tokens = 8
    r.append_output_token(5)
    r.maybe_finish(64)
    assert r.finished
    s.finish_requests([r])
    assert s.block_manager.num_free_blocks > free_before
    assert s.num_running == 0


def test_too_long_prompt_rejected():
    s = make_scheduler()
    r = make_req("a", 100)
    s.add_request(r)
    assert r.status == RequestStatus.FINISHED_LENGTH
    assert s.num_waiting == 0


def test_preempted_request_refills_all_tokens():
    s = make_scheduler(num_blocks=64)
    r = make_req("a", 8)
    s.add_request(r)
    s.schedule()
    r.num_computed_tokens = 8
    r.append_output_token(1)
    r.append_output_token(2)
    r.num_computed_tokens = 10
    s._preempt(r)
    s.running.remove(r) if r in s.running else None
    assert r.num_computed_tokens == 0
    assert r.num_tokens == 10
    batch = s._schedule_prefill()
    assert batch.requests == [r]
    assert batch.num_scheduled_tokens == [10]


def test_priority_scheduling():
    """Lower priority value runs first; FIFO within a class; a preempted
    high-priority request returns to the head of its class."""
    from kserve_amd.engine.config import CacheConfig, SchedulerConfig
    from kserve_amd.engine.request import Request
    from kserve_amd.engine.sampling_params import SamplingParams
    from kserve_amd.engine.scheduler import Scheduler

    sched = Scheduler(
        SchedulerConfig(max_num_seqs=1, max_num_batched_tokens=64,
                        max_model_len=64),
        CacheConfig(block_size=4),
        num_gpu_blocks=64,
    )

    def add(rid, prio):
        r = Request(rid, [1, 2, 3],
                    SamplingParams(max_tokens=4, priority=prio),
                    eos_token_id=-1)
        sched.add_request(r)
        return r

    add("low-a", 5)
    add("normal", 0)
    add("low-b", 5)
    urgent = add("urgent", -1)
    order = [r.request_id for r in sched.waiting]
    assert order == ["urgent", "normal", "low-a", "low-b"]

    # max_num_seqs=1: only the urgent request is scheduled
    batch = sched.schedule()
    assert [r.request_id for r in batch.requests] == ["urgent"]
    # preempt it: it must come back ahead of everything in its class
    sched._preempt(urgent)
    sched.running.remove(urgent)
    assert sched.waiting[0].request_id == "urgent"


def test_decode_not_starved_when_seats_full():
    """Prefill priority is bounded by max_num_seqs: once the running set is
    full, waiting prefills cannot block decode progress."""
    from kserve_amd.engine.config import CacheConfig, SchedulerConfig
    from kserve_amd.engine.request import Request
    from kserve_amd.engine.sampling_params import SamplingParams
    from kserve_amd.engine.scheduler import Scheduler

    sched = Scheduler(
        SchedulerConfig(max_num_seqs=2, max_num_batched_tokens=64,
                        max_model_len=64),
        CacheConfig(block_size=4),
        num_gpu_blocks=64,
    )
    for i in range(4):
        sched.add_request(
            Request(f"r{i}", [1, 2, 3], SamplingParams(max_tokens=8),
                    eos_token_id=-1)
        )
    b1 = sched.schedule()
    assert b1.is_prefill and len(b1.requests) == 2  # seats now full
    for r in b1.requests:
        r.num_computed_tokens = r.num_tokens  # prompt done
        r.append_output_token(5)
    b2 = sched.schedule()
    assert not b2.is_prefill, "decode must proceed while prefills wait"
    assert len(b2.requests) == 2
    assert sched.num_waiting == 2


def test_window_capped_not_disabled_when_admission_blocked():
    """Round 2: a waiting request that cannot be admitted (seats full) no
    longer forces one-token decode steps — the window is capped to
    BLOCKED_ADMISSION_WINDOW instead (it was unschedulable this step
    either way)."""
    from kserve_amd.engine.config import (
        CacheConfig,
        EngineConfig,
        ModelConfig,
        SchedulerConfig,
    )
    from kserve_amd.engine.engine import LLMEngine
    from kserve_amd.engine.sampling_params import SamplingParams

    cfg = EngineConfig(
        model=ModelConfig.tiny(vocab_size=64),
        cache=CacheConfig(block_size=4, num_gpu_blocks=256),
        scheduler=SchedulerConfig(
            max_num_seqs=2, max_num_batched_tokens=64, max_model_len=64,
            multi_step=8,
        ),
        device="cpu",
        eos_token_id=-1,
    )
    eng = LLMEngine(cfg)
    sp = SamplingParams(temperature=0.0, max_tokens=24, ignore_eos=True)
    for i in range(3):  # 3 requests, 2 seats -> one always waits
        eng.add_request([1 + i, 2, 3], sp, request_id=f"r{i}")
    # prefill the two admitted requests
    eng.step()
    assert eng.scheduler.num_waiting == 1
    batch = eng.scheduler.schedule()
    assert not batch.is_prefill
    k = eng.scheduler.reserve_decode_window(batch, 8)
    assert k == eng.scheduler.BLOCKED_ADMISSION_WINDOW  # capped, not 1
