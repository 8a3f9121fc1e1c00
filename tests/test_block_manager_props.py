"""Property-based invariants for the KV BlockManager (hypothesis
stateful testing): arbitrary interleavings of allocate / append / free /
swap-out / swap-in must preserve block accounting — no double ownership,
no leaks, table identity bumps on every (re)assignment.

The reference has no allocator of its own (vLLM's is external); this is
the safety net for OUR pager, complementing the example-based
tests/test_block_manager.py.
"""

import hypothesis.strategies as st
from hypothesis import settings
from hypothesis.stateful import (
    RuleBasedStateMachine,
    invariant,
    precondition,
    rule,
)

from kserve_amd.engine.block_manager import BlockManager
from kserve_amd.engine.request import Request, RequestStatus
from kserve_amd.engine.sampling_params import SamplingParams

NUM_BLOCKS = 32
BLOCK_SIZE = 4
NUM_CPU = 16


class BlockManagerMachine(RuleBasedStateMachine):
    def __init__(self):
        super().__init__()
        self.bm = BlockManager(
            NUM_BLOCKS, BLOCK_SIZE, num_cpu_blocks=NUM_CPU,
            enable_prefix_caching=False,
        )
        self.live = {}      # rid -> Request with a GPU table
        self.swapped = {}   # rid -> Request with a CPU table
        self.counter = 0

    # -- actions -----------------------------------------------------------
    @rule(prompt_len=st.integers(min_value=1, max_value=24))
    def allocate(self, prompt_len):
        rid = f"r{self.counter}"
        self.counter += 1
        req = Request(rid, list(range(prompt_len)), SamplingParams())
        req.num_computed_tokens = 0
        if self.bm.can_allocate(req):
            before = self.bm.table_seq(rid)
            self.bm.allocate(req)
            assert self.bm.table_seq(rid) != before
            req.num_computed_tokens = prompt_len
            req.status = RequestStatus.RUNNING
            self.live[rid] = req

    @precondition(lambda self: self.live)
    @rule(data=st.data())
    def append(self, data):
        rid = data.draw(st.sampled_from(sorted(self.live)))
        req = self.live[rid]
        if self.bm.can_append(req):
            req.output_token_ids.append(0)
            self.bm.append_slot(req)

    @precondition(lambda self: self.live)
    @rule(data=st.data())
    def free(self, data):
        rid = data.draw(st.sampled_from(sorted(self.live)))
        self.bm.free(self.live.pop(rid))

    @precondition(lambda self: self.live)
    @rule(data=st.data())
    def swap_out(self, data):
        rid = data.draw(st.sampled_from(sorted(self.live)))
        req = self.live[rid]
        if self.bm.can_swap_out(req):
            self.bm.swap_out(req)
            self.swapped[rid] = self.live.pop(rid)

    @precondition(lambda self: self.swapped)
    @rule(data=st.data())
    def swap_in(self, data):
        rid = data.draw(st.sampled_from(sorted(self.swapped)))
        req = self.swapped[rid]
        if self.bm.can_swap_in(req):
            before = self.bm.table_seq(rid)
            self.bm.swap_in(req)
            assert self.bm.table_seq(rid) != before
            self.live[rid] = self.swapped.pop(rid)

    # -- invariants --------------------------------------------------------
    @invariant()
    def no_double_ownership(self):
        seen = set()
        for rid, table in self.bm._tables.items():
            for blk in table:
                assert blk not in seen, f"block {blk} in two tables"
                assert blk != 0, "reserved scratch block handed out"
                seen.add(blk)
        assert not (seen & set(self.bm._free)), "live block also free"

    @invariant()
    def gpu_accounting_balances(self):
        held = sum(len(t) for t in self.bm._tables.values())
        # block 0 reserved; free + held must cover the rest exactly
        assert held + len(self.bm._free) == NUM_BLOCKS - 1

    @invariant()
    def cpu_accounting_balances(self):
        held = sum(len(t) for t in self.bm._cpu_tables.values())
        assert held + len(self.bm._cpu_free) == NUM_CPU
        for rid in self.bm._cpu_tables:
            assert rid in self.swapped, "cpu table without swapped request"

    @invariant()
    def tables_match_request_lengths(self):
        for rid, req in self.live.items():
            need = self.bm.blocks_needed(req.num_tokens)
            assert len(self.bm._tables[rid]) >= need


TestBlockManagerProperties = BlockManagerMachine.TestCase
TestBlockManagerProperties.settings = settings(
    max_examples=60, stateful_step_count=40, deadline=None
)


class PrefixCacheMachine(RuleBasedStateMachine):
    """Same accounting, with the content-addressed prefix cache on: a
    block is in exactly one of {_free, _evictable, live tables}; blocks
    shared across tables carry refcounts covering every owner; eviction
    never hands out a block still referenced."""

    def __init__(self):
        super().__init__()
        self.bm = BlockManager(
            NUM_BLOCKS, BLOCK_SIZE, enable_prefix_caching=True
        )
        self.live = {}
        self.counter = 0

    @rule(
        # tiny alphabet + short prompts -> frequent identical prefixes
        prompt=st.lists(st.integers(min_value=0, max_value=2),
                        min_size=1, max_size=16)
    )
    def allocate(self, prompt):
        rid = f"r{self.counter}"
        self.counter += 1
        req = Request(rid, prompt, SamplingParams())
        req.num_computed_tokens = 0
        self.bm.query_cached_prefix(req)
        if self.bm.can_allocate(req):
            self.bm.allocate(req)
            req.num_computed_tokens = len(prompt)
            req.status = RequestStatus.RUNNING
            self.live[rid] = req

    @precondition(lambda self: self.live)
    @rule(data=st.data())
    def publish_computed(self, data):
        rid = data.draw(st.sampled_from(sorted(self.live)))
        self.bm.register_computed_blocks(self.live[rid])

    @precondition(lambda self: self.live)
    @rule(data=st.data())
    def append(self, data):
        rid = data.draw(st.sampled_from(sorted(self.live)))
        req = self.live[rid]
        if self.bm.can_append(req):
            req.output_token_ids.append(1)
            self.bm.append_slot(req)

    @precondition(lambda self: self.live)
    @rule(data=st.data())
    def free(self, data):
        rid = data.draw(st.sampled_from(sorted(self.live)))
        self.bm.free(self.live.pop(rid))

    @invariant()
    def ownership_partition(self):
        owners = {}
        for table in self.bm._tables.values():
            for blk in table:
                assert blk != 0
                owners[blk] = owners.get(blk, 0) + 1
        free = set(self.bm._free)
        evictable = set(self.bm._evictable)
        held = set(owners)
        assert not (free & evictable)
        assert not (free & held), "live block on the free list"
        assert not (evictable & held), "live block marked evictable"
        # full coverage of the pool (block 0 reserved)
        assert len(free) + len(evictable) + len(held) == NUM_BLOCKS - 1
        # shared blocks are refcounted for every owner
        for blk, n in owners.items():
            if n > 1:
                assert self.bm._refcount.get(blk, 0) >= n, (
                    f"block {blk} shared {n}x, refcount "
                    f"{self.bm._refcount.get(blk, 0)}"
                )

    @invariant()
    def cache_index_consistent(self):
        for h, blk in self.bm._hash_to_block.items():
            assert self.bm._block_to_hash.get(blk) == h
        assert len(self.bm._hash_to_block) == len(self.bm._block_to_hash)


TestPrefixCacheProperties = PrefixCacheMachine.TestCase
TestPrefixCacheProperties.settings = settings(
    max_examples=60, stateful_step_count=40, deadline=None
)
