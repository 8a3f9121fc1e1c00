"""Property-based block-manager invariants (hypothesis): random sequences
of allocate / append / free / swap / prefix-cache operations must never
double-hand-out a block, leak blocks, or corrupt refcounts."""

from hypothesis import given, settings, strategies as st

from kserve_amd.engine.block_manager import BlockManager
from kserve_amd.engine.request import Request
from kserve_amd.engine.sampling_params import SamplingParams

NUM_BLOCKS = 24
BS = 4


def mk_req(rid, tokens):
    return Request(
        str(rid), list(tokens), SamplingParams(max_tokens=4), eos_token_id=-1
    )


def check_invariants(bm: BlockManager, live):
    """No block appears twice across live tables + free list + evictable."""
    seen = {}
    for rid, req in live.items():
        for b in bm.get_block_table(req):
            assert b != 0, "block 0 is reserved for hipGraph scratch"
            seen.setdefault(b, []).append(rid)
    # a block shared by several requests must be prefix-cached (refcount>1)
    for b, owners in seen.items():
        if len(owners) > 1:
            assert bm._refcount.get(b, 1) >= len(owners), (b, owners)
    for b in bm._free:
        assert b not in seen, f"free block {b} still referenced"
        assert b not in bm._evictable
    for b in bm._evictable:
        assert b not in seen or bm._refcount.get(b, 0) > 0
    # conservation: free + evictable + uniquely-referenced <= capacity
    assert len(bm._free) + len(bm._evictable) + len(seen) <= NUM_BLOCKS - 1


ops_strategy = st.lists(
    st.tuples(
        st.sampled_from(["alloc", "append", "free", "swap_out", "swap_in"]),
        st.integers(min_value=0, max_value=7),   # request slot
        st.integers(min_value=1, max_value=20),  # prompt length
        st.integers(min_value=0, max_value=3),   # prompt variant (cache hits)
    ),
    min_size=1,
    max_size=60,
)


@settings(max_examples=120, deadline=None)
@given(ops=ops_strategy, prefix=st.booleans())
def test_random_op_sequences_keep_invariants(ops, prefix):
    bm = BlockManager(
        NUM_BLOCKS, BS, num_cpu_blocks=8, enable_prefix_caching=prefix
    )
    live = {}
    swapped = {}
    next_id = 0
    for op, slot, plen, variant in ops:
        rid = f"r{slot}"
        if op == "alloc" and rid not in live and rid not in swapped:
            req = mk_req(f"{rid}-{next_id}", [variant] * plen)
            req.request_id = rid
            next_id += 1
            if bm.can_allocate(req):
                try:
                    bm.allocate(req)
                    live[rid] = req
                except RuntimeError:
                    pass
        elif op == "append" and rid in live:
            req = live[rid]
            if bm.can_append(req):
                req.append_output_token(99)
                try:
                    bm.append_slot(req)
                except RuntimeError:
                    pass
        elif op == "free" and rid in live:
            req = live.pop(rid)
            if prefix:
                req.num_computed_tokens = req.num_tokens
                bm.register_computed_blocks(req)
            bm.free(req)
        elif op == "swap_out" and rid in live:
            req = live[rid]
            if bm.can_swap_out(req):
                bm.swap_out(req)
                swapped[rid] = live.pop(rid)
        elif op == "swap_in" and rid in swapped:
            req = swapped[rid]
            if bm.can_swap_in(req):
                bm.swap_in(req)
                live[rid] = swapped.pop(rid)
        check_invariants(bm, live)
    # drain: free everything; all non-evictable blocks return
    for req in list(live.values()):
        bm.free(req)
    for req in list(swapped.values()):
        bm.free_cpu(req)
    assert len(bm._free) + len(bm._evictable) == NUM_BLOCKS - 1
    assert len(bm._cpu_free) == 8
