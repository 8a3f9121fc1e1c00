from kserve_amd.engine.block_manager import BlockManager
from kserve_amd.engine.request import Request
from kserve_amd.engine.sampling_params import SamplingParams


def make_req(rid, n_prompt):
    return Request(rid, list(range(n_prompt)), SamplingParams(max_tokens=4))


def test_block_zero_reserved():
    bm = BlockManager(num_blocks=8, block_size=4)
    r = make_req("a", 4)
    blocks = bm.allocate(r)
    assert 0 not in blocks


def test_allocate_and_slots():
    bm = BlockManager(num_blocks=16, block_size=4)
    r = make_req("a", 10)
    blocks = bm.allocate(r)
    assert len(blocks) == 3
    slots = bm.slot_mapping(r, 0, 10)
    assert len(slots) == 10
    assert slots[0] == blocks[0] * 4
    assert slots[5] == blocks[1] * 4 + 1


def test_append_slot_grows():
    bm = BlockManager(num_blocks=16, block_size=4)
    r = make_req("a", 4)
    bm.allocate(r)
    assert len(r.block_table) == 1
    r.append_output_token(42)  # num_tokens = 5
    bm.append_slot(r)
    assert len(r.block_table) == 2


def test_free_returns_blocks():
    bm = BlockManager(num_blocks=16, block_size=4)
    free0 = bm.num_free_blocks
    r = make_req("a", 12)
    bm.allocate(r)
    assert bm.num_free_blocks == free0 - 3
    bm.free(r)
    assert bm.num_free_blocks == free0


def test_can_allocate_watermark():
    bm = BlockManager(num_blocks=8, block_size=4, watermark=0.25)
    r = make_req("a", 24)  # needs 6 blocks; 7 free, watermark 2 -> no
    assert not bm.can_allocate(r)
    r2 = make_req("b", 16)  # 4 blocks; 7-4=3 >= 2 -> yes
    assert bm.can_allocate(r2)


def test_table_seq_changes_on_reassignment():
    """the runner's block-table cache keys on table_seq;
    swap_out+swap_in and free+reallocate must produce a NEW seq even when the
    table length is unchanged, so stale block ids are never replayed."""
    bm = BlockManager(num_blocks=16, block_size=4, num_cpu_blocks=8)
    r = make_req("a", 8)
    bm.allocate(r)
    s0 = bm.table_seq("a")
    assert s0 >= 0
    # same-length round trip through the host tier -> new ids, new seq
    bm.swap_out(r)
    assert bm.table_seq("a") == -1
    bm.swap_in(r)
    s1 = bm.table_seq("a")
    assert s1 != s0
    # recompute preemption: free + re-allocate (same length) -> new seq
    bm.free(r)
    assert bm.table_seq("a") == -1
    bm.allocate(r)
    assert bm.table_seq("a") not in (s0, s1)


def test_runner_bt_cache_invalidated_on_swap():
    """End-to-end shape of the ADVICE fix: _fill_pinned must pick up the new
    block ids after a same-length table reassignment."""
    import numpy as np
    from kserve_amd.engine.model_runner import ModelRunner

    bm = BlockManager(num_blocks=32, block_size=4, num_cpu_blocks=16)
    r = make_req("a", 8)
    bm.allocate(r)
    t0 = list(bm.get_block_table(r))

    class Batch:
        requests = [r]

    import types

    runner = ModelRunner.__new__(ModelRunner)
    runner._bt_cache = {}
    runner.config = types.SimpleNamespace(
        cache=types.SimpleNamespace(block_size=4)
    )

    captured = {}

    def fake_ensure(n, max_blocks):
        pnp = {
            "input_ids": np.zeros(n, dtype=np.int64),
            "positions": np.zeros(n, dtype=np.int64),
            "slot_mapping": np.zeros(n, dtype=np.int64),
            "context_lens": np.zeros(n, dtype=np.int32),
            "block_tables": np.zeros((n, max_blocks), dtype=np.int32),
        }
        captured["pnp"] = pnp
        return None, pnp

    runner._ensure_decode_buffers = fake_ensure
    r.num_computed_tokens = 7
    runner._fill_pinned(Batch(), bm)
    assert list(captured["pnp"]["block_tables"][0][: len(t0)]) == t0
    # same-length reassignment through the host tier; a filler request
    # grabs the just-freed blocks so swap_in lands on DIFFERENT ids
    bm.swap_out(r)
    filler = make_req("filler", 8)
    bm.allocate(filler)
    bm.swap_in(r)
    t1 = list(bm.get_block_table(r))
    assert len(t1) == len(t0) and t1 != t0
    runner._fill_pinned(Batch(), bm)
    assert list(captured["pnp"]["block_tables"][0][: len(t1)]) == t1
