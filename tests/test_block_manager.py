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
