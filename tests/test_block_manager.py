from hyperspot.engine.kv_cache import BlockManager


def test_allocate_free_roundtrip():
    bm = BlockManager(num_blocks=8, block_size=16)
    blocks = bm.allocate("a", 33)          # 3 blocks
    assert len(blocks) == 3
    assert bm.num_free == 5
    bm.free("a")
    assert bm.num_free == 8


def test_append_slot_crosses_block_boundary():
    bm = BlockManager(num_blocks=4, block_size=4)
    bm.allocate("a", 4)                    # exactly one block
    assert bm.num_free == 3
    slot = bm.append_slot("a")             # needs a fresh block
    assert bm.num_free == 2
    assert slot == bm.table("a")[1] * 4
    # fill the rest of that block: no new allocation
    for i in range(1, 4):
        bm.append_slot("a")
    assert bm.num_free == 2


def test_slot_of_matches_table():
    bm = BlockManager(num_blocks=8, block_size=4)
    bm.allocate("s", 10)
    t = bm.table("s")
    assert bm.slot_of("s", 0) == t[0] * 4
    assert bm.slot_of("s", 5) == t[1] * 4 + 1
    assert bm.slot_of("s", 9) == t[2] * 4 + 1


def test_can_allocate_watermark():
    bm = BlockManager(num_blocks=4, block_size=4)
    assert bm.can_allocate(16, watermark=0)
    assert not bm.can_allocate(16, watermark=1)
    assert bm.can_allocate(12, watermark=1)
