"""Randomized invariant checking for BlockManager + prefix caching.

Invariant after every op: every physical block is accounted exactly once
across {free list} ∪ {evictable} ∪ {referenced-by-tables}, and a shared
block's refcount equals the number of tables containing it.
"""

import random

import pytest

from hyperspot.engine.kv_cache import BlockManager


def check_invariants(bm: BlockManager):
    in_tables = {}
    for seq, row in bm.row_of.items():
        nt = int(bm.ntables_np[row])
        for b in bm.tables_np[row, :nt]:
            in_tables.setdefault(int(b), []).append(seq)
    free = set(bm.free_blocks)
    evict = set(bm.evictable.keys())
    assert not (free & evict), "block both free and evictable"
    for b, seqs in in_tables.items():
        assert b not in free, f"block {b} in a table AND free"
        assert b not in evict, f"block {b} in a table AND evictable"
        h = bm.block_hash.get(b)
        if h is not None and b in bm.block_ref:
            assert bm.block_ref[b] == len(seqs), \
                f"refcount mismatch for {b}: {bm.block_ref[b]} vs {seqs}"
        else:
            assert len(seqs) == 1, f"unshared block {b} in {seqs}"
    # total accounting
    total = len(free) + len(evict) + len(in_tables)
    assert total == bm.num_blocks, (len(free), len(evict), len(in_tables))


@pytest.mark.parametrize("seed", [0, 1, 2, 3])
def test_random_ops_preserve_invariants(seed):
    rng = random.Random(seed)
    bm = BlockManager(48, 16, capacity=16, enable_prefix_caching=True)
    prompts = {}
    live = []
    ctr = 0
    # a few shared prompt families to force cache hits
    families = [[rng.randrange(500) for _ in range(80)] for _ in range(3)]
    for step in range(400):
        op = rng.random()
        if op < 0.4 and len(live) < 12:
            ctr += 1
            sid = f"s{ctr}"
            fam = rng.choice(families)
            cut = rng.randrange(17, len(fam))
            toks = fam[:cut] + [rng.randrange(500)
                                for _ in range(rng.randrange(0, 8))]
            try:
                cached = bm.allocate_with_prefix(sid, len(toks), toks)
            except RuntimeError:
                continue
            assert cached <= len(toks)
            prompts[sid] = toks
            bm.commit_hashes(sid, len(toks))
            live.append(sid)
        elif op < 0.7 and live:
            sid = rng.choice(live)
            try:
                bm.append_slot(sid)
            except RuntimeError:
                pass
        elif live:
            sid = live.pop(rng.randrange(len(live)))
            bm.free(sid)
            prompts.pop(sid, None)
        check_invariants(bm)
    for sid in live:
        bm.free(sid)
    check_invariants(bm)
    assert len(bm.free_blocks) + len(bm.evictable) == bm.num_blocks
