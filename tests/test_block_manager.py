from production_stack_amd.engine.block_manager import BlockManager
from production_stack_amd.engine.sampling import SamplingParams
from production_stack_amd.engine.sequence import Sequence


def mkseq(rid, tokens):
    return Sequence(rid, tokens, SamplingParams(max_tokens=4))


def test_alloc_and_free():
    bm = BlockManager(8, 16)
    s = mkseq("a", list(range(40)))  # 3 blocks
    assert bm.can_allocate_prompt(s)
    bm.allocate_prompt(s)
    assert len(s.block_table) == 3
    assert bm.num_free == 5
    bm.free_seq(s)
    assert bm.num_free == 8


def test_prefix_cache_hit_and_reuse():
    bm = BlockManager(16, 16)
    prompt = list(range(100, 150))  # 50 tokens -> 3 full + 1 partial
    s1 = mkseq("s1", prompt)
    bm.allocate_prompt(s1)
    s1.num_computed = 50
    bm.register_computed_blocks(s1)
    t1 = list(s1.block_table)
    bm.free_seq(s1)
    # same prompt again: 3 full blocks should hit
    s2 = mkseq("s2", prompt)
    bm.allocate_prompt(s2)
    assert s2.num_cached_prompt_tokens == 48
    assert s2.block_table[:3] == t1[:3]
    assert bm.prefix_hits == 3
    bm.free_seq(s2)


def test_prefix_cache_never_covers_whole_prompt():
    bm = BlockManager(16, 16)
    prompt = list(range(32))  # exactly 2 blocks
    s1 = mkseq("s1", prompt)
    bm.allocate_prompt(s1)
    s1.num_computed = 32
    bm.register_computed_blocks(s1)
    bm.free_seq(s1)
    s2 = mkseq("s2", prompt)
    bm.allocate_prompt(s2)
    # only 1 of the 2 blocks may be reused (last token must be computed)
    assert s2.num_cached_prompt_tokens == 16
    bm.free_seq(s2)


def test_eviction_lru():
    bm = BlockManager(4, 16)
    s1 = mkseq("s1", list(range(64)))  # uses all 4 blocks
    bm.allocate_prompt(s1)
    s1.num_computed = 64
    bm.register_computed_blocks(s1)
    bm.free_seq(s1)
    assert bm.num_free == 4
    assert len(bm.evictable) >= 3
    # new allocation must evict and still succeed
    s2 = mkseq("s2", list(range(1000, 1064)))
    bm.allocate_prompt(s2)
    assert len(s2.block_table) == 4


def test_shared_blocks_refcounted():
    bm = BlockManager(8, 16)
    prompt = list(range(48))
    s1 = mkseq("s1", prompt)
    bm.allocate_prompt(s1)
    s1.num_computed = 48
    bm.register_computed_blocks(s1)
    s2 = mkseq("s2", prompt)
    bm.allocate_prompt(s2)
    shared = s2.block_table[0]
    assert bm.ref_count[shared] == 2
    bm.free_seq(s1)
    assert bm.ref_count[shared] == 1
    bm.free_seq(s2)
    assert bm.ref_count[shared] == 0


def test_chain_hash_differs_on_prefix():
    bm = BlockManager(8, 16)
    a = bm.chain_hash(None, tuple(range(16)))
    b = bm.chain_hash(a, tuple(range(16)))
    assert a != b
