"""Prefix caching: hash-chained full prompt blocks are shared between
requests; cached prefixes skip recompute (the prefill chunk starts
mid-prompt) and evict LRU under pool pressure.  Off by default."""

import pytest

from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams


def _engine(**kw):
    kw.setdefault("num_gpu_blocks", 64)
    cfg = EngineConfig(model="tiny-llama", max_num_seqs=4,
                       max_num_batched_tokens=512, max_model_len=256,
                       enforce_eager=True, seed=9, **kw)
    return LLMEngine(cfg, device="cpu")


PROMPT = [(i * 13) % 400 + 3 for i in range(70)]   # 4 full blocks + tail


def test_cached_prefix_skips_recompute_and_matches():
    base = _engine()
    ref = base.generate([PROMPT], SamplingParams(temperature=0.0,
                                                 max_tokens=5))[0]
    eng = _engine(enable_prefix_caching=True)
    bm = eng.runner.block_manager
    out1 = eng.generate([PROMPT], SamplingParams(temperature=0.0,
                                                 max_tokens=5))[0]
    assert out1 == ref
    assert bm.cache_hits == 0
    # identical prompt again: 3 shareable blocks (last full block is
    # never shared) are reused, the chunk starts at token 48
    out2 = eng.generate([PROMPT], SamplingParams(temperature=0.0,
                                                 max_tokens=5))[0]
    assert out2 == ref
    assert bm.cache_hits == 1


def test_partial_prefix_reuse():
    eng = _engine(enable_prefix_caching=True)
    bm = eng.runner.block_manager
    sp = SamplingParams(temperature=0.0, max_tokens=4)
    a = PROMPT[:48] + [7, 8, 9, 10]
    b = PROMPT[:48] + [11, 12, 13, 14]
    ra = eng.generate([a], sp)[0]
    rb = eng.generate([b], sp)[0]
    assert bm.cache_hits == 1          # b reused a's shared 32-token prefix
    # same outputs as an uncached engine
    plain = _engine()
    assert ra == plain.generate([a], sp)[0]
    assert rb == plain.generate([b], sp)[0]


def test_cache_eviction_under_pressure():
    eng = _engine(enable_prefix_caching=True, num_gpu_blocks=24)
    sp = SamplingParams(temperature=0.0, max_tokens=2)
    # distinct prompts fill and roll the pool; evictable blocks recycle
    for i in range(8):
        p = [(i * 97 + j) % 300 + 3 for j in range(64)]
        out = eng.generate([p], sp)[0]
        assert len(out) == 2
    bm = eng.runner.block_manager
    assert bm.num_free > 0


def test_concurrent_shared_prefix_same_batch():
    """Two same-prefix prompts admitted in ONE prefill step must not read
    each other's half-written pages (hashes commit post-step)."""
    eng = _engine(enable_prefix_caching=True)
    sp = SamplingParams(temperature=0.0, max_tokens=3)
    a = PROMPT[:48] + [21]
    b = PROMPT[:48] + [22]
    outs = eng.generate([a, b], sp)
    plain = _engine()
    ref = plain.generate([a, b], sp)
    assert outs == ref


def test_preemption_with_caching_recovers():
    """Preempted victims re-admit through the caching path and may reuse
    their own committed prefix; outputs must match a roomy engine."""
    sp = SamplingParams(temperature=0.0, max_tokens=24)
    prompts = [[(s * 41 + i) % 350 + 3 for i in range(40)]
               for s in range(3)]
    roomy = _engine(enable_prefix_caching=True, num_gpu_blocks=256)
    ref = roomy.generate(prompts, sp)
    tight = _engine(enable_prefix_caching=True, num_gpu_blocks=20)
    out = tight.generate(prompts, sp)
    assert out == ref
