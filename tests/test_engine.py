"""Engine-level tests: continuous batching, paged decode vs full-context
recompute, preemption, finish conditions."""

import pytest
import torch

from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams
from hyperspot.engine.request import FinishReason
from hyperspot.models import build_model
from hyperspot.models.llama import ForwardMeta
from hyperspot.engine.kv_cache import allocate_kv_caches


def _engine(**kw):
    cfg = EngineConfig(model="tiny-llama", max_num_seqs=8,
                       max_num_batched_tokens=512, max_model_len=256,
                       num_gpu_blocks=128, **kw)
    return LLMEngine(cfg)


def _naive_greedy(model, prompt, n_steps, num_blocks=64, block_size=16):
    """Full-context recompute each step through the SAME model."""
    spec = model.spec
    toks = list(prompt)
    for _ in range(n_steps):
        T = len(toks)
        caches = allocate_kv_caches(spec.num_layers, num_blocks,
                                    spec.num_kv_heads, block_size,
                                    spec.head_dim, model.dtype, "cpu")
        meta = ForwardMeta(
            mode="prefill",
            positions=torch.arange(T),
            slot_mapping=torch.arange(T),   # identity paging
            seq_start=torch.tensor([0, T], dtype=torch.int32),
            max_seqlen=T,
            logits_indices=torch.tensor([T - 1]),
        )
        logits = model(torch.tensor(toks), meta, caches)
        toks.append(int(logits[0].float().argmax()))
    return toks[len(prompt):]


def test_incremental_decode_matches_full_recompute():
    """The paged decode path must produce the same greedy continuation as
    recomputing the whole sequence through the prefill path each step."""
    eng = _engine()
    prompt = [3, 141, 59, 26, 535 % 512, 89, 79]
    out = eng.generate([prompt], SamplingParams(temperature=0.0, max_tokens=10))[0]
    ref = _naive_greedy(eng.runner.model, prompt, 10)
    assert out == ref


def test_batched_greedy_equals_single():
    eng = _engine()
    p1, p2 = [1, 2, 3, 4], [10, 20, 30, 40, 50, 60]
    both = eng.generate([p1, p2], SamplingParams(temperature=0.0, max_tokens=6))
    solo1 = eng.generate([p1], SamplingParams(temperature=0.0, max_tokens=6))[0]
    solo2 = eng.generate([p2], SamplingParams(temperature=0.0, max_tokens=6))[0]
    assert both[0] == solo1
    assert both[1] == solo2


def test_max_tokens_and_finish_reason():
    eng = _engine()
    rid = eng.add_request([1, 2, 3], SamplingParams(temperature=0.0, max_tokens=4))
    finished = {}
    while eng.has_work():
        for o in eng.step():
            if o.finished:
                finished[o.request_id] = o.finish_reason
    assert finished[rid] == FinishReason.LENGTH


def test_stop_token_finishes():
    eng = _engine()
    # discover the first greedy token, then use it as a stop token
    first = eng.generate([[5, 6, 7]], SamplingParams(temperature=0.0, max_tokens=1))[0][0]
    rid = eng.add_request([5, 6, 7], SamplingParams(
        temperature=0.0, max_tokens=64, stop_token_ids=(first,)))
    reasons = {}
    toks = []
    while eng.has_work():
        for o in eng.step():
            toks.append(o.token_id)
            if o.finished:
                reasons[o.request_id] = o.finish_reason
    assert reasons[rid] == FinishReason.STOP
    assert toks == [first]


def test_preemption_recompute_consistency():
    """Starve the pool so a sequence is preempted mid-decode; its final
    output must equal the unpreempted run (recompute preemption)."""
    cfg = EngineConfig(model="tiny-llama", max_num_seqs=4,
                       max_num_batched_tokens=512, max_model_len=256,
                       num_gpu_blocks=9)   # tiny pool: 9 blocks of 16 tokens
    eng = LLMEngine(cfg)
    p1 = list(range(1, 31))    # 30 tokens -> 2 blocks
    p2 = list(range(40, 70))   # 30 tokens -> 2 blocks
    sp = SamplingParams(temperature=0.0, max_tokens=40)
    out = eng.generate([p1, p2], sp)
    big = LLMEngine(EngineConfig(model="tiny-llama", max_num_seqs=4,
                                 max_num_batched_tokens=512,
                                 max_model_len=256, num_gpu_blocks=128))
    ref1 = big.generate([p1], sp)[0]
    ref2 = big.generate([p2], sp)[0]
    assert out[0] == ref1
    assert out[1] == ref2


def test_continuous_batching_admits_late_request():
    eng = _engine()
    r1 = eng.add_request([1, 2, 3], SamplingParams(temperature=0.0, max_tokens=12))
    results = {r1: []}
    for _ in range(3):
        for o in eng.step():
            results[o.request_id].append(o.token_id)
    r2 = eng.add_request([9, 8, 7], SamplingParams(temperature=0.0, max_tokens=3))
    results[r2] = []
    while eng.has_work():
        for o in eng.step():
            results[o.request_id].append(o.token_id)
    assert len(results[r1]) == 12
    assert len(results[r2]) == 3
    # late request unaffected by batching
    solo = eng.generate([[9, 8, 7]], SamplingParams(temperature=0.0, max_tokens=3))[0]
    assert results[r2] == solo


def test_seeded_sampling_deterministic():
    e1 = _engine(seed=7)
    e2 = _engine(seed=7)
    sp = SamplingParams(temperature=0.9, top_p=0.95, top_k=40, max_tokens=8)
    assert e1.generate([[1, 2, 3]], sp) == e2.generate([[1, 2, 3]], sp)


def test_moe_engine_runs():
    cfg = EngineConfig(model="tiny-moe", max_num_seqs=4,
                       max_num_batched_tokens=256, max_model_len=128,
                       num_gpu_blocks=64)
    eng = LLMEngine(cfg)
    out = eng.generate([[1, 2, 3, 4]], SamplingParams(temperature=0.0,
                                                      max_tokens=5))[0]
    assert len(out) == 5
    # deterministic
    out2 = eng.generate([[1, 2, 3, 4]], SamplingParams(temperature=0.0,
                                                       max_tokens=5))[0]
    assert out == out2


def test_long_prompt_exceeding_step_budget_is_not_starved():
    """A prompt longer than max_num_batched_tokens must still be served
    (scheduled alone), not wait forever."""
    from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams
    cfg = EngineConfig(model="tiny-llama", max_num_seqs=4,
                       max_num_batched_tokens=64, max_model_len=512,
                       num_gpu_blocks=64, enforce_eager=True)
    eng = LLMEngine(cfg, device="cpu")
    long_prompt = list(range(1, 200))          # 199 tokens > 64 budget
    out = eng.generate([long_prompt, [1, 2, 3]],
                       SamplingParams(temperature=0.0, max_tokens=4))
    assert len(out[0]) == 4 and len(out[1]) == 4


def test_chunked_prefill_matches_unchunked():
    """A 199-token prompt through a 64-token/step budget (4 chunks, the
    continuation chunks attending over the paged cache) must produce the
    same greedy tokens as whole-prompt prefill."""
    from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams
    prompt = [(i * 37) % 500 + 1 for i in range(199)]
    outs = []
    for budget in (512, 64):
        cfg = EngineConfig(model="tiny-llama", max_num_seqs=4,
                           max_num_batched_tokens=budget, max_model_len=512,
                           num_gpu_blocks=64, enforce_eager=True, seed=5)
        eng = LLMEngine(cfg, device="cpu")
        outs.append(eng.generate(
            [prompt], SamplingParams(temperature=0.0, max_tokens=6))[0])
    assert outs[0] == outs[1], outs


def test_chunked_prefill_interleaves_short_prompts():
    """Short prompts keep flowing while a long prompt chunks through."""
    from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams
    cfg = EngineConfig(model="tiny-llama", max_num_seqs=4,
                       max_num_batched_tokens=48, max_model_len=512,
                       num_gpu_blocks=96, enforce_eager=True)
    eng = LLMEngine(cfg, device="cpu")
    long_p = [(i % 90) + 2 for i in range(150)]
    out = eng.generate([long_p, [5, 6, 7], [9, 10]],
                       SamplingParams(temperature=0.0, max_tokens=4))
    assert all(len(o) == 4 for o in out), out


def test_abort_mid_chunked_prefill_frees_pages():
    from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams
    cfg = EngineConfig(model="tiny-llama", max_num_seqs=4,
                       max_num_batched_tokens=32, max_model_len=512,
                       num_gpu_blocks=64, enforce_eager=True)
    eng = LLMEngine(cfg, device="cpu")
    bm = eng.runner.block_manager
    free0 = bm.num_free
    rid = eng.add_request(list(range(2, 150)),
                          SamplingParams(temperature=0.0, max_tokens=4))
    eng.step()                       # first 32-token chunk only
    assert bm.num_free < free0
    eng.abort_request(rid)
    assert bm.num_free == free0      # pages reclaimed, no leak
    assert not eng.has_work()


def test_per_request_sampling_seed_reproducible():
    from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams

    def run(seed):
        cfg = EngineConfig(model="tiny-llama", max_num_seqs=2,
                           max_num_batched_tokens=128, max_model_len=64,
                           num_gpu_blocks=32, enforce_eager=True, seed=0)
        eng = LLMEngine(cfg, device="cpu")
        return eng.generate([[3, 4, 5]],
                            SamplingParams(temperature=1.0, top_k=50,
                                           max_tokens=8, seed=seed))[0]
    a1, a2 = run(123), run(123)
    b = run(321)
    assert a1 == a2          # same request seed => identical draws
    assert a1 != b           # different seed diverges (w.h.p.)
