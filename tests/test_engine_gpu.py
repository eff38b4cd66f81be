"""GPU engine tests: end-to-end decode on the HIP kernel path, hipGraph
capture, and native-extension enforcement."""

import pytest
import torch

import hyperspot.ops as ops
from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams

pytestmark = pytest.mark.gpu


def _cfg(**kw):
    base = dict(model="tiny-llama", max_num_seqs=8,
                max_num_batched_tokens=512, max_model_len=256,
                num_gpu_blocks=128, enforce_eager=True)
    base.update(kw)
    return EngineConfig(**base)


def test_native_extension_loaded():
    assert ops.have_native(), "GPU run requires the in-tree gfx950 extension"


def test_generate_deterministic_on_gpu():
    eng = LLMEngine(_cfg())
    sp = SamplingParams(temperature=0.0, max_tokens=12)
    a = eng.generate([[1, 2, 3, 4, 5]], sp)[0]
    b = eng.generate([[1, 2, 3, 4, 5]], sp)[0]
    assert a == b
    assert len(a) == 12


def test_decode_consistent_with_prefill_recompute_gpu():
    """Greedy continuation via incremental paged decode must match feeding
    the grown prompt through prefill again (same kernels, same dtype)."""
    eng = LLMEngine(_cfg())
    sp1 = SamplingParams(temperature=0.0, max_tokens=6)
    prompt = [3, 7, 11, 13, 17]
    out = eng.generate([prompt], sp1)[0]
    # recompute: prompt + first k outputs, ask for 1 token
    for k in range(0, 5):
        nxt = eng.generate([prompt + out[:k]],
                           SamplingParams(temperature=0.0, max_tokens=1))[0]
        assert nxt[0] == out[k], (k, nxt, out)


def test_hipgraph_decode_matches_eager():
    sp = SamplingParams(temperature=0.0, max_tokens=10)
    prompts = [[1, 2, 3, 4], [9, 8, 7], [5, 5, 5, 5, 5]]
    eager = LLMEngine(_cfg(enforce_eager=True)).generate(prompts, sp)
    graphed_engine = LLMEngine(_cfg(enforce_eager=False,
                                    graph_batch_sizes=(1, 2, 4, 8)))
    graphed = graphed_engine.generate(prompts, sp)
    assert eager == graphed
    assert len(graphed_engine.runner._graphs) > 0, "graphs were not used"


def test_batched_gpu_greedy_equals_single():
    eng = LLMEngine(_cfg())
    sp = SamplingParams(temperature=0.0, max_tokens=5)
    p1, p2 = [2, 4, 6, 8], [1, 3, 5, 7, 9, 11]
    both = eng.generate([p1, p2], sp)
    assert both[0] == eng.generate([p1], sp)[0]
    assert both[1] == eng.generate([p2], sp)[0]


def test_mixtral_tiny_runs_on_gpu():
    eng = LLMEngine(_cfg(model="tiny-moe"))
    out = eng.generate([[1, 2, 3]], SamplingParams(temperature=0.0,
                                                   max_tokens=4))[0]
    assert len(out) == 4


@pytest.mark.gpu
def test_chunked_prefill_matches_unchunked_gpu():
    """Continuation chunks route through the paged attention kernel on GPU
    (fresh_prefill=False); greedy output must match whole-prompt prefill."""
    from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams
    prompt = [(i * 131) % 120000 + 2 for i in range(700)]
    outs = []
    for budget in (8192, 256):
        cfg = EngineConfig(model="llama3-8b", max_num_seqs=4,
                           max_num_batched_tokens=budget,
                           max_model_len=1024, num_gpu_blocks=256,
                           enforce_eager=True, seed=11)
        eng = LLMEngine(cfg, device="cuda:0")
        outs.append(eng.generate(
            [prompt], SamplingParams(temperature=0.0, max_tokens=5))[0])
        del eng
        import torch
        torch.cuda.empty_cache()
    assert outs[0] == outs[1], outs


@pytest.mark.gpu
def test_prefix_caching_matches_uncached_gpu():
    from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams
    prompt = [(i * 29) % 120000 + 2 for i in range(120)]
    sp = SamplingParams(temperature=0.0, max_tokens=4)
    outs = {}
    for flag in (False, True):
        cfg = EngineConfig(model="llama3-8b", max_num_seqs=4,
                           max_num_batched_tokens=2048, max_model_len=512,
                           num_gpu_blocks=256, enforce_eager=True, seed=2,
                           enable_prefix_caching=flag)
        eng = LLMEngine(cfg, device="cuda:0")
        a = eng.generate([prompt], sp)[0]
        b = eng.generate([prompt], sp)[0]   # second run hits the cache
        assert a == b
        if flag:
            assert eng.runner.block_manager.cache_hits == 1
        outs[flag] = a
        del eng
        import torch
        torch.cuda.empty_cache()
    assert outs[False] == outs[True]


@pytest.mark.gpu
def test_prefill_graph_matches_eager():
    """The padded single-request prefill graph (TTFT fast path) must be
    token-exact vs eager prefill — including a prompt length that is
    NOT a bucket size (padding correctness) and the KV it leaves behind
    (greedy continuation must also match)."""
    from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams
    prompts = [[7 * i % 311 + 5 for i in range(n)] for n in (100, 512)]
    outs = {}
    for eager in (True, False):
        cfg = EngineConfig(model="llama3-8b", max_num_seqs=4,
                           max_model_len=1024, num_gpu_blocks=512,
                           enforce_eager=eager, seed=3)
        eng = LLMEngine(cfg, device="cuda:0")
        if not eager:
            eng.capture_graphs()
        outs[eager] = eng.generate(
            prompts, SamplingParams(temperature=0.0, max_tokens=12))
        del eng
        torch.cuda.empty_cache()
    assert outs[True] == outs[False], outs


@pytest.mark.gpu
def test_guided_json_decode_gpu():
    """Guided mask on DEVICE logits (advanced indexing + -inf fill on
    the sampler path): grammar holds on hardware exactly as on CPU."""
    import json
    from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams
    from hyperspot.engine.guided import JsonByteMachine
    eng = LLMEngine(EngineConfig(model="tiny-llama", max_num_seqs=4,
                                 max_num_batched_tokens=256,
                                 max_model_len=512, num_gpu_blocks=128,
                                 seed=0), device="cuda:0")
    outs = eng.generate(
        [[1, 10, 11], [1, 12, 13]],
        SamplingParams(temperature=1.0, max_tokens=300, seed=7,
                       response_format="json"))
    for toks in outs:
        m = JsonByteMachine()
        body = [t for t in toks if t != 2]
        for t in body:
            assert 4 <= t < 260, toks
            m.feed(t - 4)
        if toks and toks[-1] == 2:
            json.loads(bytes(t - 4 for t in body).decode(
                "utf-8", errors="replace"))
