"""fp8 (e4m3fn) weight-quant path — BASELINE config 5.

CPU tests emulate _scaled_mm numerics in fp32 (same quant model); the GPU
test (marked) runs the real hipBLASLt fp8 MFMA path and the engine e2e.
Reference contract: modules/model-registry PRD `format` field and
BASELINE.json config 5 (fp8 checkpoints, hot-swap).
"""

import pytest
import torch

from hyperspot.parallel import layers as L


@pytest.fixture(autouse=True)
def _reset_quant():
    yield
    L.set_quant_mode(None)


def test_weight_quant_roundtrip():
    w = torch.randn(64, 128) * 0.02
    q, s = L.quantize_weight_fp8(w)
    assert q.dtype == torch.float8_e4m3fn and s.shape == (64,)
    back = q.float() * s[:, None]
    rel = (back - w.float()).abs().max() / w.abs().max()
    assert rel < 0.05, rel.item()


def test_act_quant_rowwise():
    x = torch.randn(16, 128)
    q, s = L.quant_fp8_rowwise(x)
    back = q.float() * s[:, None]
    assert (back - x).abs().max() < 0.1 * x.abs().max()


def test_fp8_linear_matches_bf16_cpu():
    torch.manual_seed(0)
    L.set_quant_mode(None)
    lin = L.ColumnParallelLinear(256, 128, torch.float32, seed_tag=7)
    L.set_quant_mode("fp8")
    lin8 = L.ColumnParallelLinear(256, 128, torch.float32, seed_tag=7)
    assert lin8.quant == "fp8" and lin8.weight.dtype == torch.float8_e4m3fn
    x = torch.randn(8, 256)
    y, y8 = lin(x), lin8(x)
    rel = (y - y8).float().norm() / y.float().norm()
    assert rel < 0.06, rel.item()   # K=256: fp8 w+act quant noise ~4%


def test_fp8_engine_generates_cpu():
    from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams
    cfg = EngineConfig(model="tiny-llama", max_num_seqs=2,
                       max_num_batched_tokens=256, max_model_len=128,
                       num_gpu_blocks=64, enforce_eager=True, quant="fp8")
    eng = LLMEngine(cfg, device="cpu")
    out = eng.generate([[1, 2, 3, 4]],
                       SamplingParams(temperature=0.0, max_tokens=6))[0]
    assert len(out) == 6


def test_fp8_checkpoint_roundtrip(tmp_path):
    """fp8 state dict saves/loads; hot-swap copies into resident params."""
    from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams
    from hyperspot.engine.checkpoint import (load_checkpoint_into,
                                             save_checkpoint)
    cfg = EngineConfig(model="tiny-llama", max_num_seqs=2,
                       max_num_batched_tokens=256, max_model_len=128,
                       num_gpu_blocks=64, enforce_eager=True, quant="fp8",
                       seed=3)
    eng = LLMEngine(cfg, device="cpu")
    path = str(tmp_path / "ck.safetensors")
    save_checkpoint(eng.runner.model, path, meta={"format": "fp8"})
    before = eng.generate([[5, 6, 7]],
                          SamplingParams(temperature=0.0, max_tokens=4))[0]
    # scribble on a weight, swap back, behavior must be restored
    lin = eng.runner.model.layers[0].mlp.down
    lin.weight_scale.data.mul_(3.0)
    load_checkpoint_into(eng.runner.model, path)
    after = eng.generate([[5, 6, 7]],
                         SamplingParams(temperature=0.0, max_tokens=4))[0]
    assert before == after


@pytest.mark.gpu
def test_fp8_scaled_mm_matches_bf16_gpu():
    dev = "cuda:0"
    torch.manual_seed(0)
    L.set_quant_mode(None)
    L.set_init_device(dev)
    lin = L.ColumnParallelLinear(4096, 1024, torch.bfloat16, seed_tag=9)
    L.set_quant_mode("fp8")
    lin8 = L.ColumnParallelLinear(4096, 1024, torch.bfloat16, seed_tag=9)
    L.set_init_device("cpu")
    x = (torch.randn(64, 4096, device=dev) * 0.5).to(torch.bfloat16)
    y = lin.forward(x)
    y8 = lin8.forward(x)
    rel = (y - y8).float().norm() / y.float().norm()
    assert rel < 0.06, rel.item()   # fp8 w+act noise ~3.8% measured


@pytest.mark.gpu
def test_fp8_engine_generates_gpu():
    from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams
    cfg = EngineConfig(model="llama3-8b", max_num_seqs=4,
                       max_num_batched_tokens=2048, max_model_len=512,
                       num_gpu_blocks=256, enforce_eager=True, quant="fp8")
    eng = LLMEngine(cfg, device="cuda:0")
    out = eng.generate([[1, 2, 3, 4, 5, 6, 7, 8]],
                       SamplingParams(temperature=0.0, max_tokens=8))[0]
    assert len(out) == 8


def test_fp8_kv_engine_cpu():
    """fp8 KV cache mode: CPU reference path dequantizes transparently."""
    from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams
    cfg = EngineConfig(model="tiny-llama", max_num_seqs=2,
                       max_num_batched_tokens=256, max_model_len=128,
                       num_gpu_blocks=64, enforce_eager=True,
                       kv_dtype="fp8")
    eng = LLMEngine(cfg, device="cpu")
    assert eng.runner.kv_caches[0].dtype == torch.float8_e4m3fn
    out = eng.generate([[1, 2, 3, 4]],
                       SamplingParams(temperature=0.0, max_tokens=6))[0]
    assert len(out) == 6


@pytest.mark.gpu
def test_fp8_kv_decode_matches_dequant_ref_gpu():
    """GPU fp8-KV decode kernel vs fp32 reference over the SAME quantized
    cache values — error must be kernel-math-only, not quant noise."""
    import hyperspot.ops as ops
    from hyperspot.ops import torch_ref
    torch.manual_seed(0)
    dev = "cuda:0"
    bs, ctx, kvh, group, D, BS = 32, 96, 8, 4, 128, 16
    nblk = ctx // BS
    kc = (torch.randn(bs * nblk + 3, kvh, BS, D, device=dev)
          ).to(torch.float8_e4m3fn)
    vc = (torch.randn(bs * nblk + 3, kvh, BS, D, device=dev) * 0.5
          ).to(torch.float8_e4m3fn)
    q = (torch.randn(bs, kvh * group, D, device=dev) * 0.5).bfloat16()
    bt = torch.randperm(bs * nblk, device=dev, dtype=torch.int32
                        ).view(bs, nblk).contiguous()
    lens = torch.full((bs,), ctx, dtype=torch.int32, device=dev)
    out = ops.paged_attn_decode(q, kc, vc, bt, lens, D ** -0.5)
    ref = torch_ref.paged_attn_decode(q.float().cpu(), kc.float().cpu(),
                                      vc.float().cpu(), bt.cpu(),
                                      lens.cpu(), D ** -0.5)
    err = (out.float().cpu() - ref).abs().max().item()
    assert err < 2e-2, err


@pytest.mark.gpu
def test_fp8_kv_engine_gpu():
    from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams
    cfg = EngineConfig(model="llama3-8b", max_num_seqs=4,
                       max_num_batched_tokens=2048, max_model_len=512,
                       num_gpu_blocks=256, enforce_eager=True,
                       kv_dtype="fp8")
    eng = LLMEngine(cfg, device="cuda:0")
    out = eng.generate([[1, 2, 3, 4, 5, 6, 7, 8]],
                       SamplingParams(temperature=0.0, max_tokens=8))[0]
    assert len(out) == 8


def test_fp8_moe_engine_cpu():
    """Mixtral + fp8 mode on CPU exercises the dequant fallback path."""
    from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams
    cfg = EngineConfig(model="tiny-moe", max_num_seqs=2,
                       max_num_batched_tokens=256, max_model_len=128,
                       num_gpu_blocks=64, enforce_eager=True, quant="fp8")
    eng = LLMEngine(cfg, device="cpu")
    out = eng.generate([[1, 2, 3, 4]],
                       SamplingParams(temperature=0.0, max_tokens=5))[0]
    assert len(out) == 5


@pytest.mark.gpu
def test_fp8_moe_grouped_gemm_matches_ref_gpu():
    """fp8 grouped MFMA GEMM vs the dequantized bf16 reference."""
    import hyperspot.ops as ops
    torch.manual_seed(3)
    dev = "cuda:0"
    T, H, I, E, K = 64, 256, 256, 4, 2
    x = (torch.randn(T, H, device=dev) * 0.3).to(torch.bfloat16)
    w13 = (torch.randn(E, 2 * I, H, device=dev) * 0.05).to(torch.bfloat16)
    w2 = (torch.randn(E, H, I, device=dev) * 0.05).to(torch.bfloat16)
    logits = torch.randn(T, E, device=dev)
    wts, ids = logits.softmax(-1).topk(K, dim=-1)
    wts = (wts / wts.sum(-1, keepdim=True)).float()

    q13 = [L.quantize_weight_fp8(w13[e]) for e in range(E)]
    q2 = [L.quantize_weight_fp8(w2[e]) for e in range(E)]
    w13q = torch.stack([q for q, _ in q13])
    w13s = torch.stack([s for _, s in q13])
    w2q = torch.stack([q for q, _ in q2])
    w2s = torch.stack([s for _, s in q2])
    xq = ops.quant_fp8(x)
    out8 = ops.moe_ffn_fp8(xq, w13q, w13s, w2q, w2s, wts,
                           ids.to(torch.int32))
    ref = ops.moe_ffn(x, w13, w2, wts, ids.to(torch.int32))
    rel = (out8.float() - ref.float()).norm() / ref.float().norm()
    assert rel < 0.08, rel.item()


@pytest.mark.gpu
def test_fp8_moe_engine_gpu():
    from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams
    cfg = EngineConfig(model="tiny-moe", max_num_seqs=4,
                       max_num_batched_tokens=512, max_model_len=256,
                       num_gpu_blocks=128, enforce_eager=True, quant="fp8")
    eng = LLMEngine(cfg, device="cuda:0")
    out = eng.generate([[1, 2, 3, 4, 5]],
                       SamplingParams(temperature=0.0, max_tokens=6))[0]
    assert len(out) == 6
