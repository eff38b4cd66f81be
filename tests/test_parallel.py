"""Tensor/expert-parallel correctness on CPU (gloo, world_size=2) — the
same code path RCCL takes on the 8-GPU node (backend string is the only
difference), so TP/EP must be correct by construction here."""

import os

import pytest
import torch
import torch.multiprocessing as mp


def _run_tp_linear(rank, world, port, results):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from hyperspot.parallel import (ColumnParallelLinear, RowParallelLinear)
    from hyperspot.parallel.state import (initialize_model_parallel,
                                          destroy_model_parallel)
    initialize_model_parallel(tp_size=world)
    torch.manual_seed(0)
    x = torch.randn(5, 64)
    col = ColumnParallelLinear(64, 32, torch.float32, seed_tag=7)
    row = RowParallelLinear(32, 48, torch.float32, seed_tag=9)
    y = row(col(x))
    results[rank] = y.detach()
    dist.barrier()
    destroy_model_parallel()
    dist.destroy_process_group()


def _reference_linear():
    from hyperspot.parallel.layers import _init_weight
    torch.manual_seed(0)
    x = torch.randn(5, 64)
    wc = _init_weight(32, 64, torch.float32, 7)
    wr = _init_weight(48, 32, torch.float32, 9)
    return x @ wc.t() @ wr.t()


def test_tp2_linear_matches_single_rank():
    mgr = mp.Manager()
    results = mgr.dict()
    port = 29611
    mp.spawn(_run_tp_linear, args=(2, port, results), nprocs=2,
             join=True)
    ref = _reference_linear()
    for r in (0, 1):
        assert torch.allclose(results[r], ref, atol=1e-5), r
    # both ranks agree bitwise after the all-reduce
    assert torch.equal(results[0], results[1])


def _run_tp_engine(rank, world, port, results):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams
    from hyperspot.parallel.state import (initialize_model_parallel,
                                          destroy_model_parallel)
    initialize_model_parallel(tp_size=world)
    cfg = EngineConfig(model="tiny-llama", max_num_seqs=4,
                       max_num_batched_tokens=256, max_model_len=128,
                       num_gpu_blocks=64, tp_size=world)
    eng = LLMEngine(cfg)
    out = eng.generate([[1, 2, 3, 4, 5], [7, 8, 9]],
                       SamplingParams(temperature=0.0, max_tokens=8))
    results[rank] = out
    dist.barrier()
    destroy_model_parallel()
    dist.destroy_process_group()


def test_tp2_engine_greedy_matches_tp1():
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_run_tp_engine, args=(2, 29613, results), nprocs=2, join=True)
    # SPMD: both ranks must produce identical tokens
    assert results[0] == results[1]
    from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams
    cfg = EngineConfig(model="tiny-llama", max_num_seqs=4,
                       max_num_batched_tokens=256, max_model_len=128,
                       num_gpu_blocks=64)
    ref = LLMEngine(cfg).generate([[1, 2, 3, 4, 5], [7, 8, 9]],
                                  SamplingParams(temperature=0.0,
                                                 max_tokens=8))
    assert results[0] == ref


def _run_ep_moe(rank, world, port, results):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from hyperspot.engine.config import get_model_spec
    from hyperspot.models.mixtral import MixtralMoE
    from hyperspot.parallel.state import (initialize_model_parallel,
                                          destroy_model_parallel)
    initialize_model_parallel(tp_size=1, ep_size=world)
    spec = get_model_spec("tiny-moe")
    moe = MixtralMoE(spec, layer_idx=0, dtype=torch.float32)
    torch.manual_seed(3)
    x = torch.randn(6, spec.hidden_size)
    y = moe(x)
    results[rank] = y.detach()
    dist.barrier()
    destroy_model_parallel()
    dist.destroy_process_group()


def test_ep2_moe_matches_ep1():
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_run_ep_moe, args=(2, 29615, results), nprocs=2, join=True)
    # single-rank reference
    import torch.distributed as dist
    from hyperspot.engine.config import get_model_spec
    from hyperspot.models.mixtral import MixtralMoE
    spec = get_model_spec("tiny-moe")
    moe = MixtralMoE(spec, layer_idx=0, dtype=torch.float32)
    torch.manual_seed(3)
    x = torch.randn(6, spec.hidden_size)
    ref = moe(x)
    for r in (0, 1):
        assert torch.allclose(results[r], ref, atol=1e-4), r


def _run_ep_moe_fp8(rank, world, port, results):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from hyperspot.engine.config import get_model_spec
    from hyperspot.models.mixtral import MixtralMoE
    from hyperspot.parallel import layers as L
    from hyperspot.parallel.state import (initialize_model_parallel,
                                          destroy_model_parallel)
    initialize_model_parallel(tp_size=1, ep_size=world)
    L.set_quant_mode("fp8")
    try:
        spec = get_model_spec("tiny-moe")
        moe = MixtralMoE(spec, layer_idx=0, dtype=torch.float32)
        torch.manual_seed(3)
        x = torch.randn(6, spec.hidden_size)
        y = moe(x)
        results[rank] = y.detach()
    finally:
        L.set_quant_mode(None)
    dist.barrier()
    destroy_model_parallel()
    dist.destroy_process_group()


def test_ep2_fp8_wire_matches_single_rank():
    """fp8 EP: e4m3fn activation bytes + scales over the all-to-all must
    reproduce the single-rank fp8 MoE (same quant model both sides)."""
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_run_ep_moe_fp8, args=(2, 29617, results), nprocs=2, join=True)
    from hyperspot.engine.config import get_model_spec
    from hyperspot.models.mixtral import MixtralMoE
    from hyperspot.parallel import layers as L
    L.set_quant_mode("fp8")
    try:
        spec = get_model_spec("tiny-moe")
        moe = MixtralMoE(spec, layer_idx=0, dtype=torch.float32)
        torch.manual_seed(3)
        x = torch.randn(6, spec.hidden_size)
        ref = moe(x)
    finally:
        L.set_quant_mode(None)
    for r in (0, 1):
        assert torch.allclose(results[r], ref, atol=2e-2, rtol=2e-2), \
            (results[r] - ref).abs().max()


def _run_ep_moe_static(rank, world, port, results, quant):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from hyperspot.engine.config import get_model_spec
    from hyperspot.models.mixtral import MixtralMoE
    from hyperspot.parallel import layers as L
    from hyperspot.parallel.state import (initialize_model_parallel,
                                          destroy_model_parallel)
    initialize_model_parallel(tp_size=1, ep_size=world)
    if quant:
        L.set_quant_mode(quant)
    try:
        spec = get_model_spec("tiny-moe")
        moe = MixtralMoE(spec, layer_idx=0, dtype=torch.float32)
        moe.capacity_factor = 8.0      # ample: nothing drops
        torch.manual_seed(3)
        x = torch.randn(6, spec.hidden_size)
        y_dyn = moe(x)
        moe.force_static_ep = True
        y_static = moe(x)
        results[rank] = (y_dyn.detach(), y_static.detach(),
                         int(moe.ep_overflow))
    finally:
        if quant:
            L.set_quant_mode(None)
    dist.barrier()
    destroy_model_parallel()
    dist.destroy_process_group()


@pytest.mark.parametrize("quant", [None, "fp8"])
def test_ep2_static_dispatch_matches_dynamic(quant):
    """Capture-safe fixed-capacity dispatch == exact dynamic dispatch when
    capacity is ample (the decode-graph path at ep>1)."""
    mgr = mp.Manager()
    results = mgr.dict()
    port = 29621 if quant is None else 29622
    mp.spawn(_run_ep_moe_static, args=(2, port, results, quant),
             nprocs=2, join=True)
    for r in (0, 1):
        y_dyn, y_static, overflow = results[r]
        assert overflow == 0
        assert torch.allclose(y_static, y_dyn, atol=1e-4), r


def _run_ep_moe_overflow(rank, world, port, results):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from hyperspot.engine.config import get_model_spec
    from hyperspot.models.mixtral import MixtralMoE
    from hyperspot.parallel.state import (initialize_model_parallel,
                                          destroy_model_parallel)
    initialize_model_parallel(tp_size=1, ep_size=world)
    spec = get_model_spec("tiny-moe")
    moe = MixtralMoE(spec, layer_idx=0, dtype=torch.float32)
    moe.force_static_ep = True
    moe.capacity_factor = 0.1          # starve the buckets on purpose
    torch.manual_seed(5)
    x = torch.randn(40, spec.hidden_size)
    y = moe(x)
    results[rank] = (bool(torch.isfinite(y).all()), int(moe.ep_overflow))
    dist.barrier()
    destroy_model_parallel()
    dist.destroy_process_group()


def test_ep_static_overflow_drops_are_counted():
    """Over-capacity top-k assignments are dropped (w=0), not corrupted,
    and the drop count is observable via the ep_overflow buffer."""
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_run_ep_moe_overflow, args=(2, 29623, results), nprocs=2,
             join=True)
    for r in (0, 1):
        finite, overflow = results[r]
        assert finite and overflow > 0, (r, results[r])


def _run_ep_ckpt(rank, world, port, results, full_ckpt, ep2_ckpt):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from hyperspot.engine.checkpoint import (load_checkpoint_into,
                                             save_checkpoint)
    from hyperspot.engine.config import get_model_spec
    from hyperspot.models.mixtral import MixtralMoE
    from hyperspot.parallel.state import (initialize_model_parallel,
                                          destroy_model_parallel)
    initialize_model_parallel(tp_size=1, ep_size=world)
    spec = get_model_spec("tiny-moe")
    moe = MixtralMoE(spec, layer_idx=0, dtype=torch.float32)
    # scatter: load the full-layout (ep=1-saved) checkpoint into shards
    load_checkpoint_into(moe, full_ckpt)
    torch.manual_seed(3)
    x = torch.randn(6, spec.hidden_size)
    results[rank] = moe(x).detach()
    # gather: save back out in full layout from the sharded deployment
    save_checkpoint(moe, ep2_ckpt)
    dist.barrier()
    destroy_model_parallel()
    dist.destroy_process_group()


def test_ep2_checkpoint_scatter_gather(tmp_path):
    """EP expert tensors round-trip through the FULL [E, ...] checkpoint
    layout: an ep=1 save loads into ep=2 shards (scatter) and an ep=2
    save reassembles the identical full tensors (gather)."""
    from safetensors import safe_open
    from hyperspot.engine.checkpoint import save_checkpoint
    from hyperspot.engine.config import get_model_spec
    from hyperspot.models.mixtral import MixtralMoE
    spec = get_model_spec("tiny-moe")
    ref = MixtralMoE(spec, layer_idx=0, dtype=torch.float32)
    with torch.no_grad():            # distinct weights so the load matters
        ref.w13 *= 1.5
        ref.w2 *= 0.5
        ref.router += 0.1
    full = str(tmp_path / "full.safetensors")
    ep2 = str(tmp_path / "ep2.safetensors")
    save_checkpoint(ref, full)
    torch.manual_seed(3)
    x = torch.randn(6, spec.hidden_size)
    want = ref(x).detach()
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_run_ep_ckpt, args=(2, 29625, results, full, ep2),
             nprocs=2, join=True)
    for r in (0, 1):
        assert torch.allclose(results[r], want, atol=1e-4), r
    with safe_open(ep2, framework="pt", device="cpu") as f:
        for k in ("w13", "w2", "router"):
            assert torch.equal(f.get_tensor(k),
                               dict(ref.named_parameters())[k].data), k


def _run_tp_guided(rank, world, port, results):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams
    from hyperspot.parallel.state import (initialize_model_parallel,
                                          destroy_model_parallel)
    initialize_model_parallel(tp_size=world)
    cfg = EngineConfig(model="tiny-llama", max_num_seqs=4,
                       max_num_batched_tokens=256, max_model_len=512,
                       num_gpu_blocks=128, tp_size=world)
    eng = LLMEngine(cfg, eos_token_id=2)
    schema = {"type": "object", "required": ["n", "ok"],
              "properties": {"n": {"type": "integer"},
                             "ok": {"type": "boolean"}}}
    out = eng.generate([[1, 2, 3]],
                       SamplingParams(temperature=1.0, max_tokens=200,
                                      seed=4, response_schema=schema))
    results[rank] = out
    dist.barrier()
    destroy_model_parallel()
    dist.destroy_process_group()


def test_tp2_guided_schema_identical_and_valid():
    """Guided masks are pure functions of the (identical) request state,
    so TP ranks stay in lockstep; the result obeys the schema."""
    import json
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_run_tp_guided, args=(2, 29627, results), nprocs=2,
             join=True)
    assert results[0] == results[1]
    toks = results[0][0]
    assert toks[-1] == 2, toks
    j = json.loads(bytes(t - 4 for t in toks if t != 2).decode(
        "utf-8", errors="replace"))
    assert set(j) == {"n", "ok"} and isinstance(j["n"], int)
