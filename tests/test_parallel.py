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
