"""Checkpoint save/load + live weight hot-swap.

BASELINE config 5: model-registry hot-swaps two checkpoints from
file-storage without dropping the paged KV pool.  Checkpoints are
safetensors files (same arch, possibly different weights/precision); a swap
copies the new tensors into the resident parameters in place — the KV pool,
hipGraphs and allocator state survive because no parameter storage is
reallocated.

Checkpoints are always stored in the FULL (tp=1) layout.  Under tensor
parallelism each sharded parameter carries `_tp_slices` metadata
(parallel/layers.py `_finalize_weight`): [(full_start, length,
shard_start)] along `_tp_shard_dim`.  Load slices the full tensor into
this rank's shard; save all-gathers the shards over the TP group and
rank 0 reassembles + writes — so tp=1 and tp=8 deployments exchange the
same checkpoint files (the reference's file-storage serves one artifact
per model, modules/file-storage/docs/PRD.md:5-29).
"""

from __future__ import annotations

import json
import os
import time
from typing import Dict

import torch


def _params_and_buffers(model) -> Dict[str, torch.Tensor]:
    """named_parameters + buffers, keeping the live objects so the
    `_tp_*` shard attributes survive (state_dict() detaches and drops
    python attributes)."""
    out = dict(model.named_parameters())
    for k, v in model.named_buffers():
        out.setdefault(k, v)
    return out


def _shard_from_full(param: torch.Tensor, full: torch.Tensor) -> None:
    dim = param._tp_shard_dim
    for full_start, length, shard_start in param._tp_slices:
        src = full.narrow(dim, full_start, length)
        dst = param.data.narrow(dim, shard_start, length)
        dst.copy_(src.to(device=dst.device, dtype=dst.dtype,
                         non_blocking=True))


def _shard_group_of(param: torch.Tensor):
    """(group, my_rank, size) the parameter is sharded over: the TP group
    for the linear layers, the EP group for MoE expert tensors (which
    carry `_shard_group == "ep"`; mixtral.py block-shards whole experts
    along dim 0 with the same slice metadata)."""
    from hyperspot.parallel.state import (get_ep_group, get_ep_rank,
                                          get_ep_size, get_tp_group,
                                          get_tp_rank, get_tp_size)
    if getattr(param, "_shard_group", "tp") == "ep":
        return get_ep_group(), get_ep_rank(), get_ep_size()
    return get_tp_group(), get_tp_rank(), get_tp_size()


def _full_from_shards(param: torch.Tensor, group) -> torch.Tensor:
    """All-gather this parameter's shards over its shard group and rebuild
    the full tensor (every rank computes it; only rank 0 writes)."""
    import torch.distributed as dist
    world = dist.get_world_size(group)
    local = param.data.contiguous().cpu()
    gathered = [torch.empty_like(local) for _ in range(world)]
    # gloo/rccl gather on the device the backend supports; shards are
    # identical shape by construction
    if param.data.is_cuda:
        dev_g = [torch.empty_like(param.data) for _ in range(world)]
        dist.all_gather(dev_g, param.data.contiguous(), group=group)
        gathered = [t.cpu() for t in dev_g]
    else:
        dist.all_gather(gathered, local, group=group)
    full = torch.empty(param._tp_full_shape, dtype=local.dtype)
    dim = param._tp_shard_dim
    for r, shard in enumerate(gathered):
        # every rank r applied the same slice rule with its own rank id;
        # reconstruct by re-deriving r's slices from the uniform pattern:
        # slices are (full_start(r), length, shard_start) where
        # full_start varies with r by length*1 within each section.
        for (f0, ln, s0), (rf0, _, _) in zip(
                _slices_for_rank(param, r), param._tp_slices):
            full.narrow(dim, f0, ln).copy_(shard.narrow(dim, s0, ln))
    return full


def _slices_for_rank(param: torch.Tensor, r: int):
    """Derive rank r's slices from this rank's slice pattern.  Within
    each section the full_start is section_base + r*length; section_base
    = full_start - my_rank*length."""
    _, me, _ = _shard_group_of(param)
    out = []
    for f0, ln, s0 in param._tp_slices:
        base = f0 - me * ln
        out.append((base + r * ln, ln, s0))
    return out


def save_checkpoint(model, path: str, meta: Dict = None) -> None:
    """Write the FULL-layout checkpoint.  Collective under TP (all ranks
    must call); only rank 0 writes the file."""
    import torch.distributed as dist
    from safetensors.torch import save_file
    state = {}
    for k, v in _params_and_buffers(model).items():
        if k.endswith("cos_sin") or k.endswith("ep_overflow"):
            continue
        if getattr(v, "_tp_slices", None) is not None:
            group, _, size = _shard_group_of(v)
            state[k] = (_full_from_shards(v, group) if size > 1
                        else v.detach().contiguous().cpu())
        else:
            state[k] = v.detach().contiguous().cpu()
    if dist.is_initialized() and dist.get_rank() != 0:
        return
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    save_file(state, path, metadata={"hyperspot": json.dumps(meta or {})})


def load_checkpoint_into(model, path: str) -> float:
    """In-place weight swap; returns wall seconds.  Parameter storages are
    reused (copy_), so captured hipGraphs remain valid.  Full-layout
    checkpoints load into TP shards via the `_tp_slices` metadata."""
    from safetensors import safe_open
    t0 = time.monotonic()
    device = next(model.parameters()).device
    params = _params_and_buffers(model)
    with safe_open(path, framework="pt", device="cpu") as f:
        keys = set(f.keys())
        missing = [k for k in params if k not in keys
                   and not k.endswith(("cos_sin", "ep_overflow"))]
        if missing:
            raise ValueError(f"checkpoint misses keys: {missing[:5]}...")
        with torch.inference_mode():
            for k, p in params.items():
                if k not in keys:
                    continue
                t = f.get_tensor(k)
                if (getattr(p, "_tp_slices", None) is not None
                        and tuple(t.shape) == tuple(p._tp_full_shape)
                        and tuple(t.shape) != tuple(p.shape)):
                    _shard_from_full(p, t)
                elif tuple(t.shape) == tuple(p.shape):
                    p.data.copy_(t.to(device=device, dtype=p.dtype,
                                      non_blocking=True))
                else:
                    raise ValueError(
                        f"shape mismatch for {k}: checkpoint "
                        f"{tuple(t.shape)} vs param {tuple(p.shape)}")
    if device.type == "cuda":
        torch.cuda.synchronize(device)
    return time.monotonic() - t0
