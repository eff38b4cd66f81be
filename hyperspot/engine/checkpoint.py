"""Checkpoint save/load + live weight hot-swap.

BASELINE config 5: model-registry hot-swaps two checkpoints from
file-storage without dropping the paged KV pool.  Checkpoints are
safetensors files (same arch, possibly different weights/precision); a swap
copies the new tensors into the resident parameters in place — the KV pool,
hipGraphs and allocator state survive because no parameter storage is
reallocated.
"""

from __future__ import annotations

import json
import os
import time
from typing import Dict

import torch


def save_checkpoint(model, path: str, meta: Dict = None) -> None:
    from safetensors.torch import save_file
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    state = {k: v.detach().contiguous() for k, v in model.state_dict().items()}
    save_file(state, path, metadata={"hyperspot": json.dumps(meta or {})})


def load_checkpoint_into(model, path: str) -> float:
    """In-place weight swap; returns wall seconds.  Parameter storages are
    reused (copy_), so captured hipGraphs remain valid."""
    from safetensors import safe_open
    t0 = time.monotonic()
    device = next(model.parameters()).device
    sd = model.state_dict()
    with safe_open(path, framework="pt", device="cpu") as f:
        keys = set(f.keys())
        missing = [k for k in sd if k not in keys and not k.endswith("cos_sin")]
        if missing:
            raise ValueError(f"checkpoint misses keys: {missing[:5]}...")
        with torch.inference_mode():
            for k in sd:
                if k not in keys:
                    continue
                t = f.get_tensor(k)
                sd[k].copy_(t.to(device=device, dtype=sd[k].dtype,
                                 non_blocking=True))
    if device.type == "cuda":
        torch.cuda.synchronize(device)
    return time.monotonic() - t0
