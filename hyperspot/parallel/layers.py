"""Tensor-parallel linear layers.

GEMMs go through torch.matmul → hipBLASLt on ROCm (library GEMMs); the
hand-written kernels cover the fused non-GEMM hot ops (csrc/).  Sharding:

  - ColumnParallelLinear: weight [out/tp, in], output stays sharded
  - RowParallelLinear:    weight [out, in/tp], all-reduce on the output
  - QKVParallelLinear:    heads sharded; q|k|v fused in one GEMM
"""

from __future__ import annotations

import torch
from torch import nn

from .state import (get_tp_rank, get_tp_size,
                    tensor_model_parallel_all_reduce)


_INIT_DEVICE = "cpu"


def set_init_device(device) -> None:
    """Where random-init weights are generated (GPU init is ~20x faster for
    the 8B/70B shapes; no checkpoints exist in this offline environment)."""
    global _INIT_DEVICE
    _INIT_DEVICE = device


def _init_weight(out_f: int, in_f: int, dtype: torch.dtype, seed_tag: int):
    # Random init; deterministic per-(shape, tag, device type) so every TP
    # rank materialises the same full tensor before sharding.
    seed = (seed_tag * 1000003 + out_f * 131 + in_f) % (2**31)
    dev = torch.device(_INIT_DEVICE)
    g = torch.Generator(device=dev).manual_seed(seed)
    w = torch.empty(out_f, in_f, dtype=torch.float32, device=dev)
    w.normal_(0.0, 0.02, generator=g)
    return w.to(dtype)


class ColumnParallelLinear(nn.Module):
    def __init__(self, in_features: int, out_features: int,
                 dtype: torch.dtype, seed_tag: int = 0):
        super().__init__()
        tp, r = get_tp_size(), get_tp_rank()
        assert out_features % tp == 0, (out_features, tp)
        self.in_features, self.out_features = in_features, out_features
        shard = out_features // tp
        full = _init_weight(out_features, in_features, dtype, seed_tag)
        self.weight = nn.Parameter(full[r * shard:(r + 1) * shard], requires_grad=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return x @ self.weight.t()

    def load_full_weight(self, w: torch.Tensor) -> None:
        tp, r = get_tp_size(), get_tp_rank()
        shard = self.out_features // tp
        self.weight.data.copy_(w[r * shard:(r + 1) * shard])


class MergedColumnParallelLinear(nn.Module):
    """Several column-parallel projections fused into one GEMM
    (gate_proj | up_proj).  Each sub-projection is sharded independently so
    the shard layout matches per-projection splits downstream."""

    def __init__(self, in_features: int, out_sizes: list, dtype: torch.dtype,
                 seed_tag: int = 0):
        super().__init__()
        tp, r = get_tp_size(), get_tp_rank()
        self.out_sizes = out_sizes
        parts = []
        for i, o in enumerate(out_sizes):
            assert o % tp == 0
            full = _init_weight(o, in_features, dtype, seed_tag + i)
            sh = o // tp
            parts.append(full[r * sh:(r + 1) * sh])
        self.weight = nn.Parameter(torch.cat(parts, 0), requires_grad=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return x @ self.weight.t()


class QKVParallelLinear(nn.Module):
    def __init__(self, hidden: int, head_dim: int, num_heads: int,
                 num_kv_heads: int, dtype: torch.dtype, seed_tag: int = 0):
        super().__init__()
        tp, r = get_tp_size(), get_tp_rank()
        assert num_heads % tp == 0 and num_kv_heads % tp == 0, \
            (num_heads, num_kv_heads, tp)
        self.nh, self.nkv = num_heads // tp, num_kv_heads // tp
        self.head_dim = head_dim
        qf = _init_weight(num_heads * head_dim, hidden, dtype, seed_tag)
        kf = _init_weight(num_kv_heads * head_dim, hidden, dtype, seed_tag + 1)
        vf = _init_weight(num_kv_heads * head_dim, hidden, dtype, seed_tag + 2)
        qs, ks = self.nh * head_dim, self.nkv * head_dim
        self.weight = nn.Parameter(torch.cat([
            qf[r * qs:(r + 1) * qs], kf[r * ks:(r + 1) * ks],
            vf[r * ks:(r + 1) * ks]], 0), requires_grad=False)

    def forward(self, x: torch.Tensor):
        """Returns strided [T, n, D] views into ONE fused qkv buffer — no
        .contiguous() copies; the HIP kernels take the token stride."""
        qkv = x @ self.weight.t()
        qs, ks = self.nh * self.head_dim, self.nkv * self.head_dim
        T = x.shape[0]
        d = self.head_dim
        q = qkv[:, :qs].view(T, self.nh, d)
        k = qkv[:, qs:qs + ks].view(T, self.nkv, d)
        v = qkv[:, qs + ks:].view(T, self.nkv, d)
        return q, k, v


class RowParallelLinear(nn.Module):
    def __init__(self, in_features: int, out_features: int,
                 dtype: torch.dtype, seed_tag: int = 0):
        super().__init__()
        tp, r = get_tp_size(), get_tp_rank()
        assert in_features % tp == 0
        self.in_features, self.out_features = in_features, out_features
        shard = in_features // tp
        full = _init_weight(out_features, in_features, dtype, seed_tag)
        self.weight = nn.Parameter(full[:, r * shard:(r + 1) * shard].contiguous(),
                                   requires_grad=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        out = x @ self.weight.t()
        return tensor_model_parallel_all_reduce(out)
