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
_QUANT_MODE = None  # None (bf16) | "fp8"

FP8_MAX = 448.0  # OCP e4m3fn max-normal; gfx950 MFMA/hipBLASLt native fp8


def set_init_device(device) -> None:
    """Where random-init weights are generated (GPU init is ~20x faster for
    the 8B/70B shapes; no checkpoints exist in this offline environment)."""
    global _INIT_DEVICE
    _INIT_DEVICE = device


def set_quant_mode(mode) -> None:
    """Weight format for subsequently-constructed linear layers.

    "fp8": per-output-channel float8_e4m3fn weights + fp32 scales; GEMMs run
    torch._scaled_mm (hipBLASLt fp8 MFMA on gfx950) with per-token dynamic
    activation scales.  BASELINE config 5 (fp8 checkpoints, halved weight
    stream).  None: plain bf16.
    """
    global _QUANT_MODE
    assert mode in (None, "fp8"), mode
    _QUANT_MODE = mode


def quantize_weight_fp8(w: torch.Tensor):
    """[N,K] → (float8_e4m3fn [N,K], fp32 scale [N]) per-output-channel."""
    wf = w.float()
    s = (wf.abs().amax(dim=1) / FP8_MAX).clamp_min(1e-8)
    q = (wf / s[:, None]).clamp(-FP8_MAX, FP8_MAX).to(torch.float8_e4m3fn)
    return q, s


def quant_fp8_rowwise(x: torch.Tensor):
    """Per-token dynamic activation quant: [T,K] → (fp8 [T,K], fp32 [T])."""
    xf = x.float()
    s = (xf.abs().amax(dim=-1) / FP8_MAX).clamp_min(1e-8)
    q = (xf / s[:, None]).clamp(-FP8_MAX, FP8_MAX).to(torch.float8_e4m3fn)
    return q, s


class _LinearCompute(nn.Module):
    """Shared weight-storage / GEMM path for the TP linear layers."""

    def _finalize_weight(self, w: torch.Tensor, shard_dim: int = None,
                         slices: list = None,
                         full_shape: tuple = None) -> None:
        """`slices` describes how this rank's shard maps into the FULL
        tensor along `shard_dim`: [(full_start, length, shard_start)].
        Checkpoint load/save (engine/checkpoint.py) uses it to shard a
        full checkpoint on load and to reassemble the full tensor on
        save — TP hot-swap without a per-rank checkpoint format."""
        self.quant = _QUANT_MODE
        if self.quant == "fp8":
            q, s = quantize_weight_fp8(w)
            self.weight = nn.Parameter(q, requires_grad=False)
            self.weight_scale = nn.Parameter(s, requires_grad=False)
            if slices is not None:
                if shard_dim == 0:
                    # per-out-channel scales shard with the rows
                    self.weight_scale._tp_shard_dim = 0
                    self.weight_scale._tp_slices = slices
                    self.weight_scale._tp_full_shape = (full_shape[0],)
                # shard_dim == 1: scale is per full row — replicated
        else:
            self.weight = nn.Parameter(w, requires_grad=False)
        if slices is not None:
            self.weight._tp_shard_dim = shard_dim
            self.weight._tp_slices = slices
            self.weight._tp_full_shape = full_shape

    def _mm(self, x) -> torch.Tensor:
        from hyperspot import ops as _O   # local: avoids import cycle at init
        if isinstance(x, _O.QTensor):
            # pre-quantized by a fused producer kernel (csrc/quant.hip)
            return torch._scaled_mm(
                x.data, self.weight.t(), scale_a=x.scale.unsqueeze(1),
                scale_b=self.weight_scale.unsqueeze(0),
                out_dtype=torch.bfloat16)
        if self.quant == "fp8":
            xq, xs = quant_fp8_rowwise(x)
            if x.is_cuda:
                return torch._scaled_mm(
                    xq, self.weight.t(), scale_a=xs[:, None].contiguous(),
                    scale_b=self.weight_scale.unsqueeze(0),
                    out_dtype=x.dtype)
            # CPU test path: identical numerics model, emulated in fp32
            wf = self.weight.float() * self.weight_scale[:, None]
            return ((xq.float() * xs[:, None]) @ wf.t()).to(x.dtype)
        return x @ self.weight.t()


def _init_weight(out_f: int, in_f: int, dtype: torch.dtype, seed_tag: int):
    # Random init; deterministic per-(shape, tag, device type) so every TP
    # rank materialises the same full tensor before sharding.
    seed = (seed_tag * 1000003 + out_f * 131 + in_f) % (2**31)
    dev = torch.device(_INIT_DEVICE)
    g = torch.Generator(device=dev).manual_seed(seed)
    w = torch.empty(out_f, in_f, dtype=torch.float32, device=dev)
    w.normal_(0.0, 0.02, generator=g)
    return w.to(dtype)


class ColumnParallelLinear(_LinearCompute):
    def __init__(self, in_features: int, out_features: int,
                 dtype: torch.dtype, seed_tag: int = 0):
        super().__init__()
        tp, r = get_tp_size(), get_tp_rank()
        assert out_features % tp == 0, (out_features, tp)
        self.in_features, self.out_features = in_features, out_features
        shard = out_features // tp
        full = _init_weight(out_features, in_features, dtype, seed_tag)
        self._finalize_weight(full[r * shard:(r + 1) * shard].contiguous(),
                              shard_dim=0,
                              slices=[(r * shard, shard, 0)],
                              full_shape=(out_features, in_features))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self._mm(x)

    def load_full_weight(self, w: torch.Tensor) -> None:
        tp, r = get_tp_size(), get_tp_rank()
        shard = self.out_features // tp
        self.weight.data.copy_(w[r * shard:(r + 1) * shard])


class MergedColumnParallelLinear(_LinearCompute):
    """Several column-parallel projections fused into one GEMM
    (gate_proj | up_proj).  Each sub-projection is sharded independently so
    the shard layout matches per-projection splits downstream."""

    def __init__(self, in_features: int, out_sizes: list, dtype: torch.dtype,
                 seed_tag: int = 0):
        super().__init__()
        tp, r = get_tp_size(), get_tp_rank()
        self.out_sizes = out_sizes
        parts = []
        slices = []
        off_full = 0
        off_shard = 0
        for i, o in enumerate(out_sizes):
            assert o % tp == 0
            full = _init_weight(o, in_features, dtype, seed_tag + i)
            sh = o // tp
            parts.append(full[r * sh:(r + 1) * sh])
            slices.append((off_full + r * sh, sh, off_shard))
            off_full += o
            off_shard += sh
        self._finalize_weight(torch.cat(parts, 0), shard_dim=0,
                              slices=slices,
                              full_shape=(sum(out_sizes), in_features))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self._mm(x)


class QKVParallelLinear(_LinearCompute):
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
        nq, nk = num_heads * head_dim, num_kv_heads * head_dim
        self._finalize_weight(
            torch.cat([qf[r * qs:(r + 1) * qs], kf[r * ks:(r + 1) * ks],
                       vf[r * ks:(r + 1) * ks]], 0),
            shard_dim=0,
            slices=[(r * qs, qs, 0),
                    (nq + r * ks, ks, qs),
                    (nq + nk + r * ks, ks, qs + ks)],
            full_shape=(nq + 2 * nk, hidden))

    def forward(self, x: torch.Tensor):
        """Returns strided [T, n, D] views into ONE fused qkv buffer — no
        .contiguous() copies; the HIP kernels take the token stride."""
        qkv = self._mm(x)
        qs, ks = self.nh * self.head_dim, self.nkv * self.head_dim
        T = qkv.shape[0]
        d = self.head_dim
        q = qkv[:, :qs].view(T, self.nh, d)
        k = qkv[:, qs:qs + ks].view(T, self.nkv, d)
        v = qkv[:, qs + ks:].view(T, self.nkv, d)
        return q, k, v


class RowParallelLinear(_LinearCompute):
    def __init__(self, in_features: int, out_features: int,
                 dtype: torch.dtype, seed_tag: int = 0):
        super().__init__()
        tp, r = get_tp_size(), get_tp_rank()
        assert in_features % tp == 0
        self.in_features, self.out_features = in_features, out_features
        shard = in_features // tp
        full = _init_weight(out_features, in_features, dtype, seed_tag)
        self._finalize_weight(
            full[:, r * shard:(r + 1) * shard].contiguous(),
            shard_dim=1, slices=[(r * shard, shard, 0)],
            full_shape=(out_features, in_features))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        out = self._mm(x)
        return tensor_model_parallel_all_reduce(out)
