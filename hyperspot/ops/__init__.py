"""Op dispatch layer.

On GPU (ROCm) every hot op runs a hand-written CDNA4 HIP kernel from the
in-tree extension ``hyperspot._C`` (csrc/, built for gfx950 only).  On CPU the
pure-PyTorch references in :mod:`hyperspot.ops.torch_ref` run instead (CI has
no GPU).  There is no CUDA path, no Triton, no multi-backend dispatch — a CUDA
device without the extension is a hard error, never a silent eager fallback.
"""

from __future__ import annotations

from typing import NamedTuple

import torch

from . import torch_ref

_C = None
_C_ERR: Exception | None = None
try:  # built by `python csrc/setup.py build_ext --inplace` (gfx950)
    from hyperspot import _C as _C  # type: ignore
except Exception as e:  # pragma: no cover - exercised only on GPU boxes
    _C_ERR = e


def have_native() -> bool:
    return _C is not None


def _native():
    if _C is None:
        raise RuntimeError(
            "hyperspot._C (gfx950 HIP extension) is not built but a GPU tensor "
            "reached the op layer. Build it in-tree with "
            "`python csrc/setup.py build_ext --inplace` "
            f"(import error: {_C_ERR!r})")
    return _C


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    if x.is_cuda:
        out = torch.empty_like(x)
        _native().rmsnorm(out, x, weight, eps)
        return out
    return torch_ref.rmsnorm(x, weight, eps)


def fused_add_rmsnorm(x: torch.Tensor, residual: torch.Tensor,
                      weight: torch.Tensor, eps: float):
    """residual += x (in place); x_out = rmsnorm(residual) (in place on x)."""
    if x.is_cuda:
        _native().fused_add_rmsnorm(x, residual, weight, eps)
        return x, residual
    out, res = torch_ref.fused_add_rmsnorm(x, residual, weight, eps)
    return out, res


def rope_kv_append(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                   positions: torch.Tensor, cos_sin: torch.Tensor,
                   slot_mapping: torch.Tensor,
                   k_cache: torch.Tensor, v_cache: torch.Tensor) -> None:
    """Fused: NeoX RoPE on q,k (in place) + scatter k,v into the paged cache.

    q [T,H,D], k/v [T,KV,D]; cos_sin [max_pos, D] fp32 (cos half | sin half);
    slot_mapping [T] int64 (-1 = skip append, e.g. sliding-window drop).
    """
    if q.is_cuda:
        _native().rope_kv_append(q, k, v, positions, cos_sin, slot_mapping,
                                 k_cache, v_cache)
        return
    d = q.shape[-1]
    cs = cos_sin[positions.long()]
    cos, sin = cs[:, : d // 2], cs[:, d // 2:]
    q.copy_(torch_ref._rotate_neox(q, cos, sin))
    k.copy_(torch_ref._rotate_neox(k, cos, sin))
    mask = slot_mapping >= 0
    if mask.any():
        torch_ref.kv_cache_append(k[mask], v[mask], k_cache, v_cache,
                                  slot_mapping[mask])


import os

_ATTN_SPLIT_TARGET = int(os.environ.get("HYPERSPOT_ATTN_SPLIT_TARGET", "256"))


def _attn_split(rows: int, kvh: int) -> int:
    """Flash-decode page-split factor: bring the workgroup count up to
    ~_ATTN_SPLIT_TARGET when rows*kvh alone cannot fill the 256 CUs (small
    batch, or the kvh=1 GQA shard of 70B at TP=8).  Measured (MI355X, profiles/
    r01_decode_v3.md): splitting at >=256 WGs already loses — the fp32
    partial round trip + merge launch cost more than the occupancy buys —
    so only genuinely tiny grids (rows*kvh < 256, e.g. batch<32 at TP=8)
    split.  Static in (rows, kvh) => capture-safe."""
    s = max(1, min(16, _ATTN_SPLIT_TARGET // max(1, rows * kvh)))
    return 1 if s <= 1 else s


def paged_attn_decode(q: torch.Tensor, k_cache: torch.Tensor,
                      v_cache: torch.Tensor, block_tables: torch.Tensor,
                      seq_lens: torch.Tensor, scale: float) -> torch.Tensor:
    if q.is_cuda:
        out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        kvh = k_cache.shape[1]
        split = _attn_split(q.shape[0], kvh) if q.shape[-1] == 128 else 1
        ws = None
        if split > 1:
            ws = torch.empty(q.shape[0] * kvh * split
                             * (q.shape[1] // kvh) * 130,
                             dtype=torch.float32, device=q.device)
        _native().paged_attn(out, q, k_cache, v_cache, block_tables,
                             seq_lens, None, scale, ws, split)
        return out
    return torch_ref.paged_attn_decode(q, k_cache, v_cache, block_tables,
                                       seq_lens, scale)


def attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
              k_cache: torch.Tensor, v_cache: torch.Tensor, meta,
              scale: float) -> torch.Tensor:
    """Attention for one layer, prefill or decode, reading the paged cache.

    The ONE gfx950 kernel serves both modes: decode rows attend over
    ctx=seq_len cache slots; prefill rows attend over slots 0..pos (their
    own k/v were appended by the fused RoPE kernel just before), which
    realises causal varlen attention without a separate kernel.
    """
    if meta.mode == "decode":
        return paged_attn_decode(q, k_cache, v_cache, meta.block_tables,
                                 meta.seq_lens, scale)
    if q.is_cuda:
        out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        if q.shape[-1] == 128 and getattr(meta, "fresh_prefill", True):
            # MFMA flash prefill (K/V straight from the qkv projection)
            _native().attn_prefill_mfma(out, q, k, v, meta.seq_start,
                                        meta.max_seqlen, scale)
        else:
            # per-row attention over the (freshly appended) paged cache:
            # chunked-prefill continuation rows or odd head dims
            _native().paged_attn(out, q, k_cache, v_cache,
                                 meta.block_tables, meta.ctx_lens,
                                 meta.row_seq, scale, None, 1)
        return out
    if getattr(meta, "fresh_prefill", True):
        return torch_ref.prefill_attn(q, k, v, meta.seq_start, scale)
    return torch_ref.prefill_attn_paged(q, k_cache, v_cache,
                                        meta.block_tables, meta.ctx_lens,
                                        meta.row_seq, scale)


def silu_mul(gate_up: torch.Tensor) -> torch.Tensor:
    if gate_up.is_cuda:
        i = gate_up.shape[-1] // 2
        out = torch.empty(*gate_up.shape[:-1], i, dtype=gate_up.dtype,
                          device=gate_up.device)
        _native().silu_mul(out, gate_up)
        return out
    return torch_ref.silu_mul(gate_up)


class QTensor(NamedTuple):
    """Per-token-scaled fp8 activation: data e4m3fn [T,K], scale fp32 [T].

    Produced on GPU by the fused quant kernels (csrc/quant.hip) so fp8
    GEMM inputs never take an extra elementwise pass; consumed by the TP
    linear layers via torch._scaled_mm (hipBLASLt fp8 MFMA)."""

    data: torch.Tensor
    scale: torch.Tensor


def quant_fp8(x: torch.Tensor) -> QTensor:
    """[T,K] bf16 -> QTensor (GPU; row-per-workgroup amax+scale+pack)."""
    assert x.is_cuda
    T, K = x.shape
    out = torch.empty(T, K, dtype=torch.float8_e4m3fn, device=x.device)
    scales = torch.empty(T, dtype=torch.float32, device=x.device)
    _native().quant_fp8(out, scales, x)
    return QTensor(out, scales)


def fused_add_rmsnorm_q(x: torch.Tensor, residual: torch.Tensor,
                        weight: torch.Tensor, eps: float):
    """residual += x (in place); returns (QTensor of rmsnorm(residual)*w,
    residual).  GPU-only fused producer for the fp8 GEMM path."""
    assert x.is_cuda
    T, K = x.shape
    out = torch.empty(T, K, dtype=torch.float8_e4m3fn, device=x.device)
    scales = torch.empty(T, dtype=torch.float32, device=x.device)
    _native().fused_add_rmsnorm_fp8(out, scales, x, residual, weight, eps)
    return QTensor(out, scales), residual


def silu_mul_q(gate_up: torch.Tensor) -> QTensor:
    """silu(g)*u -> fp8 QTensor (GPU fused producer)."""
    assert gate_up.is_cuda
    i = gate_up.shape[-1] // 2
    out = torch.empty(*gate_up.shape[:-1], i, dtype=torch.float8_e4m3fn,
                      device=gate_up.device)
    scales = torch.empty(gate_up.shape[0], dtype=torch.float32,
                         device=gate_up.device)
    _native().silu_mul_fp8(out, scales, gate_up)
    return QTensor(out, scales)


MOE_BM = 256  # sorted-pair tile granularity of the MoE kernels (csrc/moe.hip)


def _moe_splitk(nblocks: int, nchunks: int, out_numel: int, dev):
    """Split-K degree + fp32 workspace for a grouped GEMM launch.

    The w2 GEMM at Mixtral decode shapes has ~288 blocks for 256 CUs —
    its ragged second wave idles most of the chip; splitting K into up
    to 8 segments (each must keep >= 3 k-chunks for the LDS ring and
    divide the chunk count) multiplies the block count back above ~4
    per CU.  Shapes-only decision: hipGraph-capture-safe.
    HS_MOE_SPLITK forces a degree (0 = auto)."""
    forced = int(os.environ.get("HS_MOE_SPLITK", "0"))
    s = 1
    if forced and nblocks >= 1024:
        forced = 0          # forcing never touches saturated launches
    while (s < 8 and nchunks % (s * 2) == 0 and nchunks // (s * 2) >= 3
           and (nblocks * s < 1024 if not forced else s * 2 <= forced)):
        s *= 2
    if forced == 1:
        s = 1
    if s == 1:
        return 1, None
    return s, torch.empty(s * out_numel, dtype=torch.float32, device=dev)


def moe_ffn(x: torch.Tensor, w13: torch.Tensor, w2: torch.Tensor,
            weights: torch.Tensor, ids: torch.Tensor) -> torch.Tensor:
    """Grouped expert SwiGLU FFN (GPU, hipGraph-capture-safe).

    x [T,H]; w13 [E_local, 2I, H]; w2 [E_local, H, I]; weights [T,K] fp32/bf16
    routing weights; ids [T,K] LOCAL expert indices.  Token->expert sorting,
    the two grouped MFMA GEMMs and the weighted combine all run on-device
    with grids that depend only on (T, K, E) — no host sync, no dynamic
    shapes (MoE dispatch/combine + grouped GEMM).
    """
    T, H = x.shape
    K = ids.shape[1]
    E = w13.shape[0]
    TK = T * K
    ntiles = (TK + E * (MOE_BM - 1) + MOE_BM - 1) // MOE_BM
    P = ntiles * MOE_BM
    dev = x.device
    flat = ids.reshape(-1).to(torch.int32)
    sorted_ids = torch.empty(P, dtype=torch.int32, device=dev)
    tile_expert = torch.empty(ntiles, dtype=torch.int32, device=dev)
    inv_pos = torch.empty(TK, dtype=torch.int32, device=dev)
    n = _native()
    n.moe_align(sorted_ids, tile_expert, inv_pos, flat, E)
    I2 = w13.shape[1]
    h1 = torch.empty(P, I2, dtype=torch.bfloat16, device=dev)
    s1, ws1 = _moe_splitk(ntiles * (I2 // 128), H // 64, P * I2, dev)
    n.moe_gemm(h1, x, w13, sorted_ids, tile_expert, K, s1, ws1)
    a = silu_mul(h1)
    y = torch.empty(P, H, dtype=torch.bfloat16, device=dev)
    I = w2.shape[2]
    s2, ws2 = _moe_splitk(ntiles * (H // 128), I // 64, P * H, dev)
    n.moe_gemm(y, a, w2, sorted_ids, tile_expert, 0, s2, ws2)
    out = torch.empty(T, H, dtype=torch.bfloat16, device=dev)
    n.moe_combine(out, y, weights.float().contiguous(), inv_pos, K)
    return out


def moe_ffn_fp8(xq: "QTensor", w13q: torch.Tensor, w13s: torch.Tensor,
                w2q: torch.Tensor, w2s: torch.Tensor,
                weights: torch.Tensor, ids: torch.Tensor) -> torch.Tensor:
    """fp8 grouped expert FFN: e4m3fn activations (QTensor from the fused
    norm producer) x e4m3fn expert weights through
    v_mfma_f32_32x32x16_fp8_fp8; the silu stage re-quantizes via the fused
    silu_mul_fp8 kernel so no extra passes appear.  ~2x the bf16 MoE
    weight bandwidth (the binding constraint, profiles/r01)."""
    T, H = xq.data.shape
    K = ids.shape[1]
    E = w13q.shape[0]
    TK = T * K
    ntiles = (TK + E * (MOE_BM - 1) + MOE_BM - 1) // MOE_BM
    P = ntiles * MOE_BM
    dev = xq.data.device
    flat = ids.reshape(-1).to(torch.int32)
    sorted_ids = torch.empty(P, dtype=torch.int32, device=dev)
    tile_expert = torch.empty(ntiles, dtype=torch.int32, device=dev)
    inv_pos = torch.empty(TK, dtype=torch.int32, device=dev)
    n = _native()
    n.moe_align(sorted_ids, tile_expert, inv_pos, flat, E)
    I2 = w13q.shape[1]
    h1 = torch.empty(P, I2, dtype=torch.bfloat16, device=dev)
    s1, ws1 = _moe_splitk(ntiles * (I2 // 128), H // 128, P * I2, dev)
    n.moe_gemm_fp8(h1, xq.data, xq.scale, w13q, w13s, sorted_ids,
                   tile_expert, K, s1, ws1)
    a = silu_mul_q(h1)
    y = torch.empty(P, H, dtype=torch.bfloat16, device=dev)
    I = w2q.shape[2]
    s2, ws2 = _moe_splitk(ntiles * (H // 128), I // 128, P * H, dev)
    n.moe_gemm_fp8(y, a.data, a.scale, w2q, w2s, sorted_ids, tile_expert,
                   0, s2, ws2)
    out = torch.empty(T, H, dtype=torch.bfloat16, device=dev)
    n.moe_combine(out, y, weights.float().contiguous(), inv_pos, K)
    return out


def greedy_sample(logits: torch.Tensor) -> torch.Tensor:
    if logits.is_cuda:
        out = torch.empty(logits.shape[0], dtype=torch.long,
                          device=logits.device)
        _native().greedy_sample(out, logits)
        return out
    return logits.float().argmax(-1)


def sample(logits: torch.Tensor, temperature: torch.Tensor,
           top_p: torch.Tensor, top_k: torch.Tensor,
           uniform: torch.Tensor) -> torch.Tensor:
    """Batch sampling. Greedy rows (temperature==0) take the argmax kernel;
    stochastic rows go through filtering + an inverse-CDF draw."""
    if not logits.is_cuda:
        return torch_ref.sample(logits, temperature, top_p, top_k, uniform)
    if bool((temperature == 0).all()):
        return greedy_sample(logits)
    # Stochastic path: torch does the (sort-based) top-k/top-p filtering,
    # the HIP kernel does temperature softmax + inverse-CDF in one pass.
    lf = logits.float()
    B, V = lf.shape
    t = temperature.clamp_min(1e-6)[:, None]
    row = lf / t
    if bool((top_k > 0).any()):
        k = top_k.clamp(0, V)
        kth = torch.where(
            k > 0,
            row.topk(int(k.max().clamp(min=1)), dim=-1).values.gather(
                1, (k.clamp(min=1) - 1)[:, None]).squeeze(1),
            torch.full_like(row[:, 0], float("-inf")))
        row = torch.where(row < kth[:, None], float("-inf"), row)
    if bool((top_p < 1.0).any()):
        sp, idx = row.softmax(-1).sort(descending=True)
        cum = sp.cumsum(-1)
        drop = (cum - sp) >= top_p[:, None]
        row = row.masked_fill(drop.gather(1, idx.argsort(-1)), float("-inf"))
    out = torch.empty(B, dtype=torch.long, device=logits.device)
    _native().inv_cdf_sample(out, row, uniform)
    greedy = temperature == 0
    if bool(greedy.any()):
        out = torch.where(greedy, greedy_sample(logits), out)
    return out
