"""Llama-3 family, MI355X-first.

Design notes (vs a CUDA port):
  - token-major [T, hidden] activations, prefill and decode separated so the
    decode step is hipGraph-capturable (static shapes per batch bucket)
  - RoPE + paged-KV append is ONE fused HIP kernel pass (hyperspot.ops)
  - rmsnorm + residual-add fused; SwiGLU fused; attention reads the paged
    cache directly (no gather/cat of past keys)
  - GEMMs via torch.matmul -> hipBLASLt; everything else hand-written CDNA4

Model set and serving contract per the reference's spec-only llm-gateway
(reference modules/llm-gateway/docs/DESIGN.md; SURVEY.md §2.9 kernel list).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

import torch
from torch import nn

from hyperspot import ops
from hyperspot.engine.config import ModelSpec
from hyperspot.ops.torch_ref import rope_cos_sin
from hyperspot.parallel import (QKVParallelLinear, RowParallelLinear,
                                MergedColumnParallelLinear)
from hyperspot.parallel.state import get_tp_size


@dataclass
class ForwardMeta:
    """Per-step attention metadata prepared by the ModelRunner."""

    mode: str                      # "prefill" | "decode"
    positions: torch.Tensor        # [T] int64
    slot_mapping: torch.Tensor     # [T] int64 into the paged cache
    # prefill
    seq_start: Optional[torch.Tensor] = None   # [num_seqs+1] int32
    max_seqlen: int = 0
    row_seq: Optional[torch.Tensor] = None     # [T] int32 row -> seq index
    ctx_lens: Optional[torch.Tensor] = None    # [T] int32 = position + 1
    # decode (block_tables also used by GPU prefill, indexed via row_seq)
    block_tables: Optional[torch.Tensor] = None  # [B, max_blocks] int32
    seq_lens: Optional[torch.Tensor] = None      # [B] int32
    # logits are computed only for these token rows (last token per seq)
    logits_indices: Optional[torch.Tensor] = None
    # embeddings mode: return final hidden states instead of logits
    return_hidden: bool = False
    # prefill: True when every row starts at position 0 (chunk == whole
    # prompt) -> the self-contained MFMA flash kernel applies; False for
    # chunked-prefill continuation batches, which must attend over the
    # paged cache (per-row ctx_lens)
    fresh_prefill: bool = True


class Attention(nn.Module):
    def __init__(self, spec: ModelSpec, layer_idx: int, dtype: torch.dtype):
        super().__init__()
        self.layer_idx = layer_idx
        self.head_dim = spec.head_dim
        self.scale = spec.head_dim ** -0.5
        self.qkv = QKVParallelLinear(spec.hidden_size, spec.head_dim,
                                     spec.num_heads, spec.num_kv_heads, dtype,
                                     seed_tag=layer_idx * 10 + 1)
        self.o_proj = RowParallelLinear(
            spec.num_heads * spec.head_dim, spec.hidden_size, dtype,
            seed_tag=layer_idx * 10 + 4)

    def forward(self, x, meta: ForwardMeta,
                kv_cache: torch.Tensor, cos_sin: torch.Tensor) -> torch.Tensor:
        q, k, v = self.qkv(x)
        T = q.shape[0]
        k_cache, v_cache = kv_cache[0], kv_cache[1]
        ops.rope_kv_append(q, k, v, meta.positions, cos_sin,
                           meta.slot_mapping, k_cache, v_cache)
        o = ops.attention(q, k, v, k_cache, v_cache, meta, self.scale)
        o = o.view(T, -1)
        if self.o_proj.quant == "fp8" and o.is_cuda:
            o = ops.quant_fp8(o)   # fused amax+scale+pack (csrc/quant.hip)
        return self.o_proj(o)


class LlamaMLP(nn.Module):
    def __init__(self, spec: ModelSpec, layer_idx: int, dtype: torch.dtype):
        super().__init__()
        self.gate_up = MergedColumnParallelLinear(
            spec.hidden_size, [spec.intermediate_size, spec.intermediate_size],
            dtype, seed_tag=layer_idx * 10 + 5)
        self.down = RowParallelLinear(spec.intermediate_size,
                                      spec.hidden_size, dtype,
                                      seed_tag=layer_idx * 10 + 7)
        self.tp = get_tp_size()
        self.inter_shard = spec.intermediate_size // self.tp

    def forward(self, x) -> torch.Tensor:
        gu = self.gate_up(x)
        if self.down.quant == "fp8" and gu.is_cuda:
            return self.down(ops.silu_mul_q(gu))
        return self.down(ops.silu_mul(gu))


class DecoderLayer(nn.Module):
    def __init__(self, spec: ModelSpec, layer_idx: int, dtype: torch.dtype,
                 mlp_cls=LlamaMLP):
        super().__init__()
        self.input_norm_w = nn.Parameter(
            torch.ones(spec.hidden_size, dtype=dtype), requires_grad=False)
        self.post_norm_w = nn.Parameter(
            torch.ones(spec.hidden_size, dtype=dtype), requires_grad=False)
        self.eps = spec.rms_eps
        self.attn = Attention(spec, layer_idx, dtype)
        self.mlp = mlp_cls(spec, layer_idx, dtype)

    def forward(self, x: torch.Tensor, residual: Optional[torch.Tensor],
                meta: ForwardMeta, kv_cache: torch.Tensor,
                cos_sin: torch.Tensor):
        # fp8 GPU path: the norm kernels emit per-token-scaled fp8 directly
        # (csrc/quant.hip), so GEMM inputs never take an extra quant pass.
        fp8 = self.attn.qkv.quant == "fp8" and x.is_cuda
        if residual is None:
            # x is the fresh embedding-gather output: safe to alias as the
            # residual (fused_add_rmsnorm mutates it in place from layer 1 on)
            residual = x
            h = ops.rmsnorm(x, self.input_norm_w, self.eps)
            if fp8:
                h = ops.quant_fp8(h)
        elif fp8:
            h, residual = ops.fused_add_rmsnorm_q(x, residual,
                                                  self.input_norm_w, self.eps)
        else:
            h, residual = ops.fused_add_rmsnorm(x, residual,
                                                self.input_norm_w, self.eps)
        h = self.attn(h, meta, kv_cache, cos_sin)
        if fp8:
            h, residual = ops.fused_add_rmsnorm_q(h, residual,
                                                  self.post_norm_w, self.eps)
        else:
            h, residual = ops.fused_add_rmsnorm(h, residual, self.post_norm_w,
                                                self.eps)
        h = self.mlp(h)
        return h, residual


class LlamaForCausalLM(nn.Module):
    mlp_cls = LlamaMLP

    def __init__(self, spec: ModelSpec, dtype: torch.dtype = torch.bfloat16):
        super().__init__()
        self.spec = spec
        self.dtype = dtype
        from hyperspot.parallel.layers import _init_weight
        # Embedding + LM head replicated (vocab GEMM needs no collective;
        # 288 GB HBM3E/GPU makes the duplicated 2 GB irrelevant at 70B).
        self.embed = nn.Parameter(
            _init_weight(spec.vocab_size, spec.hidden_size, dtype, 900001),
            requires_grad=False)
        if spec.tie_embeddings:
            self.lm_head = self.embed
        else:
            self.lm_head = nn.Parameter(
                _init_weight(spec.vocab_size, spec.hidden_size, dtype,
                             900002), requires_grad=False)
        self.layers = nn.ModuleList([
            DecoderLayer(spec, i, dtype, mlp_cls=self.mlp_cls)
            for i in range(spec.num_layers)])
        self.final_norm_w = nn.Parameter(
            torch.ones(spec.hidden_size, dtype=dtype), requires_grad=False)
        cs = rope_cos_sin(torch.arange(spec.max_position), spec.head_dim,
                          spec.rope_theta)
        self.register_buffer("cos_sin", cs, persistent=False)

    def forward(self, input_ids: torch.Tensor, meta: ForwardMeta,
                kv_caches: List[torch.Tensor]) -> torch.Tensor:
        """Returns logits [num_logit_rows, vocab]."""
        x = self.embed[input_ids]
        residual = None
        for i, layer in enumerate(self.layers):
            x, residual = layer(x, residual, meta, kv_caches[i], self.cos_sin)
        x, _ = ops.fused_add_rmsnorm(x, residual, self.final_norm_w,
                                     self.spec.rms_eps)
        if meta.return_hidden:
            return x
        if meta.logits_indices is not None:
            x = x[meta.logits_indices]
        return x @ self.lm_head.t()
