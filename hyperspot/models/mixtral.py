"""Mixtral 8x7B — MoE decoder with expert parallelism over RCCL all-to-all.

Sharding model (MI355X-first):
  - attention is tensor-parallel exactly like Llama
  - experts are EXPERT-parallel: each rank owns num_experts/ep of the FFNs
    whole (no intra-expert sharding), tokens travel rank->expert-owner->rank
    via two all-to-alls per layer (dispatch, combine).  With EP==1 (or on
    CPU tests) the same code runs the local grouped-GEMM path.
"""

from __future__ import annotations

import math
import os

import torch
import torch.distributed as dist
from torch import nn

from hyperspot import ops
from hyperspot.engine.config import ModelSpec
from hyperspot.models.llama import DecoderLayer, LlamaForCausalLM
from hyperspot.parallel.layers import (_init_weight, _QUANT_MODE,
                                       quantize_weight_fp8)
from hyperspot.parallel.state import get_ep_group, get_ep_rank, get_ep_size


class MixtralMoE(nn.Module):
    def __init__(self, spec: ModelSpec, layer_idx: int, dtype: torch.dtype):
        super().__init__()
        self.spec = spec
        self.top_k = spec.num_experts_per_tok
        self.num_experts = spec.num_experts
        self.router = nn.Parameter(
            _init_weight(spec.num_experts, spec.hidden_size, dtype,
                         layer_idx * 10 + 8), requires_grad=False)
        ep, r = get_ep_size(), get_ep_rank()
        assert spec.num_experts % ep == 0, (spec.num_experts, ep)
        self.experts_per_rank = spec.num_experts // ep
        self.local_expert_start = r * self.experts_per_rank
        # fixed-capacity (capture-safe) dispatch knobs; see _ep_moe_static
        self.capacity_factor = float(os.environ.get("HS_EP_CAPACITY", "2.0"))
        self.force_static_ep = os.environ.get("HS_EP_STATIC", "") == "1"
        self.register_buffer("ep_overflow",
                             torch.zeros((), dtype=torch.long),
                             persistent=False)
        h, i = spec.hidden_size, spec.intermediate_size
        w13, w2 = [], []
        for e in range(self.local_expert_start,
                       self.local_expert_start + self.experts_per_rank):
            tag = layer_idx * 100 + e * 3 + 11
            w13.append(torch.cat([_init_weight(i, h, dtype, tag),
                                  _init_weight(i, h, dtype, tag + 1)], 0))
            w2.append(_init_weight(h, i, dtype, tag + 2))
        # [E_local, 2I, H] and [E_local, H, I]
        from hyperspot.parallel import layers as _L
        self.quant = _L._QUANT_MODE
        if self.quant == "fp8":
            qs13 = [quantize_weight_fp8(w) for w in w13]
            qs2 = [quantize_weight_fp8(w) for w in w2]
            self.w13 = nn.Parameter(torch.stack([q for q, _ in qs13]),
                                    requires_grad=False)
            self.w13_scale = nn.Parameter(torch.stack([s for _, s in qs13]),
                                          requires_grad=False)
            self.w2 = nn.Parameter(torch.stack([q for q, _ in qs2]),
                                   requires_grad=False)
            self.w2_scale = nn.Parameter(torch.stack([s for _, s in qs2]),
                                         requires_grad=False)
        else:
            self.w13 = nn.Parameter(torch.stack(w13), requires_grad=False)
            self.w2 = nn.Parameter(torch.stack(w2), requires_grad=False)
        if ep > 1:
            # expert tensors are block-sharded along dim 0 (whole experts
            # per rank); the same slice metadata the TP layers carry lets
            # checkpoint save/load gather/scatter the FULL [E, ...] layout
            # (engine/checkpoint.py), just over the EP group instead
            for p in (self.w13, self.w2) + (
                    (self.w13_scale, self.w2_scale)
                    if self.quant == "fp8" else ()):
                p._tp_slices = [(self.local_expert_start,
                                 self.experts_per_rank, 0)]
                p._tp_full_shape = (spec.num_experts,) + tuple(p.shape[1:])
                p._tp_shard_dim = 0
                p._shard_group = "ep"

    def _expert_ffn(self, x: torch.Tensor, e_local: int) -> torch.Tensor:
        if self.quant == "fp8":   # CPU/EP fallback: dequantized math
            w13 = (self.w13[e_local].float()
                   * self.w13_scale[e_local][:, None]).to(x.dtype)
            w2 = (self.w2[e_local].float()
                  * self.w2_scale[e_local][:, None]).to(x.dtype)
            return ops.silu_mul(x @ w13.t()) @ w2.t()
        gu = x @ self.w13[e_local].t()
        return ops.silu_mul(gu) @ self.w2[e_local].t()

    def forward(self, x) -> torch.Tensor:
        xq = None
        if isinstance(x, ops.QTensor):
            # fp8 path: the fused norm producer already quantized x;
            # dequantize only the router's [T,E] logits input
            xq = x
            x = (xq.data.float() * xq.scale[:, None]).to(torch.bfloat16)
        T, H = x.shape
        weights, ids = ops.torch_ref.moe_route(x, self.router, self.top_k) \
            if not x.is_cuda else self._route_gpu(x)
        if xq is not None and get_ep_size() == 1 and ops.have_native():
            return ops.moe_ffn_fp8(xq, self.w13, self.w13_scale, self.w2,
                                   self.w2_scale, weights.float(),
                                   ids.to(torch.int32))
        if get_ep_size() == 1:
            return self._local_moe(x, weights.to(x.dtype), ids)
        return self._ep_moe(x, weights.to(x.dtype), ids)

    def _route_gpu(self, x: torch.Tensor):
        logits = x.float() @ self.router.float().t()
        probs = logits.softmax(-1)
        w, ids = probs.topk(self.top_k, dim=-1)
        return w / w.sum(-1, keepdim=True), ids

    def _local_moe(self, x, weights, ids):
        if x.is_cuda and ops.have_native():
            # grouped MFMA kernels, capture-safe (csrc/moe.hip)
            local = (ids - self.local_expert_start).to(torch.int32)
            return ops.moe_ffn(x, self.w13, self.w2, weights, local)
        out = torch.zeros_like(x)
        flat_ids = ids.reshape(-1)
        flat_w = weights.reshape(-1)
        token_idx = torch.arange(x.shape[0], device=x.device
                                 ).repeat_interleave(self.top_k)
        for e in range(self.num_experts):
            m = flat_ids == e
            if not bool(m.any()):
                continue
            rows = token_idx[m]
            y = self._expert_ffn(x[rows], e - self.local_expert_start) \
                if self.local_expert_start <= e < self.local_expert_start + self.experts_per_rank \
                else None
            if y is None:
                continue
            out.index_add_(0, rows, y * flat_w[m, None])
        return out

    def _ep_moe(self, x, weights, ids):
        """Dispatch tokens to expert owners (all-to-all), run local experts,
        combine back (all-to-all), weight and reduce.  At quant=fp8 the
        dispatched activations travel as e4m3fn bytes + per-token f32
        scales — half the xGMI wire bytes — and feed the fp8 grouped GEMM
        directly on the owner rank.

        Two dispatch modes:
          - dynamic (exact): per-rank counts exchanged first, variable-split
            all-to-alls.  Needs a host sync (`.tolist()`) so it cannot be
            captured in a hipGraph — used for eager/prefill forwards.
          - static (capture-safe): fixed-capacity buckets, see
            _ep_moe_static — used under graph capture (decode buckets)."""
        if self.force_static_ep or (
                x.is_cuda and torch.cuda.is_current_stream_capturing()):
            return self._ep_moe_static(x, weights, ids)
        ep = get_ep_size()
        group = get_ep_group()
        T, H = x.shape
        K = self.top_k
        fp8_wire = self.quant == "fp8"
        flat_ids = ids.reshape(-1)                         # [T*K]
        owner = flat_ids // self.experts_per_rank          # [T*K]
        order = torch.argsort(owner, stable=True)
        send_x = x.repeat_interleave(K, dim=0)[order]      # [T*K, H]
        send_eids = flat_ids[order]
        send_counts = torch.bincount(owner, minlength=ep)
        # exchange counts
        recv_counts = torch.empty_like(send_counts)
        dist.all_to_all_single(recv_counts, send_counts, group=group)
        sc = send_counts.tolist()
        rc = recv_counts.tolist()
        recv_eids = torch.empty(sum(rc), dtype=send_eids.dtype, device=x.device)
        dist.all_to_all_single(recv_eids, send_eids, rc, sc, group=group)
        if fp8_wire:
            from hyperspot.parallel.layers import quant_fp8_rowwise
            sq, ss = quant_fp8_rowwise(send_x)
            recv_q = torch.empty(sum(rc), H, dtype=torch.uint8,
                                 device=x.device)
            recv_s = torch.empty(sum(rc), dtype=torch.float32,
                                 device=x.device)
            # fp8 bytes travel as uint8 (gloo/rccl dtype-agnostic payload)
            dist.all_to_all_single(recv_q, sq.view(torch.uint8), rc, sc,
                                   group=group)
            dist.all_to_all_single(recv_s, ss, rc, sc, group=group)
            recv_x = None
        else:
            recv_x = torch.empty(sum(rc), H, dtype=x.dtype, device=x.device)
            dist.all_to_all_single(recv_x, send_x, rc, sc, group=group)
        # local grouped FFN
        local = recv_eids - self.local_expert_start
        if fp8_wire and x.is_cuda and ops.have_native():
            xq = ops.QTensor(recv_q.view(torch.float8_e4m3fn), recv_s)
            ones = torch.ones(recv_q.shape[0], 1, device=x.device,
                              dtype=torch.float32)
            y = ops.moe_ffn_fp8(xq, self.w13, self.w13_scale, self.w2,
                                self.w2_scale, ones,
                                local.view(-1, 1).to(torch.int32))
        elif not fp8_wire and x.is_cuda and ops.have_native():
            ones = torch.ones(recv_x.shape[0], 1, device=x.device,
                              dtype=torch.float32)
            y = ops.moe_ffn(recv_x, self.w13, self.w2, ones,
                            local.view(-1, 1).to(torch.int32))
        else:
            if fp8_wire:    # CPU/EP test path: dequantize the wire bytes
                recv_x = (recv_q.view(torch.float8_e4m3fn).float()
                          * recv_s[:, None]).to(x.dtype)
            y = torch.empty_like(recv_x)
            for e in range(self.experts_per_rank):
                m = local == e
                if bool(m.any()):
                    y[m] = self._expert_ffn(recv_x[m], e)
        # combine back
        back = torch.empty(sum(sc), H, dtype=x.dtype, device=x.device)
        dist.all_to_all_single(back, y, sc, rc, group=group)
        out = torch.zeros_like(x)
        token_idx = torch.arange(T, device=x.device).repeat_interleave(K)[order]
        out.index_add_(0, token_idx, back * weights.reshape(-1)[order, None])
        return out

    def _ep_capacity(self, T: int) -> int:
        """Tokens-per-destination-rank bucket size: mean load x factor."""
        ep = get_ep_size()
        c = math.ceil(T * self.top_k / ep * self.capacity_factor)
        return max(8, min(T * self.top_k, c))

    def _ep_moe_static(self, x, weights, ids):
        """Capture-safe EP dispatch: fixed-capacity buckets + EQUAL-split
        all-to-alls.

        Every tensor shape here is a function of (T, ep, capacity) only —
        no `.tolist()`/`.item()` host syncs — so hipGraph capture records
        the RCCL all-to-alls exactly like TP's decode all-reduce, and
        ep>1 decode no longer forces eager mode.  The trade is the
        standard fixed-capacity MoE one: a rank sending more than
        `capacity` tokens to one owner drops the excess top-k assignments
        (counted in `ep_overflow`; remaining experts still serve the
        token).  Capacity = mean load x HS_EP_CAPACITY (default 2.0).
        Eager forwards (prefill) keep the exact dynamic dispatch above.
        """
        ep = get_ep_size()
        group = get_ep_group()
        T, H = x.shape
        K = self.top_k
        C = self._ep_capacity(T)
        dev = x.device
        fp8_wire = self.quant == "fp8"
        flat_ids = ids.reshape(-1).to(torch.long)          # [T*K]
        owner = flat_ids // self.experts_per_rank
        order = torch.argsort(owner, stable=True)
        sorted_owner = owner[order]
        counts = torch.bincount(owner, minlength=ep)
        offsets = torch.cumsum(counts, 0) - counts         # exclusive
        slot = torch.arange(T * K, device=dev) - offsets[sorted_owner]
        keep = slot < C
        trash = torch.full_like(slot, ep * C)              # one spare row
        dest = torch.where(keep, sorted_owner * C + slot, trash)
        src_rows = torch.arange(T, device=dev).repeat_interleave(K)[order]
        self.ep_overflow += (~keep).sum()
        # scatter tokens into their destination buckets (padding rows stay 0)
        send_x = x.new_zeros(ep * C + 1, H)
        send_x.index_copy_(0, dest, x[src_rows])
        send_eid = torch.full((ep * C + 1,), -1, dtype=torch.long, device=dev)
        send_eid.index_copy_(0, dest, flat_ids[order])
        recv_eid = torch.empty(ep * C, dtype=torch.long, device=dev)
        dist.all_to_all_single(recv_eid, send_eid[:ep * C], group=group)
        if fp8_wire:
            from hyperspot.parallel.layers import quant_fp8_rowwise
            sq, ss = quant_fp8_rowwise(send_x[:ep * C])
            recv_q = torch.empty(ep * C, H, dtype=torch.uint8, device=dev)
            recv_s = torch.empty(ep * C, dtype=torch.float32, device=dev)
            dist.all_to_all_single(recv_q, sq.view(torch.uint8), group=group)
            dist.all_to_all_single(recv_s, ss, group=group)
            recv_x = None
        else:
            recv_x = torch.empty(ep * C, H, dtype=x.dtype, device=dev)
            dist.all_to_all_single(recv_x, send_x[:ep * C], group=group)
        # local grouped FFN over all ep*C rows; padding rows (eid -1) are
        # zero inputs routed to local expert 0 and discarded on combine
        local = (recv_eid - self.local_expert_start).clamp(
            0, self.experts_per_rank - 1)
        if fp8_wire and x.is_cuda and ops.have_native():
            xq = ops.QTensor(recv_q.view(torch.float8_e4m3fn), recv_s)
            ones = torch.ones(ep * C, 1, device=dev, dtype=torch.float32)
            y = ops.moe_ffn_fp8(xq, self.w13, self.w13_scale, self.w2,
                                self.w2_scale, ones,
                                local.view(-1, 1).to(torch.int32))
        elif not fp8_wire and x.is_cuda and ops.have_native():
            ones = torch.ones(ep * C, 1, device=dev, dtype=torch.float32)
            y = ops.moe_ffn(recv_x, self.w13, self.w2, ones,
                            local.view(-1, 1).to(torch.int32))
        else:
            if fp8_wire:    # CPU/EP test path: dequantize the wire bytes
                recv_x = (recv_q.view(torch.float8_e4m3fn).float()
                          * recv_s[:, None]).to(x.dtype)
            y = torch.empty_like(recv_x)
            for e in range(self.experts_per_rank):
                m = local == e
                if bool(m.any()):
                    y[m] = self._expert_ffn(recv_x[m], e)
        back = torch.empty_like(y)
        dist.all_to_all_single(back, y, group=group)
        # gather each kept assignment's result back; dropped ones get w=0
        gidx = torch.where(keep, dest, torch.zeros_like(dest))
        w = weights.reshape(-1)[order] * keep.to(weights.dtype)
        out = torch.zeros_like(x)
        out.index_add_(0, src_rows, back[gidx] * w[:, None])
        return out


class _MoEAdapter(nn.Module):
    """Matches the LlamaMLP call signature inside DecoderLayer."""

    def __init__(self, spec: ModelSpec, layer_idx: int, dtype: torch.dtype):
        super().__init__()
        self.moe = MixtralMoE(spec, layer_idx, dtype)

    def forward(self, x):
        return self.moe(x)


class MixtralForCausalLM(LlamaForCausalLM):
    mlp_cls = _MoEAdapter
