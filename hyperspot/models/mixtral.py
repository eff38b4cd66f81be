"""Mixtral 8x7B — MoE decoder with expert parallelism over RCCL all-to-all.

Sharding model (MI355X-first):
  - attention is tensor-parallel exactly like Llama
  - experts are EXPERT-parallel: each rank owns num_experts/ep of the FFNs
    whole (no intra-expert sharding), tokens travel rank->expert-owner->rank
    via two all-to-alls per layer (dispatch, combine).  With EP==1 (or on
    CPU tests) the same code runs the local grouped-GEMM path.
"""

from __future__ import annotations

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
        directly on the owner rank."""
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


class _MoEAdapter(nn.Module):
    """Matches the LlamaMLP call signature inside DecoderLayer."""

    def __init__(self, spec: ModelSpec, layer_idx: int, dtype: torch.dtype):
        super().__init__()
        self.moe = MixtralMoE(spec, layer_idx, dtype)

    def forward(self, x):
        return self.moe(x)


class MixtralForCausalLM(LlamaForCausalLM):
    mlp_cls = _MoEAdapter
