"""Pure-PyTorch reference implementations of every engine op.

These are the numerics ground truth for the CDNA4 HIP kernels (tests compare
the gfx950 kernels against these in fp32) and the CPU execution path for
CI — this container has no GPU.  They are written for clarity, not speed.

Conventions shared with the HIP kernels (csrc/):
  - token-major activations: x is [T, hidden] (T = flattened tokens of the
    whole batch, prefill + decode mixed)
  - q/k/v are [T, n_heads, head_dim] after the qkv projection split
  - paged KV cache: k_cache/v_cache are [num_blocks, kv_heads, block, head_dim]
  - slot_mapping[t] = block_id * block_size + offset for token t
  - RoPE is GPT-NeoX style (rotate halves, not interleaved pairs)
"""

from __future__ import annotations

import torch


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    dt = x.dtype
    xf = x.float()
    out = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (out * weight.float()).to(dt)


def fused_add_rmsnorm(x: torch.Tensor, residual: torch.Tensor,
                      weight: torch.Tensor, eps: float):
    """residual += x; return (rmsnorm(residual), residual)."""
    res = (residual.float() + x.float())
    out = res * torch.rsqrt(res.pow(2).mean(-1, keepdim=True) + eps)
    out = (out * weight.float()).to(x.dtype)
    return out, res.to(x.dtype)


def rope_cos_sin(positions: torch.Tensor, head_dim: int, theta: float,
                 dtype: torch.dtype = torch.float32) -> torch.Tensor:
    """Return [T, head_dim] table: first half cos, second half sin."""
    inv = 1.0 / (theta ** (torch.arange(0, head_dim, 2, dtype=torch.float32,
                                        device=positions.device) / head_dim))
    freqs = positions.float()[:, None] * inv[None, :]          # [T, D/2]
    return torch.cat([freqs.cos(), freqs.sin()], dim=-1).to(dtype)


def _rotate_neox(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor):
    # x: [T, H, D]; cos/sin: [T, D/2]
    d2 = x.shape[-1] // 2
    x1, x2 = x[..., :d2].float(), x[..., d2:].float()
    c, s = cos[:, None, :], sin[:, None, :]
    return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1).to(x.dtype)


def apply_rope(q: torch.Tensor, k: torch.Tensor, positions: torch.Tensor,
               theta: float):
    """In-place-semantics RoPE on q [T,H,D] and k [T,KV,D]."""
    d = q.shape[-1]
    table = rope_cos_sin(positions, d, theta)
    cos, sin = table[:, : d // 2], table[:, d // 2:]
    return _rotate_neox(q, cos, sin), _rotate_neox(k, cos, sin)


def kv_cache_append(k: torch.Tensor, v: torch.Tensor,
                    k_cache: torch.Tensor, v_cache: torch.Tensor,
                    slot_mapping: torch.Tensor) -> None:
    """Scatter k/v [T, KV, D] into paged caches [NB, KV, BS, D]."""
    nb, kvh, bs, d = k_cache.shape
    blk = slot_mapping // bs
    off = slot_mapping % bs
    k_cache[blk, :, off] = k.to(k_cache.dtype)
    v_cache[blk, :, off] = v.to(v_cache.dtype)


def paged_attn_decode(q: torch.Tensor, k_cache: torch.Tensor,
                      v_cache: torch.Tensor, block_tables: torch.Tensor,
                      seq_lens: torch.Tensor, scale: float) -> torch.Tensor:
    """One-token-per-sequence attention over the paged cache.

    q: [B, H, D]; block_tables: [B, max_blocks] int32; seq_lens: [B] int32
    (length INCLUDING the current token, whose k/v are already in the cache).
    """
    B, H, D = q.shape
    nb, KV, BS, _ = k_cache.shape
    group = H // KV
    out = torch.empty_like(q)
    for b in range(B):
        n = int(seq_lens[b])
        nblk = (n + BS - 1) // BS
        blocks = block_tables[b, :nblk].long()
        k = k_cache[blocks].permute(1, 0, 2, 3).reshape(KV, nblk * BS, D)[:, :n]
        v = v_cache[blocks].permute(1, 0, 2, 3).reshape(KV, nblk * BS, D)[:, :n]
        qb = q[b].float().view(KV, group, D)                   # [KV, G, D]
        att = torch.einsum("kgd,knd->kgn", qb, k.float()) * scale
        p = att.softmax(-1)
        o = torch.einsum("kgn,knd->kgd", p, v.float())
        out[b] = o.reshape(H, D).to(q.dtype)
    return out


def prefill_attn_paged(q: torch.Tensor, k_cache: torch.Tensor,
                       v_cache: torch.Tensor, block_tables: torch.Tensor,
                       ctx_lens: torch.Tensor, row_seq: torch.Tensor,
                       scale: float) -> torch.Tensor:
    """Per-row causal attention over the paged cache (chunked-prefill
    continuation reference: row t attends slots 0..ctx_lens[t])."""
    T, H, D = q.shape
    nb, KV, BS, _ = k_cache.shape
    group = H // KV
    out = torch.empty_like(q)
    for t in range(T):
        si = int(row_seq[t])
        c = int(ctx_lens[t])
        nblk = (c + BS - 1) // BS
        blocks = block_tables[si, :nblk].long()
        k = k_cache[blocks].permute(1, 0, 2, 3).reshape(KV, nblk * BS,
                                                        D)[:, :c].float()
        v = v_cache[blocks].permute(1, 0, 2, 3).reshape(KV, nblk * BS,
                                                        D)[:, :c].float()
        qt = q[t].float().view(KV, group, D)
        att = torch.einsum("kgd,knd->kgn", qt, k) * scale
        p = att.softmax(-1)
        o = torch.einsum("kgn,knd->kgd", p, v)
        out[t] = o.reshape(H, D).to(q.dtype)
    return out


def prefill_attn(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                 seq_start: torch.Tensor, scale: float) -> torch.Tensor:
    """Varlen causal self-attention for prefill.

    q [T,H,D], k/v [T,KV,D]; seq_start: [num_seqs+1] int32 cumulative offsets.
    """
    T, H, D = q.shape
    KV = k.shape[1]
    group = H // KV
    out = torch.empty_like(q)
    for i in range(seq_start.numel() - 1):
        s, e = int(seq_start[i]), int(seq_start[i + 1])
        n = e - s
        qs = q[s:e].float().view(n, KV, group, D)
        ks = k[s:e].float()
        vs = v[s:e].float()
        att = torch.einsum("mkgd,nkd->kgmn", qs, ks) * scale
        mask = torch.full((n, n), float("-inf"), device=q.device).triu(1)
        att = att + mask
        p = att.softmax(-1)
        o = torch.einsum("kgmn,nkd->mkgd", p, vs)
        out[s:e] = o.reshape(n, H, D).to(q.dtype)
    return out


def silu_mul(gate_up: torch.Tensor) -> torch.Tensor:
    """SwiGLU activation: input [T, 2*I] (gate | up) -> [T, I]."""
    i = gate_up.shape[-1] // 2
    g, u = gate_up[..., :i].float(), gate_up[..., i:].float()
    return (torch.nn.functional.silu(g) * u).to(gate_up.dtype)


def sample(logits: torch.Tensor, temperature: torch.Tensor,
           top_p: torch.Tensor, top_k: torch.Tensor,
           uniform: torch.Tensor) -> torch.Tensor:
    """Per-row temperature / top-k / top-p sampling.

    logits [B, V] float; temperature/top_p [B] float; top_k [B] int
    (0 = off); uniform [B] in [0,1) drives the categorical draw so the HIP
    kernel and the reference are comparable given the same randoms.
    temperature == 0 selects greedy argmax for that row.
    """
    B, V = logits.shape
    out = torch.empty(B, dtype=torch.long, device=logits.device)
    lf = logits.float()
    for b in range(B):
        t = float(temperature[b])
        if t == 0.0:
            out[b] = int(lf[b].argmax())
            continue
        row = lf[b] / t
        k = int(top_k[b])
        if 0 < k < V:
            kth = row.topk(k).values[-1]
            row = row.masked_fill(row < kth, float("-inf"))
        probs = row.softmax(-1)
        p = float(top_p[b])
        if p < 1.0:
            sp, idx = probs.sort(descending=True)
            cum = sp.cumsum(-1)
            keep = cum - sp < p          # keep tokens until cumulative >= p
            sp = sp * keep
            sp = sp / sp.sum()
            # inverse-CDF draw on the sorted distribution
            c = sp.cumsum(-1)
            j = int(torch.searchsorted(c, uniform[b].to(c.dtype), right=False).clamp(max=V - 1))
            out[b] = int(idx[j])
        else:
            c = probs.cumsum(-1)
            j = int(torch.searchsorted(c, uniform[b].to(c.dtype), right=False).clamp(max=V - 1))
            out[b] = j
    return out


def moe_route(hidden: torch.Tensor, router_w: torch.Tensor, top_k: int):
    """Softmax-topk routing: returns (weights [T,K], expert_ids [T,K])."""
    logits = hidden.float() @ router_w.float().t()
    probs = logits.softmax(-1)
    w, ids = probs.topk(top_k, dim=-1)
    w = w / w.sum(-1, keepdim=True)
    return w, ids
