"""CPU tests of the torch reference ops (the numerics ground truth that the
gfx950 HIP kernels are compared against in tests/test_ops_gpu.py)."""

import math

import pytest
import torch

from hyperspot.ops import torch_ref as R


def test_rmsnorm_matches_manual():
    torch.manual_seed(0)
    x = torch.randn(7, 64)
    w = torch.randn(64)
    out = R.rmsnorm(x, w, 1e-5)
    ref = x / torch.sqrt((x ** 2).mean(-1, keepdim=True) + 1e-5) * w
    assert torch.allclose(out, ref, atol=1e-5)


def test_fused_add_rmsnorm():
    torch.manual_seed(0)
    x, res = torch.randn(5, 32), torch.randn(5, 32)
    w = torch.randn(32)
    out, new_res = R.fused_add_rmsnorm(x, res, w, 1e-5)
    assert torch.allclose(new_res, x + res, atol=1e-6)
    assert torch.allclose(out, R.rmsnorm(x + res, w, 1e-5), atol=1e-6)


def test_rope_preserves_norm_and_position0_identity():
    torch.manual_seed(0)
    q = torch.randn(4, 2, 32)
    k = torch.randn(4, 1, 32)
    pos = torch.tensor([0, 1, 5, 100])
    q2, k2 = R.apply_rope(q.clone(), k.clone(), pos, theta=10000.0)
    # position 0 is identity
    assert torch.allclose(q2[0], q[0], atol=1e-6)
    # rotation preserves pairwise norms
    d2 = 16
    n1 = (q[..., :d2] ** 2 + q[..., d2:] ** 2)
    n2 = (q2[..., :d2] ** 2 + q2[..., d2:] ** 2)
    assert torch.allclose(n1, n2, atol=1e-4)


def test_rope_relative_property():
    # <rope(q,m), rope(k,n)> depends only on m-n
    torch.manual_seed(1)
    q = torch.randn(1, 1, 64)
    k = torch.randn(1, 1, 64)
    def dot(m, n):
        qm, kn = R.apply_rope(q.clone(), k.clone(),
                              torch.tensor([0]), theta=10000.0)
        q1, _ = R.apply_rope(q.clone(), k.clone(), torch.tensor([m]), 10000.0)
        _, k1 = R.apply_rope(q.clone(), k.clone(), torch.tensor([n]), 10000.0)
        return float((q1 * k1).sum())
    assert math.isclose(dot(5, 3), dot(12, 10), rel_tol=1e-4)


def test_kv_append_and_paged_decode_match_dense_attention():
    torch.manual_seed(0)
    KV, BS, D, H = 2, 4, 16, 4
    n = 10
    k_cache = torch.zeros(8, KV, BS, D)
    v_cache = torch.zeros(8, KV, BS, D)
    k = torch.randn(n, KV, D)
    v = torch.randn(n, KV, D)
    blocks = [3, 0, 5]
    slots = torch.tensor([blocks[i // BS] * BS + i % BS for i in range(n)])
    R.kv_cache_append(k, v, k_cache, v_cache, slots)
    q = torch.randn(1, H, D)
    bt = torch.tensor([blocks], dtype=torch.int32)
    out = R.paged_attn_decode(q, k_cache, v_cache, bt,
                              torch.tensor([n], dtype=torch.int32),
                              scale=D ** -0.5)
    # dense reference
    g = H // KV
    qf = q[0].view(KV, g, D)
    att = torch.einsum("kgd,nkd->kgn", qf, k) * D ** -0.5
    p = att.softmax(-1)
    ref = torch.einsum("kgn,nkd->kgd", p, v).reshape(H, D)
    assert torch.allclose(out[0], ref, atol=1e-5)


def test_prefill_attn_varlen_causal():
    torch.manual_seed(0)
    H, KV, D = 4, 2, 16
    lens = [5, 3]
    T = sum(lens)
    q = torch.randn(T, H, D)
    k = torch.randn(T, KV, D)
    v = torch.randn(T, KV, D)
    ss = torch.tensor([0, 5, 8], dtype=torch.int32)
    out = R.prefill_attn(q, k, v, ss, scale=D ** -0.5)
    # per-position dense causal reference
    g = H // KV
    for s, e in [(0, 5), (5, 8)]:
        for i in range(s, e):
            qf = q[i].view(KV, g, D)
            ks, vs = k[s:i + 1], v[s:i + 1]
            att = torch.einsum("kgd,nkd->kgn", qf, ks) * D ** -0.5
            ref = torch.einsum("kgn,nkd->kgd", att.softmax(-1), vs).reshape(H, D)
            assert torch.allclose(out[i], ref, atol=1e-5), i


def test_silu_mul():
    x = torch.randn(3, 8)
    out = R.silu_mul(x)
    g, u = x[:, :4], x[:, 4:]
    assert torch.allclose(out, torch.nn.functional.silu(g) * u, atol=1e-6)


def test_sample_greedy_and_temperature():
    torch.manual_seed(0)
    logits = torch.randn(4, 50)
    t0 = R.sample(logits, torch.zeros(4), torch.ones(4),
                  torch.zeros(4, dtype=torch.int32), torch.rand(4))
    assert torch.equal(t0, logits.argmax(-1))
    # temperature sampling with u -> 0 picks the first index with mass;
    # statistically, low temperature concentrates on argmax
    hits = 0
    for trial in range(50):
        u = torch.rand(4)
        t = R.sample(logits, torch.full((4,), 0.01), torch.ones(4),
                     torch.zeros(4, dtype=torch.int32), u)
        hits += int((t == logits.argmax(-1)).sum())
    assert hits > 190  # 200 draws, nearly all at the mode


def test_sample_top_k_restricts_support():
    torch.manual_seed(0)
    logits = torch.randn(2, 100)
    topk = logits.topk(5, dim=-1).indices
    for _ in range(20):
        t = R.sample(logits, torch.ones(2), torch.ones(2),
                     torch.full((2,), 5, dtype=torch.int32), torch.rand(2))
        for b in range(2):
            assert t[b] in topk[b]


def test_sample_top_p_restricts_support():
    logits = torch.tensor([[10.0, 9.0, -5.0, -5.0, -5.0]])
    for _ in range(20):
        t = R.sample(logits, torch.ones(1), torch.tensor([0.9]),
                     torch.zeros(1, dtype=torch.int32), torch.rand(1))
        assert t[0] in (0, 1)
