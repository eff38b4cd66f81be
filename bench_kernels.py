#!/usr/bin/env python3
"""Isolated kernel microbenchmarks (GPU only) — us/call + achieved GB/s.

Usage: python bench_kernels.py [--which attn|gemm|all] [--iters N]

Shapes mirror the flagship bench (Llama-3-8B bf16, bs=256 ctx=512 decode):
the numbers here are the per-kernel budget lines behind bench.py's step
time, compared against the HBM3E stream roofline (~8 TB/s).
"""

from __future__ import annotations

import argparse
import json

import torch


def timed(fn, iters, warmup=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) * 1e3 / iters  # us/call


def bench_attn(iters):
    import hyperspot.ops as ops
    from hyperspot.ops import torch_ref
    torch.manual_seed(0)
    dev = "cuda:0"
    for (bs, ctx, kvh, group, D) in [(256, 512, 8, 4, 128),
                                     (256, 2048, 8, 4, 128),
                                     (64, 512, 8, 4, 128),
                                     (256, 512, 1, 8, 128),
                                     (64, 8192, 8, 4, 128),
                                     (16, 32768, 8, 4, 128)]:
        BS = 16
        nblk = (ctx + BS - 1) // BS
        tot_blocks = bs * nblk + 7
        k_cache = torch.randn(tot_blocks, kvh, BS, D, dtype=torch.bfloat16,
                              device=dev) * 0.3
        v_cache = torch.randn_like(k_cache) * 0.3
        q = torch.randn(bs, kvh * group, D, dtype=torch.bfloat16,
                        device=dev) * 0.5
        perm = torch.randperm(bs * nblk, device=dev, dtype=torch.int32)
        block_tables = perm.view(bs, nblk).contiguous()
        seq_lens = torch.full((bs,), ctx, dtype=torch.int32, device=dev)
        scale = D ** -0.5

        out = ops.paged_attn_decode(q, k_cache, v_cache, block_tables,
                                    seq_lens, scale)
        ref = torch_ref.paged_attn_decode(
            q.float().cpu(), k_cache.float().cpu(), v_cache.float().cpu(),
            block_tables.cpu(), seq_lens.cpu(), scale)
        err = (out.float().cpu() - ref).abs().max().item()
        us = timed(lambda: ops.paged_attn_decode(
            q, k_cache, v_cache, block_tables, seq_lens, scale), iters)
        kv_bytes = bs * ctx * kvh * D * 2 * 2
        roof_us = kv_bytes / 8e12 * 1e6
        print(json.dumps({
            "kernel": "paged_attn_decode",
            "shape": f"bs{bs}_ctx{ctx}_kvh{kvh}_g{group}_d{D}",
            "us": round(us, 1), "GBps": round(kv_bytes / us / 1e3, 0),
            "roofline_us": round(roof_us, 1),
            "x_roofline": round(us / roof_us, 2),
            "max_abs_err": round(err, 5)}), flush=True)


def bench_gemm(iters):
    dev = "cuda:0"
    for (m, k, n, tag) in [(256, 4096, 6144, "qkv"),
                           (256, 4096, 4096, "o"),
                           (256, 4096, 28672, "gate_up"),
                           (256, 14336, 4096, "down"),
                           (64, 4096, 6144, "qkv_bs64"),
                           (8192, 4096, 28672, "prefill_gate_up")]:
        x = torch.randn(m, k, dtype=torch.bfloat16, device=dev)
        w = torch.randn(n, k, dtype=torch.bfloat16, device=dev)
        us = timed(lambda: torch.nn.functional.linear(x, w), iters)
        bytes_ = (m * k + n * k + m * n) * 2
        flops = 2 * m * k * n
        print(json.dumps({
            "kernel": f"linear_{tag}", "shape": f"{m}x{k}x{n}",
            "us": round(us, 1),
            "GBps": round(bytes_ / us / 1e3, 0),
            "TFs": round(flops / us / 1e6, 0),
            "roofline_us": round(max(bytes_ / 8e12, flops / 2.5e15) * 1e6,
                                 1)}), flush=True)


def bench_fp8_gemm(iters):
    from hyperspot.parallel.layers import quant_fp8_rowwise, quantize_weight_fp8
    dev = "cuda:0"
    for (m, k, n, tag) in [(1024, 4096, 6144, "qkv"),
                           (1024, 4096, 28672, "gate_up"),
                           (1024, 14336, 4096, "down"),
                           (256, 4096, 28672, "gate_up_bs256")]:
        x = torch.randn(m, k, dtype=torch.bfloat16, device=dev)
        w = torch.randn(n, k, dtype=torch.bfloat16, device=dev) * 0.02
        wq, ws = quantize_weight_fp8(w)
        ws_row = ws[None, :].contiguous()
        us_bf16 = timed(lambda: torch.nn.functional.linear(x, w), iters)

        def fp8_full():
            xq, xs = quant_fp8_rowwise(x)
            return torch._scaled_mm(xq, wq.t(), scale_a=xs[:, None],
                                    scale_b=ws_row, out_dtype=torch.bfloat16)
        us_fp8 = timed(fp8_full, iters)
        xq, xs = quant_fp8_rowwise(x)
        xs_col = xs[:, None].contiguous()
        us_fp8_mm = timed(lambda: torch._scaled_mm(
            xq, wq.t(), scale_a=xs_col, scale_b=ws_row,
            out_dtype=torch.bfloat16), iters)
        print(json.dumps({
            "kernel": f"fp8_{tag}", "shape": f"{m}x{k}x{n}",
            "us_bf16": round(us_bf16, 1), "us_fp8_withquant": round(us_fp8, 1),
            "us_fp8_mm_only": round(us_fp8_mm, 1)}), flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--which", default="attn")
    ap.add_argument("--iters", type=int, default=100)
    args = ap.parse_args()
    assert torch.cuda.is_available()
    if args.which in ("attn", "all"):
        bench_attn(args.iters)
    if args.which in ("gemm", "all"):
        bench_gemm(args.iters)
    if args.which in ("fp8", "all"):
        bench_fp8_gemm(args.iters)


if __name__ == "__main__":
    main()
