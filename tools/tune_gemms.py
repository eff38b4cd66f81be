#!/usr/bin/env python3
"""Offline hipBLASLt algorithm tuning for the serving GEMM shapes (GPU).

Runs PyTorch TunableOp tuning over every GEMM the engine issues (decode at
the bucketed batch sizes, chunked prefill, TTFT single-prompt prefill,
logits head) and writes profiles/tunableop_gfx950.csv, which ModelRunner
loads read-only at engine init.

Run on an MI355X box:  python tools/tune_gemms.py [--out FILE]
"""

from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

ap = argparse.ArgumentParser()
ap.add_argument("--out", default="profiles/tunableop_gfx950.csv")
ap.add_argument("--include-70b", action="store_true")
args = ap.parse_args()

# TunableOp reads its env at tuning-context construction — set everything
# before torch loads.
os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
os.environ["PYTORCH_TUNABLEOP_TUNING"] = "1"
os.environ["PYTORCH_TUNABLEOP_FILENAME"] = args.out
os.environ["PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS"] = "10"
os.environ["PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS"] = "30"
os.environ["PYTORCH_TUNABLEOP_VERBOSE"] = "1"

import torch  # noqa: E402

# Llama-3-8B
SHAPES_8B = dict(hidden=4096, inter=14336, q=4096, kv=1024, vocab=128256)
# Llama-3-70B at TP=8 (per-rank shards) and TP=1 (single 288GB GPU)
SHAPES_70B_TP8 = dict(hidden=8192, inter=28672 // 8, q=8192 // 8,
                      kv=1024 // 8, vocab=128256 // 8)
SHAPES_70B_TP1 = dict(hidden=8192, inter=28672, q=8192, kv=1024,
                      vocab=128256)


def gemm_shapes(include_70b=False):
    ms = [64, 128, 256, 384, 512, 768, 1024, 2048, 8192]
    specs = (SHAPES_8B, SHAPES_70B_TP8, SHAPES_70B_TP1) if include_70b         else (SHAPES_8B,)
    for spec in specs:
        h, it = spec["hidden"], spec["inter"]
        nqkv = spec["q"] + 2 * spec["kv"]
        for m in ms:
            yield m, h, nqkv            # qkv proj
            yield m, spec["q"], h       # o proj
            yield m, h, 2 * it          # gate_up
            yield m, it, h              # down
        for m in [64, 256, 512, 1024]:
            yield m, h, spec["vocab"]   # logits head


def main():
    assert torch.cuda.is_available()
    dev = "cuda:0"
    seen = set()
    shapes = [s for s in gemm_shapes(args.include_70b)
              if not (s in seen or seen.add(s))]
    for i, (m, k, n) in enumerate(shapes):
        x = torch.randn(m, k, dtype=torch.bfloat16, device=dev)
        w = torch.randn(n, k, dtype=torch.bfloat16, device=dev)
        torch.nn.functional.linear(x, w)   # tuning happens on first call
        torch.cuda.synchronize()
        print(f"[{i + 1}/{len(shapes)}] tuned {m}x{k}x{n}", flush=True)
        del x, w
    # fp8 path (ScaledGemm): same shapes, e4m3fn + rowwise scales
    from hyperspot.parallel.layers import (quant_fp8_rowwise,
                                           quantize_weight_fp8)
    for i, (m, k, n) in enumerate(shapes):
        x = torch.randn(m, k, dtype=torch.bfloat16, device=dev)
        w = torch.randn(n, k, dtype=torch.bfloat16, device=dev) * 0.02
        wq, ws = quantize_weight_fp8(w)
        xq, xs = quant_fp8_rowwise(x)
        torch._scaled_mm(xq, wq.t(), scale_a=xs[:, None].contiguous(),
                         scale_b=ws[None, :].contiguous(),
                         out_dtype=torch.bfloat16)
        torch.cuda.synchronize()
        print(f"[fp8 {i + 1}/{len(shapes)}] tuned {m}x{k}x{n}", flush=True)
        del x, w, wq, xq
    # results are flushed by TunableOp at interpreter exit; make sure the
    # directory exists so the write succeeds
    d = os.path.dirname(args.out)
    if d:
        os.makedirs(d, exist_ok=True)
    print("tuning done; results flush to", args.out, "at exit")


if __name__ == "__main__":
    main()
