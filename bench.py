#!/usr/bin/env python3
"""Flagship serving benchmark — BASELINE.json north-star metric.

Measures steady-state decode throughput (tokens/s, whole-job aggregate) of
the llm-gateway engine on Llama-3-8B bf16, synthetic prompts, random-init
weights, plus p50 TTFT (prefill latency of one prompt).

Default (driver contract): one engine replica per GPU rank, TP=1 per
replica (weak scaling — per-GPU work fixed as N grows; the idiomatic
serving scale-out).  ``--tp N`` instead runs ONE engine tensor-parallel
over all N ranks (strong scaling; used for the 70B TP=8 config).

Launch (driver): python -m torch.distributed.run --nnodes=1
  --nproc-per-node N --master-addr 127.0.0.1 bench.py --gpus N ...
"""

from __future__ import annotations

import argparse
import json
import os
import statistics
import time

import torch
import torch.distributed as dist


def log(rank, msg):
    if rank == 0:
        print(f"[bench] {msg}", flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=64)
    ap.add_argument("--warmup", type=int, default=8)
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--tp", type=int, default=1,
                    help=">1: one tensor-parallel engine over all ranks")
    ap.add_argument("--batch", type=int, default=2048)
    ap.add_argument("--prompt-len", type=int, default=512)
    ap.add_argument("--eager", action="store_true")
    ap.add_argument("--quant", default=None, choices=[None, "fp8"],
                    help="fp8: e4m3fn weights + per-token act scales "
                         "(separate config line, NOT the bf16 headline)")
    ap.add_argument("--kv-dtype", default="bfloat16",
                    choices=["bfloat16", "fp8"],
                    help="paged KV cache dtype (fp8 halves attention HBM)")
    ap.add_argument("--device", default=None)
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    on_gpu = torch.cuda.is_available() and args.device != "cpu"
    if on_gpu:
        torch.cuda.set_device(local_rank)

    from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams
    from hyperspot.parallel.state import initialize_model_parallel

    if world > 1:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        dist.init_process_group("nccl" if on_gpu else "gloo")
    tp = args.tp if args.tp > 1 else 1
    if tp > 1:
        assert world == tp, "TP mode needs world == tp"
        initialize_model_parallel(tp_size=tp)

    max_len = args.prompt_len + args.warmup + args.steps + 64
    cfg = EngineConfig(
        model=args.model, max_num_seqs=args.batch,
        max_num_batched_tokens=max(8192, args.prompt_len),
        max_model_len=max_len, enforce_eager=args.eager or not on_gpu,
        tp_size=tp, quant=args.quant, kv_dtype=args.kv_dtype,
        seed=1234 + (0 if tp > 1 else rank))
    t0 = time.monotonic()
    eng = LLMEngine(cfg, device=args.device)
    log(rank, f"engine init {time.monotonic() - t0:.1f}s "
              f"(model={args.model}, blocks={eng.runner.num_blocks})")

    # ---- TTFT: single-request prefill latency (p50 of 5) ----
    g = torch.Generator().manual_seed(7)
    vocab = cfg.spec().vocab_size
    def mk_prompt(n):
        return torch.randint(0, vocab, (n,), generator=g).tolist()
    ttfts = []
    for _ in range(5):
        rid = eng.add_request(mk_prompt(args.prompt_len),
                              SamplingParams(temperature=0.0, max_tokens=1))
        t = time.monotonic()
        while eng.has_work():
            eng.step()
        if on_gpu:
            torch.cuda.synchronize()
        ttfts.append((time.monotonic() - t) * 1000)
    ttft_p50 = statistics.median(ttfts)
    log(rank, f"ttft p50 {ttft_p50:.1f} ms")

    # ---- throughput: saturate with args.batch concurrent sequences ----
    sp = SamplingParams(temperature=0.0, max_tokens=10 ** 9)
    for i in range(args.batch):
        eng.add_request(mk_prompt(args.prompt_len), sp)
    t = time.monotonic()
    while eng.num_waiting > 0:          # run all prefills
        eng.step()
    if on_gpu:
        torch.cuda.synchronize()
    log(rank, f"prefill of {args.batch} x {args.prompt_len} done "
              f"({time.monotonic() - t:.1f}s); capturing graphs")
    if not cfg.enforce_eager:
        eng.capture_graphs()

    for _ in range(args.warmup):
        eng.step()
    if on_gpu:
        torch.cuda.synchronize()
    if world > 1:
        dist.barrier()

    t0 = time.monotonic()
    produced = 0
    for _ in range(args.steps):
        produced += len(eng.step())
    if on_gpu:
        torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    elapsed = time.monotonic() - t0

    # max over ranks
    if world > 1:
        e = torch.tensor([elapsed])
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = float(e[0])
        p = torch.tensor([produced], dtype=torch.long)
        dist.all_reduce(p)
        produced = int(p[0])
    else:
        produced = produced

    n_gpus = world if world > 1 else args.gpus
    value = produced / elapsed
    ms_per_step = elapsed / args.steps * 1000
    if rank == 0:
        out = {
            "metric": "tokens/sec (llm-gateway decode throughput)",
            "value": round(value, 2),
            "unit": "tokens/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "strong" if tp > 1 else "weak",
            "vs_baseline": None,
            "dtype": (((args.quant or "bf16")
                       + ("+fp8kv" if args.kv_dtype == "fp8" else ""))
                      if on_gpu else "fp32(cpu-dev-run)"),
            "data": "synthetic",
            "ttft_ms_p50": round(ttft_p50, 2),
            "config": {
                "model": args.model,
                "global_batch": args.batch * (1 if tp > 1 else n_gpus),
                "seq_len": args.prompt_len,
                "parallelism": (f"tp{tp}" if tp > 1 else f"dp{n_gpus}"),
            },
        }
        print(json.dumps(out), flush=True)
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
