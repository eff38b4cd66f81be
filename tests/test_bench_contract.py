"""Driver-contract smoke test for bench.py (REST mode, CPU dev-run).

The driver runs `python bench.py --gpus N --steps K --warmup W` and parses
ONE JSON line from stdout; this guards the contract on every CPU CI run.
"""

import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_rest_json_line():
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"),
         "--steps", "4", "--warmup", "1", "--ttft-iters", "3"],
        cwd=ROOT, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    j = json.loads(lines[0])
    assert j["metric"].startswith("tokens/sec + p50 TTFT via llm-gateway"
                                  " REST")
    assert j["unit"] == "tokens/s"
    assert j["value"] > 0
    assert j["steps"] == 4 and j["warmup"] == 1
    assert j["ms_per_step"] > 0
    assert j["higher_is_better"] is True
    assert j["scaling"] == "weak"
    assert j["data"] == "synthetic"
    assert j["ttft_ms_p50"] > 0
    assert set(j["config"]) >= {"model", "global_batch", "seq_len",
                                "parallelism"}
    # REST-vs-engine overhead is quantified (VERDICT round-1 item 2)
    assert "engine_tokens_per_s" in j and "gateway_overhead_pct" in j


def test_bench_rest_two_rank_launch():
    """The driver's N>1 shape: torchrun two ranks, rank 0 orchestrates
    the host with a 2-worker DP fleet, barriers synchronize ranks."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29713",
         os.path.join(ROOT, "bench.py"), "--gpus", "2",
         "--steps", "4", "--warmup", "1", "--ttft-iters", "2"],
        cwd=ROOT, capture_output=True, text=True, timeout=420)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    j = json.loads(lines[0])
    assert j["n_gpus"] == 2
    assert j["config"]["parallelism"].startswith("dp2")
    assert j["value"] > 0


def test_bench_engine_direct_json_line():
    """--engine-direct (the rocprof/kernel-work mode) keeps the same
    one-JSON-line contract."""
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"),
         "--engine-direct", "--model", "tiny-llama", "--batch", "8",
         "--prompt-len", "32", "--steps", "2", "--warmup", "1",
         "--ttft-iters", "1"],
        cwd=ROOT, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    j = json.loads(lines[0])
    assert j["metric"].startswith("tokens/sec (engine-direct")
    assert j["steps"] == 2 and j["value"] > 0 and j["ms_per_step"] > 0
    assert j["config"]["model"] == "tiny-llama"
