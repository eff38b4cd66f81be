"""Model and engine configuration.

The reference (cyberfabric/cyberfabric-core) ships no model code; the model set
here is the one named by BASELINE.json: Llama-3-8B (TP=1), Llama-3-70B (TP=8),
Mixtral 8x7B (EP).  Architecture hyper-parameters follow the public
architectures; weights are random-init (no network in this environment).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional


@dataclass
class ModelSpec:
    """Architecture description for a decoder-only transformer."""

    name: str
    hidden_size: int
    num_layers: int
    num_heads: int
    num_kv_heads: int
    head_dim: int
    intermediate_size: int
    vocab_size: int
    rope_theta: float = 500000.0
    rms_eps: float = 1e-5
    max_position: int = 8192
    tie_embeddings: bool = False
    # MoE (0 experts => dense MLP)
    num_experts: int = 0
    num_experts_per_tok: int = 0
    dtype: str = "bfloat16"

    @property
    def is_moe(self) -> bool:
        return self.num_experts > 0

    def param_count(self) -> int:
        h, l, i, v = self.hidden_size, self.num_layers, self.intermediate_size, self.vocab_size
        qkv = h * (self.num_heads + 2 * self.num_kv_heads) * self.head_dim
        o = self.num_heads * self.head_dim * h
        if self.is_moe:
            mlp = 3 * h * i * self.num_experts + h * self.num_experts
        else:
            mlp = 3 * h * i
        per_layer = qkv + o + mlp + 2 * h
        emb = v * h * (1 if self.tie_embeddings else 2)
        return l * per_layer + emb + h


# Preset registry (keys are the canonical model ids used by the model-registry
# module; reference modules/model-registry/docs/PRD.md:11 canonical id format).
_PRESETS = {
    "llama3-8b": ModelSpec(
        name="llama3-8b", hidden_size=4096, num_layers=32, num_heads=32,
        num_kv_heads=8, head_dim=128, intermediate_size=14336,
        vocab_size=128256, rope_theta=500000.0,
    ),
    "llama3-70b": ModelSpec(
        name="llama3-70b", hidden_size=8192, num_layers=80, num_heads=64,
        num_kv_heads=8, head_dim=128, intermediate_size=28672,
        vocab_size=128256, rope_theta=500000.0,
    ),
    "mixtral-8x7b": ModelSpec(
        name="mixtral-8x7b", hidden_size=4096, num_layers=32, num_heads=32,
        num_kv_heads=8, head_dim=128, intermediate_size=14336,
        vocab_size=32000, rope_theta=1000000.0, num_experts=8,
        num_experts_per_tok=2,
    ),
    # Tiny models for CPU tests / CI (same code paths, small shapes).
    "tiny-llama": ModelSpec(
        name="tiny-llama", hidden_size=256, num_layers=2, num_heads=4,
        num_kv_heads=2, head_dim=64, intermediate_size=512, vocab_size=512,
        rope_theta=10000.0, max_position=1024,
    ),
    "tiny-moe": ModelSpec(
        name="tiny-moe", hidden_size=256, num_layers=2, num_heads=4,
        num_kv_heads=2, head_dim=64, intermediate_size=256, vocab_size=512,
        rope_theta=10000.0, max_position=1024, num_experts=4,
        num_experts_per_tok=2,
    ),
}


def get_model_spec(name: str) -> ModelSpec:
    key = name.lower()
    if key not in _PRESETS:
        raise KeyError(f"unknown model preset '{name}' (have: {sorted(_PRESETS)})")
    return _PRESETS[key]


@dataclass
class EngineConfig:
    """Engine/runtime configuration (the serverless-runtime worker side).

    KV sizing targets 288 GB HBM3E per MI355X GPU: after weights, the pool
    takes ``gpu_memory_utilization`` of free memory in BLOCK_SIZE-token pages.
    """

    model: str = "llama3-8b"
    block_size: int = 16                 # tokens per KV page
    max_num_seqs: int = 256              # max concurrent sequences
    max_num_batched_tokens: int = 8192   # per-step token budget (chunked prefill)
    max_model_len: int = 8192
    gpu_memory_utilization: float = 0.90
    num_gpu_blocks: Optional[int] = None  # override (tests / CPU)
    enforce_eager: bool = False           # disable hipGraph decode capture
    tp_size: int = 1
    ep_size: int = 1
    dtype: str = "bfloat16"
    quant: Optional[str] = None          # None (bf16) | "fp8" (e4m3fn weights)
    kv_dtype: str = "bfloat16"           # "bfloat16" | "fp8" (e4m3fn cache)
    enable_prefix_caching: bool = False  # shared-prompt KV block reuse
    seed: int = 0
    # decode hipGraph capture batch buckets (padded up to nearest)
    graph_batch_sizes: tuple = (1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128,
                                192, 256, 384, 512, 768, 1024, 1536, 2048)
    # single-request whole-prompt prefill graphs (TTFT fast path):
    # tokens pad up to the bucket, padded slots write no KV
    prefill_graph_sizes: tuple = (128, 256, 512, 1024, 2048)

    def spec(self) -> ModelSpec:
        return get_model_spec(self.model)


@dataclass
class SamplingParams:
    """Per-request sampling parameters.

    The reference request schema has *no* sampling fields
    (llm-gateway-sdk/schemas/core/request.v1.schema.json — verified in
    SURVEY.md Appendix B); these are the engine-side extension exposed via the
    v2 request surface.
    """

    temperature: float = 1.0
    top_p: float = 1.0
    top_k: int = 0            # 0 => disabled
    max_tokens: int = 128
    min_tokens: int = 0
    stop_token_ids: tuple = ()
    ignore_eos: bool = False
    seed: Optional[int] = None
    # "json": constrained decoding — the sampler masks every token that
    # would break RFC-8259 validity and only allows EOS once the
    # top-level value closes; "tool_call" forces the ToolCall skeleton
    # (engine/guided.py)
    response_format: Optional[str] = None
    # a JSON Schema dict: SCHEMA-shaped constrained decoding — the
    # object skeleton (known keys, declared value types) is forced
    # byte-exactly (guided.SchemaMachine); overrides response_format
    response_schema: Optional[dict] = None
    # declared tool names: a forced tool call ("tool_call" format)
    # narrows the emitted name to one of these (guided.ToolCallMachine)
    tool_names: tuple = ()

    @property
    def greedy(self) -> bool:
        return self.temperature == 0.0
