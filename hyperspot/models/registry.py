"""Model registry: spec name -> model class (the in-engine half; the
tenant-facing model-registry module with canonical `{provider}::{model}` ids
lives in the C++ host plane, reference modules/model-registry/docs/PRD.md)."""

from __future__ import annotations

import torch

from hyperspot.engine.config import ModelSpec, get_model_spec


def build_model(spec_or_name, dtype: torch.dtype = torch.bfloat16):
    from .llama import LlamaForCausalLM
    from .mixtral import MixtralForCausalLM
    spec = spec_or_name if isinstance(spec_or_name, ModelSpec) \
        else get_model_spec(spec_or_name)
    cls = MixtralForCausalLM if spec.is_moe else LlamaForCausalLM
    return cls(spec, dtype=dtype)
