from .llama import LlamaForCausalLM
from .mixtral import MixtralForCausalLM
from .registry import build_model
