from .config import EngineConfig, ModelSpec, SamplingParams, get_model_spec
from .engine import LLMEngine
from .request import FinishReason, Request, StepOutput
