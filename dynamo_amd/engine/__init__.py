from .config import EngineConfig, ModelConfig, PRESETS
from .engine import LLMEngine, StepOutput, ForwardPassMetrics
from .scheduler import Request, SamplingParams

__all__ = [
    "EngineConfig", "ModelConfig", "PRESETS", "LLMEngine", "StepOutput",
    "ForwardPassMetrics", "Request", "SamplingParams",
]
