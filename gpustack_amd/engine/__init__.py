from .config import EngineConfig, ModelSpec, PRESETS
from .engine import LLMEngine
from .sequence import SamplingParams, Sequence, StepOutput

__all__ = ["EngineConfig", "ModelSpec", "PRESETS", "LLMEngine", "SamplingParams", "Sequence", "StepOutput"]
