from .config import EngineConfig, ModelConfig
from .engine import LLMEngine
from .sequence import SamplingParams, Sequence, SeqStatus
