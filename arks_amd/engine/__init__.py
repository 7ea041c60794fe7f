from .engine import LLMEngine  # noqa: F401
from .sequence import SamplingParams, Sequence, SeqStatus, StepOutput  # noqa: F401
