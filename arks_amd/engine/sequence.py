"""Request/sequence state for the continuous-batching engine."""

from __future__ import annotations

import enum
import time
from dataclasses import dataclass


@dataclass
class SamplingParams:
    max_tokens: int = 128
    temperature: float = 0.0  # 0 => greedy
    top_p: float = 1.0  # applied via logit filtering when < 1.0
    top_k: int = 0  # 0 => disabled
    min_p: float = 0.0  # drop tokens with prob < min_p * max-prob (vLLM)
    presence_penalty: float = 0.0  # flat penalty on seen tokens
    frequency_penalty: float = 0.0  # per-occurrence penalty
    repetition_penalty: float = 1.0  # >1 divides positive seen-logits
    stop_token_ids: tuple[int, ...] = ()
    ignore_eos: bool = False
    logprobs: int | None = None  # None = off; 0 = chosen only; N = top-N too
    seed: int | None = None  # per-request RNG seed (reproducible sampling)
    logit_bias: dict[int, float] | None = None  # OpenAI logit_bias
    min_tokens: int = 0  # suppress eos/stop tokens until this many emitted
    prompt_logprobs: bool = False  # score prompt tokens at prefill (echo)

    @property
    def has_penalties(self) -> bool:
        return (self.presence_penalty != 0.0 or self.frequency_penalty != 0.0
                or self.repetition_penalty != 1.0)


class SeqStatus(enum.Enum):
    WAITING = "waiting"
    RUNNING = "running"
    PREEMPTED = "preempted"
    FINISHED = "finished"
    ABORTED = "aborted"


class Sequence:
    """One request: prompt + generated tokens + its KV block table."""

    _counter = 0

    def __init__(
        self,
        prompt_token_ids: list[int],
        sampling: SamplingParams | None = None,
        request_id: str | None = None,
        arrival_time: float | None = None,
    ):
        Sequence._counter += 1
        self.seq_id = Sequence._counter
        self.request_id = request_id or f"seq-{self.seq_id}"
        self.prompt_token_ids = list(prompt_token_ids)
        self.output_token_ids: list[int] = []
        self.sampling = sampling or SamplingParams()
        self.status = SeqStatus.WAITING
        self.block_table: list[int] = []
        # Disaggregated prefill: keep KV pages allocated after finish so the
        # decode instance can pull them (scheduler.held).
        self.hold_pages = False
        self.num_cached_tokens = 0  # tokens whose KV is already in cache
        # prompt-token logprobs (sampling.prompt_logprobs): value at index
        # q-1 scores prompt token q given tokens < q (first token unscored)
        self.prompt_logprob_values: list[float] = []
        self.arrival_time = arrival_time if arrival_time is not None else time.time()
        self.first_token_time: float | None = None
        self.finish_time: float | None = None
        self.finish_reason: str | None = None

    # --- lengths ---
    @property
    def num_prompt_tokens(self) -> int:
        return len(self.prompt_token_ids)

    @property
    def num_output_tokens(self) -> int:
        return len(self.output_token_ids)

    @property
    def num_tokens(self) -> int:
        return self.num_prompt_tokens + self.num_output_tokens

    @property
    def all_token_ids(self) -> list[int]:
        return self.prompt_token_ids + self.output_token_ids

    @property
    def is_finished(self) -> bool:
        return self.status in (SeqStatus.FINISHED, SeqStatus.ABORTED)

    def last_token(self) -> int:
        return self.output_token_ids[-1] if self.output_token_ids else self.prompt_token_ids[-1]

    def append_token(self, token_id: int) -> None:
        if self.first_token_time is None:
            self.first_token_time = time.time()
        self.output_token_ids.append(token_id)

    def check_finished(self, eos_token_id: int) -> bool:
        sp = self.sampling
        if self.num_output_tokens >= sp.max_tokens:
            self.finish_reason = "length"
        elif not sp.ignore_eos and self.output_token_ids and (
            self.output_token_ids[-1] == eos_token_id
            or self.output_token_ids[-1] in sp.stop_token_ids
        ):
            self.finish_reason = "stop"
        else:
            return False
        self.status = SeqStatus.FINISHED
        self.finish_time = time.time()
        return True


@dataclass
class StepOutput:
    """Per-sequence result of one engine step."""

    request_id: str
    seq_id: int
    new_token_id: int
    finished: bool
    finish_reason: str | None
    num_prompt_tokens: int
    num_output_tokens: int
    logprob: float | None = None  # chosen token's logprob (when requested)
    top_logprobs: dict[int, float] | None = None
    # prompt-token logprobs (echo): set on the sequence's FIRST output
    prompt_logprobs: list[float] | None = None
