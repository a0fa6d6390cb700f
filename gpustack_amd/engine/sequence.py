"""Request / sequence state for the continuous-batching engine."""
from __future__ import annotations

import enum
import time
from dataclasses import dataclass, field


class SeqStatus(enum.Enum):
    WAITING = "waiting"
    RUNNING = "running"
    FINISHED = "finished"


@dataclass
class SamplingParams:
    temperature: float = 0.0        # 0 => greedy
    top_p: float = 1.0
    top_k: int = 0                  # 0 => disabled
    min_p: float = 0.0              # vLLM-style min-p nucleus floor
    presence_penalty: float = 0.0   # OpenAI semantics: subtract if seen
    frequency_penalty: float = 0.0  # OpenAI semantics: subtract per count
    repetition_penalty: float = 1.0  # HF semantics: divide/multiply seen
    logit_bias: dict | None = None  # {token_id: bias}
    # guided decoding: output must be exactly one of these token sequences
    # (constrained per-step logit masking; engine_server encodes the
    # user-facing `guided_choice` strings)
    guided_token_seqs: tuple | None = None
    # grammar-constrained JSON output (engine/guided.py): True = any valid
    # JSON; a dict = JSON-Schema subset compiled to a template machine
    guided_json: "bool | dict | None" = None
    guided_regex: str | None = None  # regex-subset NFA (engine/guided.py)
    guided_grammar: str | None = None  # EBNF CFG, Earley (engine/guided.py)
    eos_token_id: int = 0           # used by guided decoding to terminate
    max_tokens: int = 128
    min_tokens: int = 0        # suppress EOS/stop until this many tokens
    ignore_eos: bool = False
    stop_token_ids: tuple[int, ...] = ()
    seed: int | None = None
    logprobs: bool = False
    priority: int = 0           # higher = admitted sooner, preempted last
    top_logprobs: int = 0       # OpenAI top-k alternative logprobs per token
    # dynamic multi-LoRA: name of a live adapter (engine.add_lora) applied
    # to this request's rows only (models/lora.py)
    lora_name: str | None = None

    @property
    def needs_logit_processing(self) -> bool:
        return bool(self.logit_bias) or bool(self.guided_token_seqs) \
            or self.guided_json is not None \
            or self.guided_regex is not None \
            or self.guided_grammar is not None \
            or self.presence_penalty != 0.0 \
            or self.frequency_penalty != 0.0 \
            or self.repetition_penalty != 1.0 \
            or (self.min_tokens > 0 and not self.ignore_eos)

    @property
    def spec_safe(self) -> bool:
        """Speculative drafts verify greedily row by row; stateful guided
        decoding cannot validate draft rows, so such seqs decode plain."""
        return (self.greedy and self.guided_json is None
                and self.guided_regex is None
                and self.guided_grammar is None)

    @property
    def greedy(self) -> bool:
        return self.temperature <= 0.0


@dataclass
class Sequence:
    request_id: str
    prompt_token_ids: list[int]
    params: SamplingParams = field(default_factory=SamplingParams)
    output_token_ids: list[int] = field(default_factory=list)
    status: SeqStatus = SeqStatus.WAITING
    block_table: list[int] = field(default_factory=list)
    host_block_table: list[int] = field(default_factory=list)  # offload tier
    num_cached_tokens: int = 0      # tokens whose KV is already in the pool
    swap_outs: int = 0
    pending_tokens: int = 0         # async decode: sampled on device, not yet read back
    next_draft: list | None = None  # draft-model speculative window (engine/eagle.py)
    lora_slot: int = 0              # 0 = no adapter (models/lora.py LoraBank)
    block_hashes: list | None = None  # prefix-cache chain (engine/kv_cache.py)
    guided_sm: object | None = None   # guided-JSON machine (engine/guided.py)
    guided_consumed: int = 0          # output tokens already fed to guided_sm
    arrival_time: float = field(default_factory=time.monotonic)
    first_token_time: float | None = None
    finish_time: float | None = None
    finish_reason: str | None = None
    preemptions: int = 0

    @property
    def num_tokens(self) -> int:
        return len(self.prompt_token_ids) + len(self.output_token_ids) + self.pending_tokens

    @property
    def all_token_ids(self) -> list[int]:
        return self.prompt_token_ids + self.output_token_ids

    @property
    def num_prompt_tokens(self) -> int:
        return len(self.prompt_token_ids)

    def record_first_token(self) -> None:
        if self.first_token_time is None:
            self.first_token_time = time.monotonic()

    def finish(self, reason: str) -> None:
        self.status = SeqStatus.FINISHED
        self.finish_reason = reason
        self.finish_time = time.monotonic()

    @property
    def ttft(self) -> float | None:
        if self.first_token_time is None:
            return None
        return self.first_token_time - self.arrival_time


@dataclass
class StepOutput:
    request_id: str
    token_id: int
    finished: bool
    finish_reason: str | None = None
    logprob: float | None = None
    top_logprobs: "list[tuple[int, float]] | None" = None
