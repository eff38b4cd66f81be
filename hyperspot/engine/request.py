"""Request / sequence state for the continuous-batching engine."""

from __future__ import annotations

import enum
import time
from dataclasses import dataclass, field
from typing import List, Optional

from .config import SamplingParams


class RequestState(enum.Enum):
    WAITING = "waiting"
    RUNNING = "running"
    PREEMPTED = "preempted"
    FINISHED = "finished"


class FinishReason(str, enum.Enum):
    # mirrors the reference stream_chunk finish_reason enum
    # (llm-gateway-sdk/schemas/core/stream_chunk.v1.schema.json)
    STOP = "stop"
    LENGTH = "length"
    ABORT = "abort"


@dataclass
class Request:
    request_id: str
    prompt_token_ids: List[int]
    sampling: SamplingParams
    tenant_id: str = "default"
    arrival_time: float = field(default_factory=time.monotonic)
    output_token_ids: List[int] = field(default_factory=list)
    num_generated: int = 0      # survives preemption/recompute
    # chunked prefill progress (tokens whose KV is computed) + the chunk
    # scheduled for the CURRENT step
    num_computed_tokens: int = 0
    chunk_start: int = 0
    chunk_len: int = 0
    blocks_preallocated: bool = False   # prefix-caching whole-prompt alloc
    state: RequestState = RequestState.WAITING
    finish_reason: Optional[FinishReason] = None
    first_token_time: Optional[float] = None
    finish_time: Optional[float] = None

    @property
    def num_prompt_tokens(self) -> int:
        return len(self.prompt_token_ids)

    @property
    def num_tokens(self) -> int:
        return len(self.prompt_token_ids) + len(self.output_token_ids)

    @property
    def last_token_id(self) -> int:
        return self.output_token_ids[-1] if self.output_token_ids \
            else self.prompt_token_ids[-1]

    def append_output(self, token_id: int, eos_token_id: Optional[int] = None):
        if self.first_token_time is None:
            self.first_token_time = time.monotonic()
        self.output_token_ids.append(token_id)
        self.num_generated += 1
        n = self.num_generated
        if n >= self.sampling.max_tokens:
            self._finish(FinishReason.LENGTH)
        elif n >= self.sampling.min_tokens and not self.sampling.ignore_eos:
            if (eos_token_id is not None and token_id == eos_token_id) or \
                    token_id in self.sampling.stop_token_ids:
                self._finish(FinishReason.STOP)

    def _finish(self, reason: FinishReason):
        self.state = RequestState.FINISHED
        self.finish_reason = reason
        self.finish_time = time.monotonic()

    @property
    def finished(self) -> bool:
        return self.state == RequestState.FINISHED


@dataclass
class StepOutput:
    request_id: str
    token_id: int
    finished: bool
    finish_reason: Optional[FinishReason]
