"""Request state machine for the per-node serving engine.

Capability parity with the reference's src/parallax/server/request.py:83,157,268
(`Request` / `InitialRequest` / `IntermediateRequest`): the head peer owns full
request state (prompt, generated tokens, sampling params); every other peer sees
only the wire packet (rid, position, hidden states / next token, routing table).
Fresh MI355X-first design: hidden states are torch tensors that stay on-device for
the in-host (xGMI) path and are only serialized for the multi-host transport.
"""

from __future__ import annotations

import enum
import time
import uuid
from dataclasses import dataclass, field
from typing import List, Optional

import torch

from .sampling_params import SamplingParams


class RequestStatus(enum.Enum):
    WAITING = enum.auto()        # admitted to the engine, not yet scheduled
    PREFILLING = enum.auto()     # running prefill (possibly chunked)
    DECODING = enum.auto()       # running decode
    FINISHED_EOS = enum.auto()
    FINISHED_STOP = enum.auto()
    FINISHED_LENGTH = enum.auto()
    FINISHED_ABORT = enum.auto()

    @property
    def is_finished(self) -> bool:
        return self in (
            RequestStatus.FINISHED_EOS,
            RequestStatus.FINISHED_STOP,
            RequestStatus.FINISHED_LENGTH,
            RequestStatus.FINISHED_ABORT,
        )

    @property
    def finish_reason(self) -> Optional[str]:
        return {
            RequestStatus.FINISHED_EOS: "stop",
            RequestStatus.FINISHED_STOP: "stop",
            RequestStatus.FINISHED_LENGTH: "length",
            RequestStatus.FINISHED_ABORT: "abort",
        }.get(self)


def new_request_id() -> str:
    return uuid.uuid4().hex


@dataclass
class Request:
    """Base request: identity + pipeline routing."""

    rid: str
    status: RequestStatus = RequestStatus.WAITING
    # Ordered node/stage ids forming the pipeline for this request. For the
    # single-host engine these are PP ranks; for the decentralized layer they
    # are peer ids assigned by the scheduler.
    routing_table: List[str] = field(default_factory=list)
    arrival_time: float = field(default_factory=time.monotonic)
    lora_path: Optional[str] = None

    @property
    def is_finished(self) -> bool:
        return self.status.is_finished

    def next_hop(self, self_id: str) -> Optional[str]:
        """Next stage in the pipeline; wraps around so the sampled token returns
        to the head peer (reference p2p/server.py:640 behavior)."""
        if not self.routing_table:
            return None
        idx = self.routing_table.index(self_id)
        return self.routing_table[(idx + 1) % len(self.routing_table)]


@dataclass
class InitialRequest(Request):
    """Full request state. Lives only on the head (first-stage) peer."""

    prompt_token_ids: List[int] = field(default_factory=list)
    output_token_ids: List[int] = field(default_factory=list)
    sampling_params: SamplingParams = field(default_factory=SamplingParams)
    eos_token_ids: List[int] = field(default_factory=list)
    # chunked prefill progress: number of prompt tokens whose KV is computed
    num_prefilled_tokens: int = 0
    # timestamps for TTFT / TPOT metrics
    first_token_time: Optional[float] = None
    finish_time: Optional[float] = None
    # set when the client disconnected / abort was requested
    abort_requested: bool = False
    # lazily-built constrained-decoding matcher (sampling_params.json_schema);
    # lives on the rank that samples, never serialized
    grammar: Optional[object] = field(default=None, repr=False, compare=False)

    @property
    def prompt_len(self) -> int:
        return len(self.prompt_token_ids)

    @property
    def num_output_tokens(self) -> int:
        return len(self.output_token_ids)

    @property
    def total_len(self) -> int:
        return self.prompt_len + self.num_output_tokens

    @property
    def prefill_done(self) -> bool:
        return self.num_prefilled_tokens >= self.prompt_len

    @property
    def all_token_ids(self) -> List[int]:
        return self.prompt_token_ids + self.output_token_ids

    def commit_new_token(self, token_id: int) -> None:
        if self.first_token_time is None:
            self.first_token_time = time.monotonic()
        self.output_token_ids.append(token_id)

    def check_finished(self, stop_token_ids_extra: Optional[List[int]] = None) -> bool:
        """EOS / stop-token / length finish checks (reference server/scheduler.py:192).
        String-stop matching happens at the detokenizing frontend."""
        if self.status.is_finished:
            return True
        sp = self.sampling_params
        if self.abort_requested:
            self.status = RequestStatus.FINISHED_ABORT
        elif self.num_output_tokens >= sp.max_new_tokens:
            self.status = RequestStatus.FINISHED_LENGTH
        elif self.num_output_tokens >= sp.min_new_tokens and self.output_token_ids:
            last = self.output_token_ids[-1]
            stop_ids = set(sp.stop_token_ids)
            if stop_token_ids_extra:
                stop_ids |= set(stop_token_ids_extra)
            if not sp.ignore_eos and last in self.eos_token_ids:
                self.status = RequestStatus.FINISHED_EOS
            elif last in stop_ids:
                self.status = RequestStatus.FINISHED_STOP
        if self.status.is_finished and self.finish_time is None:
            self.finish_time = time.monotonic()
        return self.status.is_finished


@dataclass
class IntermediateRequest(Request):
    """The wire packet between pipeline stages.

    Prefill hop: hidden_states is (prompt_chunk_len, hidden); decode hop: (1, hidden).
    The final (last→head) hop carries next_token_id instead of hidden states.
    """

    current_position: int = 0          # context length BEFORE this step's tokens
    num_new_tokens: int = 1            # tokens being processed this step
    hidden_states: Optional[torch.Tensor] = None
    next_token_id: Optional[int] = None
    sampling_params: Optional[SamplingParams] = None
    # token ids for this step (needed by the first stage for embedding; carried
    # so any stage can re-embed after elastic reallocation)
    input_ids: Optional[List[int]] = None
    is_prefill: bool = False
    return_logprob: bool = False
    token_logprob: Optional[float] = None

    @classmethod
    def from_initial(
        cls,
        req: InitialRequest,
        hidden_states: Optional[torch.Tensor],
        *,
        is_prefill: bool,
        position: int,
        num_new_tokens: int,
    ) -> "IntermediateRequest":
        return cls(
            rid=req.rid,
            status=req.status,
            routing_table=list(req.routing_table),
            current_position=position,
            num_new_tokens=num_new_tokens,
            hidden_states=hidden_states,
            sampling_params=req.sampling_params,
            is_prefill=is_prefill,
            lora_path=req.lora_path,
        )
