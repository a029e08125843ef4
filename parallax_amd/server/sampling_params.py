"""Sampling parameter model.

Capability parity with the reference's src/parallax/server/sampling/sampling_params.py
(temperature / top-p / top-k / min-p / penalties / stop / max tokens); fresh design.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Union


@dataclass
class SamplingParams:
    temperature: float = 1.0
    top_p: float = 1.0
    top_k: int = -1  # -1 = disabled
    min_p: float = 0.0
    repetition_penalty: float = 1.0
    presence_penalty: float = 0.0
    frequency_penalty: float = 0.0
    max_new_tokens: int = 128
    min_new_tokens: int = 0
    stop: List[str] = field(default_factory=list)
    stop_token_ids: List[int] = field(default_factory=list)
    ignore_eos: bool = False
    seed: Optional[int] = None
    # structured output hook (reference carries json_schema through the wire)
    json_schema: Optional[str] = None
    # OpenAI logit_bias: token id -> additive bias (applied pre-sampling)
    logit_bias: Optional[Dict[int, float]] = None
    # return per-token logprob of the sampled token (reference wire field
    # token_prob / return_probs, proto forward.proto)
    logprobs: bool = False

    def __post_init__(self):
        if self.temperature < 0.0:
            raise ValueError(f"temperature must be >= 0, got {self.temperature}")
        if not 0.0 < self.top_p <= 1.0:
            raise ValueError(f"top_p must be in (0, 1], got {self.top_p}")
        if self.top_k < -1 or self.top_k == 0:
            raise ValueError(f"top_k must be -1 (off) or >= 1, got {self.top_k}")
        if not 0.0 <= self.min_p <= 1.0:
            raise ValueError(f"min_p must be in [0, 1], got {self.min_p}")
        if self.max_new_tokens < 1:
            raise ValueError("max_new_tokens must be >= 1")

    @property
    def is_greedy(self) -> bool:
        return self.temperature == 0.0

    def to_dict(self) -> dict:
        return {
            "temperature": self.temperature,
            "top_p": self.top_p,
            "top_k": self.top_k,
            "min_p": self.min_p,
            "repetition_penalty": self.repetition_penalty,
            "presence_penalty": self.presence_penalty,
            "frequency_penalty": self.frequency_penalty,
            "max_new_tokens": self.max_new_tokens,
            "min_new_tokens": self.min_new_tokens,
            "stop": list(self.stop),
            "stop_token_ids": list(self.stop_token_ids),
            "ignore_eos": self.ignore_eos,
            "seed": self.seed,
            "json_schema": self.json_schema,
            "logprobs": self.logprobs,
            "logit_bias": self.logit_bias,
        }

    @classmethod
    def from_dict(cls, d: dict) -> "SamplingParams":
        known = {k: v for k, v in d.items() if k in cls.__dataclass_fields__}
        return cls(**known)

    @classmethod
    def from_openai(cls, body: dict, default_max_tokens: int = 512) -> "SamplingParams":
        """Build from an OpenAI chat/completions request body."""
        stop = body.get("stop") or []
        if isinstance(stop, str):
            stop = [stop]
        return cls(
            temperature=float(body.get("temperature", 1.0)),
            top_p=float(body.get("top_p", 1.0)),
            top_k=int(body.get("top_k", -1)),
            min_p=float(body.get("min_p", 0.0)),
            repetition_penalty=float(body.get("repetition_penalty", 1.0)),
            presence_penalty=float(body.get("presence_penalty", 0.0)),
            frequency_penalty=float(body.get("frequency_penalty", 0.0)),
            max_new_tokens=int(
                body.get("max_tokens")
                or body.get("max_completion_tokens")
                or default_max_tokens
            ),
            min_new_tokens=int(body.get("min_tokens", 0)),
            stop=stop,
            ignore_eos=bool(body.get("ignore_eos", False)),
            seed=body.get("seed"),
            logprobs=bool(body.get("logprobs", False)),
            json_schema=_extract_json_schema(body),
            logit_bias={int(k): float(v)
                        for k, v in body["logit_bias"].items()}
            if isinstance(body.get("logit_bias"), dict) else None,
        )


def _extract_json_schema(body: dict):
    """OpenAI structured-output surfaces: `response_format={"type":
    "json_schema", "json_schema": {"schema": {...}}}` or a raw `json_schema`
    field (the reference's wire name). Returns the schema as a JSON string."""
    import json as _json

    raw = body.get("json_schema")
    rf = body.get("response_format")
    if raw is None and isinstance(rf, dict) and rf.get("type") == "json_schema":
        inner = rf.get("json_schema") or {}
        raw = inner.get("schema", inner)
    if raw is None and isinstance(rf, dict) and rf.get("type") == "json_object":
        # schema-less JSON mode: any object with any keys and any JSON values
        raw = ANY_JSON_OBJECT_SCHEMA
    if raw is None:
        return None
    return raw if isinstance(raw, str) else _json.dumps(raw)


#: "any JSON value" expressed in the FSM's schema subset (recursive union)
ANY_JSON_OBJECT_SCHEMA = {
    "$defs": {"any": {"anyOf": [
        {"type": "string"}, {"type": "number"}, {"type": "boolean"},
        {"type": "null"},
        {"type": "array", "items": {"$ref": "#/$defs/any"}},
        {"type": "object", "additionalProperties": {"$ref": "#/$defs/any"}},
    ]}},
    "type": "object",
    "additionalProperties": {"$ref": "#/$defs/any"},
}
