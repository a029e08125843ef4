"""Wire codec for inter-stage packets (multi-host path).

Reference analogue: p2p/proto/forward.proto + message_util.py (protobuf with
safetensors-serialized tensors). Fresh design: msgpack framing with raw
little-endian tensor buffers described by (dtype, shape) — no tensor-library
round-trip on the wire; bf16 hidden states travel as raw bytes.

Packet kinds: "forward" (batch of intermediate requests), "token" (sampled
token back to head), "release" (free cache state), "abort".
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Tuple

import msgpack
import torch

from ..server.request import IntermediateRequest
from ..server.sampling_params import SamplingParams

_DTYPES = {
    "bf16": torch.bfloat16,
    "f16": torch.float16,
    "f32": torch.float32,
    "i64": torch.int64,
    "i32": torch.int32,
}
_DTYPE_NAMES = {v: k for k, v in _DTYPES.items()}


def _pack_tensor(t: Optional[torch.Tensor]) -> Optional[dict]:
    if t is None:
        return None
    t = t.detach().contiguous().cpu()
    return {
        "dtype": _DTYPE_NAMES[t.dtype],
        "shape": list(t.shape),
        "data": t.view(torch.uint8).numpy().tobytes()
        if t.dtype == torch.bfloat16
        else t.numpy().tobytes(),
    }


def _unpack_tensor(d: Optional[dict]) -> Optional[torch.Tensor]:
    if d is None:
        return None
    dtype = _DTYPES[d["dtype"]]
    raw = torch.frombuffer(bytearray(d["data"]), dtype=torch.uint8)
    return raw.view(dtype).reshape(d["shape"])


def encode_forward(reqs: List[IntermediateRequest]) -> bytes:
    payload = {
        "kind": "forward",
        "reqs": [
            {
                "rid": r.rid,
                "routing_table": r.routing_table,
                "current_position": r.current_position,
                "num_new_tokens": r.num_new_tokens,
                "is_prefill": r.is_prefill,
                "hidden": _pack_tensor(r.hidden_states),
                "next_token_id": r.next_token_id,
                "input_ids": r.input_ids,
                "sampling_params": r.sampling_params.to_dict()
                if r.sampling_params
                else None,
                "lora_path": r.lora_path,
            }
            for r in reqs
        ],
    }
    return msgpack.packb(payload, use_bin_type=True)


def encode_control(kind: str, rids: List[str]) -> bytes:
    assert kind in ("release", "abort")
    return msgpack.packb({"kind": kind, "rids": rids}, use_bin_type=True)


def encode_tokens(tokens) -> bytes:
    """Sampled (rid, token_id) or (rid, token_id, logprob) tuples from the
    last stage back to the head (logprob = reference proto token_prob)."""
    return msgpack.packb({"kind": "token", "tokens": tokens}, use_bin_type=True)


def decode(data: bytes) -> Dict[str, Any]:
    msg = msgpack.unpackb(data, raw=False)
    if msg["kind"] == "forward":
        reqs = []
        for r in msg["reqs"]:
            reqs.append(
                IntermediateRequest(
                    rid=r["rid"],
                    routing_table=list(r["routing_table"]),
                    current_position=r["current_position"],
                    num_new_tokens=r["num_new_tokens"],
                    is_prefill=r["is_prefill"],
                    hidden_states=_unpack_tensor(r["hidden"]),
                    next_token_id=r.get("next_token_id"),
                    input_ids=list(r["input_ids"]) if r.get("input_ids") else None,
                    sampling_params=SamplingParams.from_dict(r["sampling_params"])
                    if r.get("sampling_params")
                    else None,
                    lora_path=r.get("lora_path"),
                )
            )
        msg["reqs"] = reqs
    return msg
