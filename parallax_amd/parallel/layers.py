"""Tensor-parallel linear layers (RCCL all-reduce over xGMI at tp_size > 1).

Column-parallel: weight is split on the output dim, no communication on forward
(outputs stay sharded for the following row-parallel layer). Row-parallel:
weight split on the input dim; forward ends with a TP all-reduce. At tp_size=1
these are plain GEMMs through hipBLASLt (torch.nn.functional.linear).
Mirrors the capability of the reference's shard_linear / SGLang TP layers
(SURVEY.md §2.3 TP row) as a fresh design.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .comm import get_comm
from .. import ops


class _Fp8WeightMixin:
    """Per-tensor fp8-E4M3 weight storage for the hipBLASLt W8A8 route
    (BASELINE DeepSeek-V3 fp8 config). quantize_fp8() swaps the bf16 weight
    parameter for an fp8 buffer + device scale; forward then dispatches
    ops.linear_fp8 (activations quantize per-call on device)."""

    fp8 = False

    def quantize_fp8(self) -> None:
        if self.fp8:
            return
        w = self.weight.data
        amax = w.float().abs().amax().clamp_min(1e-8)
        scale = (amax / 448.0).reshape(1).float()
        q = (w.float() / scale).clamp(-448.0, 448.0).to(torch.float8_e4m3fn)
        del self._parameters["weight"]
        self.register_buffer("weight_fp8", q)
        self.register_buffer("weight_scale", scale.to(w.device))
        self.fp8 = True

    def _linear(self, x: torch.Tensor) -> torch.Tensor:
        if self.fp8:
            return ops.linear_fp8(x, self.weight_fp8, self.weight_scale,
                                  getattr(self, "bias", None))
        return ops.linear(x, self.weight, getattr(self, "bias", None))


def _shard(dim_size: int, tp_size: int, tp_rank: int) -> tuple:
    """(start, size) of this rank's shard; dims that do not divide evenly are
    REPLICATED (e.g. odd vocab sizes in the lm_head) — the layer then behaves
    as tp_size == 1 for that tensor."""
    if dim_size % tp_size != 0:
        return 0, dim_size
    per = dim_size // tp_size
    return per * tp_rank, per


class ColumnParallelLinear(nn.Module, _Fp8WeightMixin):
    """Y = X W^T with W sharded along output features."""

    def __init__(self, in_features: int, out_features: int, bias: bool = False,
                 gather_output: bool = False):
        super().__init__()
        comm = get_comm()
        self.full_out_features = out_features
        _, self.out_per_rank = _shard(out_features, comm.tp_size, comm.tp_rank)
        self.replicated = self.out_per_rank == out_features and comm.tp_size > 1
        self.tp_size = 1 if self.replicated else comm.tp_size
        self.weight = nn.Parameter(
            torch.empty(self.out_per_rank, in_features), requires_grad=False
        )
        self.bias = (
            nn.Parameter(torch.empty(self.out_per_rank), requires_grad=False)
            if bias else None
        )
        self.gather_output = gather_output

    def load_full_weight(
        self, w: Optional[torch.Tensor], b: Optional[torch.Tensor] = None
    ) -> None:
        comm = get_comm()
        start, per = _shard(self.full_out_features, comm.tp_size, comm.tp_rank)
        if w is not None:
            self.weight.data.copy_(w[start : start + per])
        if self.bias is not None and b is not None:
            self.bias.data.copy_(b[start : start + per])

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = self._linear(x)
        if self.gather_output and self.tp_size > 1:
            y = get_comm().tp_all_gather(y, dim=-1)
        return y


class MergedColumnParallelLinear(nn.Module, _Fp8WeightMixin):
    """Several column-parallel projections fused into one GEMM (e.g. QKV or
    gate+up). Each sub-projection is sharded independently so per-rank layout is
    [q_shard | k_shard | v_shard]."""

    def __init__(self, in_features: int, out_sizes: List[int], bias: bool = False):
        super().__init__()
        comm = get_comm()
        self.tp_size = comm.tp_size
        self.out_sizes = list(out_sizes)
        self.shard_sizes = [s // comm.tp_size for s in out_sizes]
        total = sum(self.shard_sizes)
        self.weight = nn.Parameter(torch.empty(total, in_features), requires_grad=False)
        self.bias = nn.Parameter(torch.empty(total), requires_grad=False) if bias else None

    def load_full_weight_part(
        self, idx: int, w: Optional[torch.Tensor], b: Optional[torch.Tensor] = None
    ) -> None:
        """Load one sub-projection (e.g. k_proj into the fused QKV) from its
        full (unsharded) HF tensor."""
        comm = get_comm()
        off = sum(self.shard_sizes[:idx])
        start, per = _shard(self.out_sizes[idx], comm.tp_size, comm.tp_rank)
        if w is not None:
            self.weight.data[off : off + per].copy_(w[start : start + per])
        if self.bias is not None and b is not None:
            self.bias.data[off : off + per].copy_(b[start : start + per])

    def load_full_weights(
        self, ws: List[torch.Tensor], bs: Optional[List[Optional[torch.Tensor]]] = None
    ) -> None:
        comm = get_comm()
        off = 0
        for i, w in enumerate(ws):
            start, per = _shard(self.out_sizes[i], comm.tp_size, comm.tp_rank)
            self.weight.data[off : off + per].copy_(w[start : start + per])
            if self.bias is not None and bs is not None and bs[i] is not None:
                self.bias.data[off : off + per].copy_(bs[i][start : start + per])
            off += per

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self._linear(x)

    def split_output(self, y: torch.Tensor) -> List[torch.Tensor]:
        return list(torch.split(y, self.shard_sizes, dim=-1))


class RowParallelLinear(nn.Module, _Fp8WeightMixin):
    """Y = X W^T with W sharded along input features; all-reduce on forward."""

    def __init__(self, in_features: int, out_features: int, bias: bool = False):
        super().__init__()
        comm = get_comm()
        self.full_in_features = in_features
        _, self.in_per_rank = _shard(in_features, comm.tp_size, comm.tp_rank)
        self.replicated = self.in_per_rank == in_features and comm.tp_size > 1
        self.tp_size = 1 if self.replicated else comm.tp_size
        self.weight = nn.Parameter(
            torch.empty(out_features, self.in_per_rank), requires_grad=False
        )
        # bias is replicated; it is added after the all-reduce (appears once)
        self.bias = (
            nn.Parameter(torch.empty(out_features), requires_grad=False) if bias else None
        )

    def load_full_weight(
        self, w: Optional[torch.Tensor], b: Optional[torch.Tensor] = None
    ) -> None:
        if w is not None:
            comm = get_comm()
            start, per = _shard(self.full_in_features, comm.tp_size, comm.tp_rank)
            self.weight.data.copy_(w[:, start : start + per])
        if self.bias is not None and b is not None:
            self.bias.data.copy_(b)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        comm = get_comm()
        if self.fp8:
            y = ops.linear_fp8(x, self.weight_fp8, self.weight_scale)
        else:
            y = ops.linear(x, self.weight)
        if not self.replicated:
            y = comm.tp_all_reduce(y)
        # after the all-reduce every rank holds the full sum; bias is added once
        if self.bias is not None:
            y = y + self.bias
        return y


class VocabEmbedding(nn.Module):
    """Token embedding (not TP-sharded for now: 288 GB HBM3E holds full vocab
    embeddings comfortably even for 128k vocabs)."""

    def __init__(self, vocab_size: int, hidden_size: int):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(vocab_size, hidden_size), requires_grad=False)

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        return F.embedding(input_ids, self.weight)
