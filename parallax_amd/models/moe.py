"""Mixture-of-Experts layer.

v0 compute path: dense per-expert GEMMs over gathered token groups (rocBLAS via
torch). The MFMA grouped-GEMM HIP kernel replaces the inner loop later; the
module boundary (routing -> grouped expert FFN -> scatter-add) is already the
kernel's contract. Expert-parallel sharding (experts split across EP ranks with
all-to-all token exchange) hangs off the same routing output.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from ..parallel.comm import get_comm
from .config import ModelConfig


class MoERouter(nn.Module):
    """Top-k router. Supports softmax scoring (Qwen/GLM style) and sigmoid
    scoring with bias correction + group-limited top-k (DeepSeek-V3 style)."""

    def __init__(self, cfg: ModelConfig):
        super().__init__()
        self.num_experts = cfg.num_experts
        self.top_k = cfg.num_experts_per_tok
        self.norm_topk_prob = cfg.norm_topk_prob
        self.scoring_func = cfg.scoring_func
        self.routed_scaling_factor = cfg.routed_scaling_factor
        self.n_group = cfg.n_group
        self.topk_group = cfg.topk_group
        self.weight = nn.Parameter(
            torch.empty(cfg.num_experts, cfg.hidden_size), requires_grad=False
        )
        if cfg.scoring_func == "sigmoid":
            # DeepSeek-V3 e_score_correction_bias
            self.e_score_correction_bias = nn.Parameter(
                torch.zeros(cfg.num_experts), requires_grad=False
            )

    def forward(self, x: torch.Tensor):
        """x: [T, H] -> (topk_ids [T, k] long, topk_weights [T, k] float)."""
        logits = F.linear(x.float(), self.weight.float())
        if self.scoring_func == "sigmoid":
            scores = logits.sigmoid()
            select = scores + self.e_score_correction_bias
            if self.n_group > 1:
                T = x.shape[0]
                gs = select.view(T, self.n_group, -1)
                # group score = sum of top-2 expert scores per group (DeepSeek-V3)
                group_scores = gs.topk(min(2, gs.shape[-1]), dim=-1).values.sum(-1)
                keep_groups = group_scores.topk(self.topk_group, dim=-1).indices
                mask = torch.zeros_like(group_scores, dtype=torch.bool)
                mask.scatter_(1, keep_groups, True)
                select = select.masked_fill(
                    ~mask.unsqueeze(-1).expand_as(gs).reshape(T, -1), float("-inf")
                )
            topk_ids = select.topk(self.top_k, dim=-1).indices
            topk_w = scores.gather(1, topk_ids)
            if self.norm_topk_prob:
                topk_w = topk_w / topk_w.sum(dim=-1, keepdim=True).clamp_min(1e-20)
            topk_w = topk_w * self.routed_scaling_factor
        else:
            probs = logits.softmax(dim=-1)
            topk_w, topk_ids = probs.topk(self.top_k, dim=-1)
            if self.norm_topk_prob:
                topk_w = topk_w / topk_w.sum(dim=-1, keepdim=True).clamp_min(1e-20)
        return topk_ids, topk_w


class FusedMoE(nn.Module):
    """Routed experts with fused gate_up/down weights stored as stacked 3-D
    tensors [E, 2I, H] / [E, H, I] — the exact layout the MFMA grouped-GEMM
    kernel consumes.

    Expert parallelism: when the engine runs TP > 1 and the expert count
    divides, experts are SHARDED across the TP group (each rank holds
    E/tp_size experts, computes its local experts' contributions for the
    whole batch, and the partial outputs all-reduce) — a real EP
    implementation where the reference only plumbs moe_ep_size=1
    (sglang/model_runner.py:65-66). Non-divisible counts fall back to
    replication."""

    def __init__(self, cfg: ModelConfig, intermediate_size: Optional[int] = None,
                 act_limit: float = 0.0):
        super().__init__()
        self.cfg = cfg
        # act_limit > 0: gpt-oss/minimax-m3 clamped swiglu
        #   glu = min(gate, L) * sigmoid(1.702 * min(gate, L));
        #   out = (clamp(up, -L, L) + 1) * glu
        self.act_limit = act_limit
        inter = intermediate_size or cfg.moe_intermediate_size or cfg.intermediate_size
        self.intermediate_size = inter
        E, H = cfg.num_experts, cfg.hidden_size
        comm = get_comm()
        self._comm = comm
        self.ep_size = comm.tp_size if (comm.tp_size > 1 and E % comm.tp_size == 0) else 1
        self.ep_rank = comm.tp_rank if self.ep_size > 1 else 0
        self.num_local_experts = E // self.ep_size
        self.expert_offset = self.ep_rank * self.num_local_experts
        self.router = MoERouter(cfg)
        self.fp8 = False
        self.w_gate_up = nn.Parameter(
            torch.empty(self.num_local_experts, 2 * inter, H), requires_grad=False
        )
        self.w_down = nn.Parameter(
            torch.empty(self.num_local_experts, H, inter), requires_grad=False
        )
        # expert-routing observability (reference parity:
        # vllm_executor enable_return_routed_experts): persistent GLOBAL
        # per-expert token counters, graph-safe (fixed-shape scatter_add
        # on a buffer that exists before capture). None = disabled, free.
        self.routing_counts: Optional[torch.Tensor] = None

    def enable_routing_stats(self) -> None:
        self.routing_counts = torch.zeros(
            self.cfg.num_experts + 1, dtype=torch.int64,
            device=self.w_gate_up.device,
        )

    # -- fp8 (W8A8) path ---------------------------------------------------------

    def quantize_fp8(self) -> None:
        """Convert expert weights to fp8-E4M3 with per-output-channel dequant
        scales — the BASELINE DeepSeek-V3 "fp8 MFMA" configuration. The GPU
        path then runs the fp8 grouped-GEMM kernels (ops.fused_moe_forward_fp8)
        with per-token activation quantization; the CPU fallback dequantizes.
        Quantization is chunked over experts to bound the fp32 temporary."""
        if self.fp8:
            return
        E = self.num_local_experts
        gu, dn = self.w_gate_up.data, self.w_down.data
        q_gu = torch.empty(gu.shape, dtype=torch.float8_e4m3fn, device=gu.device)
        s_gu = torch.empty(gu.shape[:2], dtype=torch.float32, device=gu.device)
        q_dn = torch.empty(dn.shape, dtype=torch.float8_e4m3fn, device=dn.device)
        s_dn = torch.empty(dn.shape[:2], dtype=torch.float32, device=dn.device)
        step = max(1, min(8, E))
        for e0 in range(0, E, step):
            sl = slice(e0, min(e0 + step, E))
            q_gu[sl], s_gu[sl] = ops.quantize_fp8_weight(gu[sl])
            q_dn[sl], s_dn[sl] = ops.quantize_fp8_weight(dn[sl])
        del self._parameters["w_gate_up"], self._parameters["w_down"]
        self.register_buffer("w_gate_up_fp8", q_gu)
        self.register_buffer("w_gu_scale", s_gu)
        self.register_buffer("w_down_fp8", q_dn)
        self.register_buffer("w_down_scale", s_dn)
        self.fp8 = True

    def _expert_gu(self, e: int, dtype: torch.dtype) -> torch.Tensor:
        if self.fp8:
            return (self.w_gate_up_fp8[e].float()
                    * self.w_gu_scale[e].unsqueeze(-1)).to(dtype)
        return self.w_gate_up[e]

    def _expert_down(self, e: int, dtype: torch.dtype) -> torch.Tensor:
        if self.fp8:
            return (self.w_down_fp8[e].float()
                    * self.w_down_scale[e].unsqueeze(-1)).to(dtype)
        return self.w_down[e]

    # -- EP-aware weight loading (used by every family's loader) ----------------

    def local_expert(self, e: int) -> Optional[int]:
        l = e - self.expert_offset
        return l if 0 <= l < self.num_local_experts else None

    def load_expert_gate(self, e: int, t: torch.Tensor) -> None:
        l = self.local_expert(e)
        if l is not None:
            self.w_gate_up.data[l, : self.intermediate_size].copy_(t)

    def load_expert_up(self, e: int, t: torch.Tensor) -> None:
        l = self.local_expert(e)
        if l is not None:
            self.w_gate_up.data[l, self.intermediate_size :].copy_(t)

    def load_expert_down(self, e: int, t: torch.Tensor) -> None:
        l = self.local_expert(e)
        if l is not None:
            self.w_down.data[l].copy_(t)

    def load_fused_gate_up(self, t: torch.Tensor) -> None:
        o = self.expert_offset
        self.w_gate_up.data.copy_(t[o : o + self.num_local_experts])

    def load_fused_down(self, t: torch.Tensor) -> None:
        o = self.expert_offset
        self.w_down.data.copy_(t[o : o + self.num_local_experts])

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        T, H = x.shape
        topk_ids, topk_w = self.router(x)           # [T,k] global expert ids
        if self.routing_counts is not None:
            self.routing_counts.scatter_add_(
                0, topk_ids.reshape(-1),
                torch.ones_like(topk_ids.reshape(-1)),
            )
        if self.ep_size > 1:
            # remap to local ids; foreign experts get the sentinel id
            # num_local_experts, which the grouped-GEMM segment table and the
            # CPU loop both skip — no wasted compute, partials all-reduce
            local = topk_ids - self.expert_offset
            valid = (local >= 0) & (local < self.num_local_experts)
            topk_ids = torch.where(
                valid, local, torch.full_like(local, self.num_local_experts)
            )
            topk_w = topk_w * valid
        if x.is_cuda:
            if self.fp8:
                out = ops.fused_moe_forward_fp8(
                    x, self.w_gate_up_fp8, self.w_gu_scale,
                    self.w_down_fp8, self.w_down_scale, topk_ids, topk_w,
                    limit=self.act_limit,
                )
            else:
                out = ops.fused_moe_forward(
                    x, self.w_gate_up, self.w_down, topk_ids, topk_w,
                    limit=self.act_limit,
                )
            if self.ep_size > 1:
                out = self._comm.tp_all_reduce(out)
            return out.to(x.dtype)
        out = torch.zeros(T, H, dtype=torch.float32, device=x.device)
        flat_ids = topk_ids.reshape(-1)              # [T*k]
        flat_w = topk_w.reshape(-1)
        token_idx = (
            torch.arange(T, device=x.device).unsqueeze(1).expand_as(topk_ids).reshape(-1)
        )
        used_experts = torch.unique(flat_ids)
        for e in used_experts.tolist():
            if e >= self.num_local_experts:
                continue  # EP sentinel
            sel = (flat_ids == e).nonzero(as_tuple=True)[0]
            toks = token_idx[sel]
            xe = x[toks]                              # [n_e, H]
            h = F.linear(xe, self._expert_gu(e, xe.dtype))  # [n_e, 2I]
            if self.act_limit > 0:
                I = self.intermediate_size
                gate = h[:, :I].clamp(max=self.act_limit)
                up = h[:, I:].clamp(min=-self.act_limit, max=self.act_limit)
                h = (up + 1.0) * (gate * torch.sigmoid(gate * 1.702))
            else:
                h = ops.silu_and_mul(h)
            ye = F.linear(h, self._expert_down(e, h.dtype)).float()  # [n_e, H]
            out.index_add_(0, toks, ye * flat_w[sel].unsqueeze(-1))
        if self.ep_size > 1:
            out = self._comm.tp_all_reduce(out)
        return out.to(x.dtype)


class MoEBlock(nn.Module):
    """Routed experts + optional shared experts (DeepSeek/GLM/Qwen2-MoE)."""

    def __init__(self, cfg: ModelConfig):
        super().__init__()
        from .llama import LlamaMLP  # avoid import cycle

        self.experts = FusedMoE(cfg)
        self.shared = None
        if cfg.num_shared_experts > 0:
            inter = (cfg.moe_intermediate_size or cfg.intermediate_size)
            self.shared = LlamaMLP(cfg, intermediate_size=inter * cfg.num_shared_experts)
        elif cfg.shared_expert_intermediate_size > 0:
            self.shared = LlamaMLP(
                cfg, intermediate_size=cfg.shared_expert_intermediate_size
            )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = self.experts(x)
        if self.shared is not None:
            y = y + self.shared(x)
        return y
