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
    kernel consumes."""

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
        self.router = MoERouter(cfg)
        self.w_gate_up = nn.Parameter(torch.empty(E, 2 * inter, H), requires_grad=False)
        self.w_down = nn.Parameter(torch.empty(E, H, inter), requires_grad=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        T, H = x.shape
        topk_ids, topk_w = self.router(x)           # [T,k]
        if x.is_cuda:
            return ops.fused_moe_forward(
                x, self.w_gate_up, self.w_down, topk_ids, topk_w,
                limit=self.act_limit,
            ).to(x.dtype)
        out = torch.zeros(T, H, dtype=torch.float32, device=x.device)
        flat_ids = topk_ids.reshape(-1)              # [T*k]
        flat_w = topk_w.reshape(-1)
        token_idx = (
            torch.arange(T, device=x.device).unsqueeze(1).expand_as(topk_ids).reshape(-1)
        )
        used_experts = torch.unique(flat_ids)
        for e in used_experts.tolist():
            sel = (flat_ids == e).nonzero(as_tuple=True)[0]
            toks = token_idx[sel]
            xe = x[toks]                              # [n_e, H]
            h = F.linear(xe, self.w_gate_up[e])       # [n_e, 2I]
            if self.act_limit > 0:
                I = self.intermediate_size
                gate = h[:, :I].clamp(max=self.act_limit)
                up = h[:, I:].clamp(min=-self.act_limit, max=self.act_limit)
                h = (up + 1.0) * (gate * torch.sigmoid(gate * 1.702))
            else:
                h = ops.silu_and_mul(h)
            ye = F.linear(h, self.w_down[e]).float()  # [n_e, H]
            out.index_add_(0, toks, ye * flat_w[sel].unsqueeze(-1))
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
