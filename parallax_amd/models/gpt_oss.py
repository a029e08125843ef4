"""GPT-OSS: sliding-window/full alternating layers, per-head attention sinks,
MoE with clamped interleaved-SwiGLU experts and biased projections.

Reference analogue: src/parallax/models/gpt_oss.py (sliding window + sinks);
the sink logit rides the softmax denominator inside our decode/prefill HIP
kernels (ops sinks argument)."""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from .config import ModelConfig
from .llama import LlamaDecoderLayer, LlamaShardModel
from .registry import register_model


class GptOssMoE(nn.Module):
    """Router: linear+bias -> top-k -> softmax over the top-k logits.
    Experts: interleaved gate/up with clamping: glu = clamp(gate) *
    sigmoid(1.702 * gate); out = (clamp(up) + 1) * glu (HF modeling_gpt_oss).

    GPU path: the grouped-GEMM MoE kernel (deinterleaved [E, 2I, H] layout
    built at load time) with the gate/up BIAS folded into the kernel epilogue
    and the down bias added as a capture-safe torch gather — the round-1
    per-expert torch loop used torch.unique, which is illegal inside hipGraph
    capture and serialized experts on the host."""

    alpha = 1.702
    limit = 7.0

    def __init__(self, cfg: ModelConfig):
        super().__init__()
        E, H, I = cfg.num_experts, cfg.hidden_size, cfg.intermediate_size
        self.top_k = cfg.num_experts_per_tok
        self.intermediate_size = I
        self.router_weight = nn.Parameter(torch.empty(E, H), requires_grad=False)
        self.router_bias = nn.Parameter(torch.empty(E), requires_grad=False)
        # kernel layout: [E, 2I, H] out-major, gate rows then up rows
        self.w_gate_up = nn.Parameter(torch.empty(E, 2 * I, H), requires_grad=False)
        self.b_gate_up = nn.Parameter(torch.empty(E, 2 * I), requires_grad=False)
        self.w_down = nn.Parameter(torch.empty(E, H, I), requires_grad=False)
        self.b_down = nn.Parameter(torch.empty(E, H), requires_grad=False)

    # -- HF weight conversion (interleaved input-major -> kernel layout) --------

    def load_hf_expert_weight(self, leaf: str, t: torch.Tensor) -> bool:
        t = t.to(torch.float32)
        if leaf == "gate_up_proj":          # [E, H, 2I] interleaved
            gate = t[:, :, 0::2].transpose(1, 2)   # [E, I, H]
            up = t[:, :, 1::2].transpose(1, 2)
            self.w_gate_up.data.copy_(torch.cat([gate, up], dim=1))
        elif leaf == "gate_up_proj_bias":   # [E, 2I] interleaved
            self.b_gate_up.data.copy_(
                torch.cat([t[:, 0::2], t[:, 1::2]], dim=1)
            )
        elif leaf == "down_proj":           # [E, I, H] input-major
            self.w_down.data.copy_(t.transpose(1, 2))
        elif leaf == "down_proj_bias":
            self.b_down.data.copy_(t)
        else:
            return False
        return True

    def _route(self, x: torch.Tensor):
        logits = F.linear(x.float(), self.router_weight.float(),
                          self.router_bias.float())
        top_vals, top_ids = logits.topk(self.top_k, dim=-1)
        weights = top_vals.softmax(dim=-1)
        return top_ids, weights

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        T, H = x.shape
        top_ids, weights = self._route(x)
        if x.is_cuda:
            out = ops.fused_moe_forward(
                x, self.w_gate_up, self.w_down, top_ids, weights,
                limit=self.limit,
                bias_gate_up=self.b_gate_up.to(torch.bfloat16),
            )
            # down bias: sum_k route_w_k * b_down[e_k]  (capture-safe gather)
            out = out + torch.einsum(
                "tk,tkh->th", weights.float(), self.b_down[top_ids].float()
            )
            return out.to(x.dtype)
        out = torch.zeros(T, H, dtype=torch.float32, device=x.device)
        I = self.intermediate_size
        flat_ids = top_ids.reshape(-1)
        flat_w = weights.reshape(-1)
        token_idx = (
            torch.arange(T, device=x.device).unsqueeze(1).expand_as(top_ids).reshape(-1)
        )
        for e in torch.unique(flat_ids).tolist():
            sel = (flat_ids == e).nonzero(as_tuple=True)[0]
            toks = token_idx[sel]
            xe = x[toks]
            gu = F.linear(xe, self.w_gate_up[e], self.b_gate_up[e])
            gate, up = gu[..., :I], gu[..., I:]
            gate = gate.clamp(max=self.limit)
            up = up.clamp(min=-self.limit, max=self.limit)
            glu = gate * torch.sigmoid(gate * self.alpha)
            act = (up + 1) * glu
            ye = F.linear(act, self.w_down[e], self.b_down[e]).float()
            out.index_add_(0, toks, ye * flat_w[sel].unsqueeze(-1))
        return out.to(x.dtype)


class GptOssDecoderLayer(LlamaDecoderLayer):
    def __init__(self, cfg: ModelConfig, layer_idx: int, local_layer_idx: int):
        super().__init__(cfg, layer_idx, local_layer_idx)
        self.mlp = GptOssMoE(cfg)


@register_model("GptOssForCausalLM")
class GptOssShardModel(LlamaShardModel):
    decoder_layer_cls = GptOssDecoderLayer

    def load_hf_weight(self, name: str, tensor: torch.Tensor) -> bool:
        local = self.map_global_layer(name)
        if local is None:
            return False
        parts = local.split(".")
        if len(parts) >= 3 and parts[0] == "layers" and parts[2] == "mlp":
            layer = self.layers[int(parts[1])]
            moe: GptOssMoE = layer.mlp
            t = tensor.to(torch.bfloat16)
            sub = parts[3]
            if sub == "router":
                if parts[4] == "weight":
                    moe.router_weight.data.copy_(t)
                else:
                    moe.router_bias.data.copy_(t)
                return True
            if sub == "experts":
                return moe.load_hf_expert_weight(parts[4], tensor)
            return False
        if ".self_attn.sinks" in name:
            layer = self.layers[int(parts[1])]
            layer.self_attn.sinks.data.copy_(tensor.float())
            return True
        return super().load_hf_weight(name, tensor)
