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

from .config import ModelConfig
from .llama import LlamaDecoderLayer, LlamaShardModel
from .registry import register_model


class GptOssMoE(nn.Module):
    """Router: linear+bias -> top-k -> softmax over the top-k logits.
    Experts: interleaved gate/up with clamping: glu = clamp(gate) *
    sigmoid(1.702 * gate); out = (clamp(up) + 1) * glu (HF modeling_gpt_oss)."""

    alpha = 1.702
    limit = 7.0

    def __init__(self, cfg: ModelConfig):
        super().__init__()
        E, H, I = cfg.num_experts, cfg.hidden_size, cfg.intermediate_size
        self.top_k = cfg.num_experts_per_tok
        self.intermediate_size = I
        self.router_weight = nn.Parameter(torch.empty(E, H), requires_grad=False)
        self.router_bias = nn.Parameter(torch.empty(E), requires_grad=False)
        # HF fused layout: [E, H, 2I] / [E, I, H] (input-major)
        self.gate_up_proj = nn.Parameter(torch.empty(E, H, 2 * I), requires_grad=False)
        self.gate_up_proj_bias = nn.Parameter(torch.empty(E, 2 * I), requires_grad=False)
        self.down_proj = nn.Parameter(torch.empty(E, I, H), requires_grad=False)
        self.down_proj_bias = nn.Parameter(torch.empty(E, H), requires_grad=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        T, H = x.shape
        logits = F.linear(x.float(), self.router_weight.float(),
                          self.router_bias.float())
        top_vals, top_ids = logits.topk(self.top_k, dim=-1)
        weights = top_vals.softmax(dim=-1)
        out = torch.zeros(T, H, dtype=torch.float32, device=x.device)
        flat_ids = top_ids.reshape(-1)
        flat_w = weights.reshape(-1)
        token_idx = (
            torch.arange(T, device=x.device).unsqueeze(1).expand_as(top_ids).reshape(-1)
        )
        for e in torch.unique(flat_ids).tolist():
            sel = (flat_ids == e).nonzero(as_tuple=True)[0]
            toks = token_idx[sel]
            xe = x[toks]
            gu = xe @ self.gate_up_proj[e] + self.gate_up_proj_bias[e]
            gate, up = gu[..., ::2], gu[..., 1::2]
            gate = gate.clamp(max=self.limit)
            up = up.clamp(min=-self.limit, max=self.limit)
            glu = gate * torch.sigmoid(gate * self.alpha)
            act = (up + 1) * glu
            ye = (act @ self.down_proj[e] + self.down_proj_bias[e]).float()
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
                leaf = parts[4]
                getattr(moe, leaf).data.copy_(t)
                return True
            return False
        if ".self_attn.sinks" in name:
            layer = self.layers[int(parts[1])]
            layer.self_attn.sinks.data.copy_(tensor.float())
            return True
        return super().load_hf_weight(name, tensor)
