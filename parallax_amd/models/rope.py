"""RoPE frequency computation: default / linear / llama3 / yarn scaling.

Host-side precomputation (the CDNA guide: on-device trig turns RoPE
VALU-bound); the resulting cos/sin table feeds the fused rope HIP kernels."""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch

from .config import ModelConfig


def compute_inv_freq_and_scale(
    rot_dim: int,
    rope_theta: float,
    rope_scaling: Optional[dict],
    max_position_embeddings: int,
) -> Tuple[torch.Tensor, float]:
    """Returns (inv_freq [rot_dim/2], mscale) applying the configured scaling."""
    base = rope_theta
    inv_freq = 1.0 / (
        base ** (torch.arange(0, rot_dim, 2, dtype=torch.float32) / rot_dim)
    )
    if not rope_scaling:
        return inv_freq, 1.0
    rtype = rope_scaling.get("rope_type") or rope_scaling.get("type") or "default"
    factor = float(rope_scaling.get("factor", 1.0))

    if rtype == "linear":
        return inv_freq / factor, 1.0

    if rtype == "llama3":
        low_factor = float(rope_scaling.get("low_freq_factor", 1.0))
        high_factor = float(rope_scaling.get("high_freq_factor", 4.0))
        old_ctx = float(rope_scaling.get("original_max_position_embeddings", 8192))
        wavelen = 2 * math.pi / inv_freq
        low_wl = old_ctx / low_factor
        high_wl = old_ctx / high_factor
        out = torch.where(wavelen > low_wl, inv_freq / factor, inv_freq)
        smooth = (old_ctx / wavelen - low_factor) / (high_factor - low_factor)
        smoothed = (1 - smooth) / factor * inv_freq + smooth * inv_freq
        is_mid = (wavelen <= low_wl) & (wavelen >= high_wl)
        out = torch.where(is_mid, smoothed, out)
        return out, 1.0

    if rtype in ("yarn", "deepseek_yarn"):
        old_ctx = float(
            rope_scaling.get("original_max_position_embeddings", 4096)
        )
        beta_fast = float(rope_scaling.get("beta_fast", 32.0))
        beta_slow = float(rope_scaling.get("beta_slow", 1.0))

        def find_dim(num_rot: float) -> float:
            return (
                rot_dim * math.log(old_ctx / (num_rot * 2 * math.pi))
            ) / (2 * math.log(base))

        low = max(math.floor(find_dim(beta_fast)), 0)
        high = min(math.ceil(find_dim(beta_slow)), rot_dim // 2 - 1)
        ramp = torch.clamp(
            (torch.arange(rot_dim // 2, dtype=torch.float32) - low)
            / max(high - low, 1e-3),
            0.0, 1.0,
        )
        mask = 1.0 - ramp  # 1 where extrapolation-free (high freq)
        inv_freq_interp = inv_freq / factor
        out = inv_freq_interp * (1 - mask) + inv_freq * mask
        # attention temperature (yarn mscale)
        mscale_all = float(rope_scaling.get("mscale", 1.0))
        mscale_all_dim = float(rope_scaling.get("mscale_all_dim", 0.0))

        def yarn_mscale(scale: float, m: float) -> float:
            if scale <= 1.0 or m == 0.0:
                return 1.0
            return 0.1 * m * math.log(scale) + 1.0

        mscale = (
            yarn_mscale(factor, mscale_all) / yarn_mscale(factor, mscale_all_dim)
            if mscale_all_dim
            else yarn_mscale(factor, mscale_all)
        )
        # transformers' default yarn applies sqrt-temperature to cos/sin
        if "mscale" not in rope_scaling and "mscale_all_dim" not in rope_scaling:
            mscale = float(
                0.1 * math.log(factor) + 1.0
            )
        return out, mscale

    return inv_freq, 1.0


def build_rope_cache_for(cfg: ModelConfig, rot_dim: Optional[int] = None) -> torch.Tensor:
    rot = rot_dim or cfg.head_dim
    inv_freq, mscale = compute_inv_freq_and_scale(
        rot, cfg.rope_theta, cfg.rope_scaling, cfg.max_position_embeddings
    )
    t = torch.arange(cfg.max_position_embeddings, dtype=torch.float32)
    freqs = torch.outer(t, inv_freq)
    return torch.cat([freqs.cos() * mscale, freqs.sin() * mscale], dim=-1)
