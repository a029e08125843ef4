"""Op dispatch: hand-written HIP/CDNA4 kernels on GPU, eager torch on CPU.

Policy (per the MI355X-first design): on a CUDA (= ROCm/HIP) device the compiled
extension `parallax_amd._C` is REQUIRED for the core hot ops — if it is missing
the op raises instead of silently falling back to eager torch, so a GPU run can
never "pass" on a Python fallback. On CPU the fp32 reference implementations run
(plumbing tests, no GPU in CI).
"""

from __future__ import annotations

import os
from typing import List, Optional, Tuple

import torch

from . import reference as ref
from ..utils.logging_config import get_logger

logger = get_logger("ops")

_EXT = None
_EXT_ERR: Optional[str] = None
_warned_prefill = False


def _try_load_extension():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from . import _C  # built in-tree by setup.py build_ext --inplace

        _EXT = _C
        logger.info("loaded HIP extension parallax_amd.ops._C")
    except ImportError as e:  # pragma: no cover - exercised only without build
        _EXT_ERR = str(e)
    return _EXT


def has_extension() -> bool:
    return _try_load_extension() is not None


def _require_ext(op_name: str):
    ext = _try_load_extension()
    if ext is None:
        raise RuntimeError(
            f"parallax_amd HIP extension is required for {op_name} on GPU but is not "
            f"built (import error: {_EXT_ERR}). Build it with "
            f"`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950)."
        )
    return ext


# -- normalization ------------------------------------------------------------


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    if x.is_cuda:
        ext = _require_ext("rmsnorm")
        out = torch.empty_like(x)
        ext.rmsnorm(out, x.contiguous(), weight, eps)
        return out
    return ref.rmsnorm(x, weight, eps)


def fused_add_rmsnorm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Returns (normed, new_residual). On GPU this mutates x (normed) and
    residual (sum) in place to avoid two extra HBM round trips."""
    if x.is_cuda:
        ext = _require_ext("fused_add_rmsnorm")
        ext.fused_add_rmsnorm(x, residual, weight, eps)
        return x, residual
    return ref.fused_add_rmsnorm(x, residual, weight, eps)


# -- rotary -------------------------------------------------------------------

build_rope_cache = ref.build_rope_cache


def rope_inplace(
    q: torch.Tensor,
    k: Optional[torch.Tensor],
    positions: torch.Tensor,
    cos_sin: torch.Tensor,
    is_neox: bool = True,
) -> None:
    if q.is_cuda:
        ext = _require_ext("rope")
        ext.rope_inplace(
            q,
            k if k is not None else q[:0],
            positions.to(torch.int32),
            cos_sin,
            is_neox,
        )
        return
    ref.rope_inplace(q, k, positions, cos_sin, is_neox)


# -- cache scatter ---------------------------------------------------------------


def rope_and_cache(
    q: torch.Tensor,             # [T, Hq, D] (rows may be strided, e.g. QKV views)
    k: torch.Tensor,             # [T, Hk, D]
    v: torch.Tensor,             # [T, Hk, D]
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    positions: torch.Tensor,
    cos_sin: torch.Tensor,
    slot_mapping: torch.Tensor,
    is_neox: bool = True,
    k_scale: float = 1.0,
    v_scale: float = 1.0,
) -> None:
    """Fused: rope(q) in place; rope(k) + v scattered straight into the paged
    cache (quantizing when the cache is fp8_e4m3fn). One kernel per layer
    instead of three."""
    if q.is_cuda:
        ext = _require_ext("rope_and_cache")
        ext.rope_and_cache(
            q, k, v, k_cache, v_cache, positions.to(torch.int32), cos_sin,
            slot_mapping.to(torch.int64), is_neox, k_scale, v_scale,
        )
        return
    ref.rope_inplace(q, k, positions, cos_sin, is_neox)
    ref.reshape_and_cache(k, v, k_cache, v_cache, slot_mapping)


def reshape_and_cache(
    k: torch.Tensor,
    v: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,
) -> None:
    if k.is_cuda:
        ext = _require_ext("reshape_and_cache")
        ext.reshape_and_cache(
            k.contiguous(), v.contiguous(), k_cache, v_cache, slot_mapping.to(torch.int64)
        )
        return
    ref.reshape_and_cache(k, v, k_cache, v_cache, slot_mapping)


def mla_reshape_and_cache(
    kv_latent: torch.Tensor,
    k_rope: torch.Tensor,
    cache: torch.Tensor,
    slot_mapping: torch.Tensor,
) -> None:
    if kv_latent.is_cuda:
        ext = _require_ext("mla_reshape_and_cache")
        ext.mla_reshape_and_cache(
            kv_latent.contiguous(), k_rope.contiguous(), cache, slot_mapping.to(torch.int64)
        )
        return
    ref.mla_reshape_and_cache(kv_latent, k_rope, cache, slot_mapping)


# -- attention ---------------------------------------------------------------------


def paged_attention_decode(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,
    seq_lens: torch.Tensor,
    scale: float,
    sliding_window: int = -1,
    softcap: float = 0.0,
    sinks: Optional[torch.Tensor] = None,
    max_seq_len: Optional[int] = None,
    k_scale: float = 1.0,
    v_scale: float = 1.0,
) -> torch.Tensor:
    if q.is_cuda:
        ext = _require_ext("paged_attention_decode")
        out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        if max_seq_len is None:
            max_seq_len = int(seq_lens.max().item())
        ext.paged_attention_decode(
            out,
            q,
            k_cache,
            v_cache,
            block_tables.to(torch.int32),
            seq_lens.to(torch.int32),
            scale,
            sliding_window,
            softcap,
            sinks if sinks is not None else q.new_empty(0),
            max_seq_len,
            k_scale,
            v_scale,
        )
        return out
    return ref.paged_attention_decode(
        q, k_cache, v_cache, block_tables, seq_lens, scale, sliding_window, softcap, sinks
    )


def prefill_attention(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,
    seq_lens: torch.Tensor,
    query_lens: torch.Tensor,
    scale: float,
    sliding_window: int = -1,
    softcap: float = 0.0,
    sinks: Optional[torch.Tensor] = None,
    k_scale: float = 1.0,
    v_scale: float = 1.0,
) -> torch.Tensor:
    if q.is_cuda:
        ext = _require_ext("prefill_attention")
        out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        ql = query_lens.cpu().tolist()
        cu = [0]
        tile_req, tile_row0 = [], []
        for i, l in enumerate(ql):
            cu.append(cu[-1] + int(l))
            for r0 in range(0, int(l), 32):
                tile_req.append(i)
                tile_row0.append(r0)
        dev = q.device
        ext.prefill_attention(
            out,
            q,
            k_cache,
            v_cache,
            block_tables.to(torch.int32),
            seq_lens.to(torch.int32),
            torch.tensor(cu, dtype=torch.int32, device=dev),
            torch.tensor(tile_req, dtype=torch.int32, device=dev),
            torch.tensor(tile_row0, dtype=torch.int32, device=dev),
            scale,
            sliding_window,
            softcap,
            sinks if sinks is not None else q.new_empty(0),
            k_scale,
            v_scale,
        )
        return out
    return ref.prefill_attention(
        q, k_cache, v_cache, block_tables, seq_lens, query_lens, scale,
        sliding_window, softcap, sinks,
    )


def mla_paged_attention_decode(
    q_latent: torch.Tensor,
    q_rope: torch.Tensor,
    cache: torch.Tensor,
    block_tables: torch.Tensor,
    seq_lens: torch.Tensor,
    scale: float,
    max_seq_len: Optional[int] = None,
) -> torch.Tensor:
    if q_latent.is_cuda:
        ext = _require_ext("mla_paged_attention_decode")
        out = torch.empty_like(q_latent)
        if max_seq_len is None:
            max_seq_len = int(seq_lens.max().item())
        ext.mla_paged_attention_decode(
            out, q_latent.contiguous(), q_rope.contiguous(), cache,
            block_tables.to(torch.int32), seq_lens.to(torch.int32), scale,
            max_seq_len, q_latent.new_empty(0, dtype=torch.int32),
        )
        return out
    return ref.mla_paged_attention_decode(
        q_latent, q_rope, cache, block_tables, seq_lens, scale
    )


def dsa_paged_attention_decode(
    q_latent: torch.Tensor,
    q_rope: torch.Tensor,
    cache: torch.Tensor,
    block_tables: torch.Tensor,
    seq_lens: torch.Tensor,
    topk_indices: torch.Tensor,
    scale: float,
    max_seq_len: Optional[int] = None,
) -> torch.Tensor:
    """DeepSeek-V3.2 sparse MLA decode: attention restricted to each row's
    top-k token positions (-1 padded; a row starting -1 falls back dense)."""
    if q_latent.is_cuda:
        ext = _require_ext("dsa_paged_attention_decode")
        out = torch.empty_like(q_latent)
        if max_seq_len is None:
            max_seq_len = int(seq_lens.max().item())
        ext.mla_paged_attention_decode(
            out, q_latent.contiguous(), q_rope.contiguous(), cache,
            block_tables.to(torch.int32), seq_lens.to(torch.int32), scale,
            max(max_seq_len, topk_indices.shape[1]),
            topk_indices.to(torch.int32).contiguous(),
        )
        return out
    return ref.dsa_paged_attention_decode(
        q_latent, q_rope, cache, block_tables, seq_lens, topk_indices, scale
    )


def msa_paged_attention_decode(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,
    seq_lens: torch.Tensor,
    token_positions: torch.Tensor,
    scale: float,
) -> torch.Tensor:
    """MiniMax-style sparse attention over explicit token positions."""
    if q.is_cuda:
        ext = _require_ext("msa_paged_attention_decode")
        out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        ext.msa_paged_attention_decode(
            out, q, k_cache, v_cache, block_tables.to(torch.int32),
            seq_lens.to(torch.int32),
            token_positions.to(torch.int32).contiguous(), scale,
        )
        return out
    return ref.msa_paged_attention_decode(
        q, k_cache, v_cache, block_tables, seq_lens, token_positions, scale
    )


# indexer / block-score helpers run as torch compositions on both CPU and GPU
# (GEMV-scale work; the hot sparse gathers above are the HIP kernels)
def dsa_indexer_scores(q_index, index_cache, head_weights, block_tables,
                       seq_lens, max_ctx=None):
    """Weighted relu(q.k) scores over the paged index cache (DeepSeek-V3.2
    indexer). HIP MFMA kernel on GPU for the shared-key layout; torch
    reference elsewhere (reference Metal analogue: dsa_indexer.metal).
    Pass max_ctx (e.g. the graph ctx bucket) to stay hipGraph-capturable —
    int(seq_lens.max()) syncs the host otherwise."""
    if (
        q_index.is_cuda and index_cache.dim() == 3
        and q_index.dtype == torch.bfloat16
        and index_cache.dtype == torch.bfloat16
        and q_index.shape[1] <= 64 and q_index.shape[2] in (64, 128)
    ):
        ext = _require_ext("dsa_indexer_scores")
        if max_ctx is None:
            max_ctx = int(seq_lens.max())
        return ext.dsa_indexer_scores(
            q_index, index_cache, head_weights.float(),
            block_tables.int(), seq_lens.int(), max_ctx,
        )
    return ref.dsa_indexer_scores(
        q_index, index_cache, head_weights, block_tables, seq_lens
    )


def store_indexer_cache(index_keys, index_cache, slot_mapping):
    """Scatter indexer keys by slot (pad slots -1 land in the trash block)."""
    if index_keys.is_cuda and index_keys.dtype == torch.bfloat16 \
            and index_cache.dtype == torch.bfloat16:
        ext = _require_ext("store_indexer_cache")
        bs = index_cache.shape[1]
        trash_slot = (index_cache.shape[0] - 1) * bs
        ext.store_indexer_cache(
            index_keys.reshape(index_keys.shape[0], -1),
            index_cache, slot_mapping.long(), trash_slot,
        )
        return
    ref.store_indexer_cache(index_keys, index_cache, slot_mapping)


def msa_block_scores(q, k_cache, block_tables, seq_lens, sparse_block):
    """Mean-pooled per-sparse-block K scores vs the head-mean query
    (MiniMax-M3 indexer phase 1; reference msa_indexer.metal)."""
    if q.is_cuda and q.dtype == torch.bfloat16 \
            and k_cache.dtype == torch.bfloat16 and q.shape[2] in (64, 128):
        ext = _require_ext("msa_block_scores")
        max_sb = (int(seq_lens.max()) + sparse_block - 1) // sparse_block
        return ext.msa_block_scores(
            q, k_cache, block_tables.int(), seq_lens.int(), sparse_block,
            max(1, max_sb),
        )
    return ref.msa_block_scores(q, k_cache, block_tables, seq_lens, sparse_block)


def msa_topk_tokens(block_scores, seq_lens, sparse_block, topk_blocks,
                    init_blocks=1, local_blocks=2):
    """Top-k sparse blocks expanded to sorted token positions (-1 padded),
    always keeping the init/local blocks (phase 2)."""
    if block_scores.is_cuda:
        ext = _require_ext("msa_topk_tokens")
        keep_blocks = topk_blocks + init_blocks + local_blocks
        max_sb = block_scores.shape[1]
        max_positions = min(int(seq_lens.max()),
                            min(keep_blocks, max_sb) * sparse_block)
        return ext.msa_topk_tokens(
            block_scores.float(), seq_lens.int(), sparse_block, topk_blocks,
            init_blocks, local_blocks, max(1, max_positions),
        )
    return ref.msa_topk_tokens(
        block_scores, seq_lens, sparse_block, topk_blocks, init_blocks,
        local_blocks,
    )


# -- activations ----------------------------------------------------------------------


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    if x.is_cuda:
        ext = _require_ext("silu_and_mul")
        half = x.shape[-1] // 2
        out = x.new_empty(*x.shape[:-1], half)
        ext.silu_and_mul(out, x.contiguous())
        return out
    return ref.silu_and_mul(x)


def gelu_and_mul(x: torch.Tensor) -> torch.Tensor:
    if x.is_cuda:
        ext = _require_ext("gelu_and_mul")
        half = x.shape[-1] // 2
        out = x.new_empty(*x.shape[:-1], half)
        ext.gelu_and_mul(out, x.contiguous())
        return out
    return ref.gelu_and_mul(x)


# -- dense linear (skinny-M fast path) ------------------------------------------

# Disabled by default: hipBLASLt reaches 3.4-5.8 TB/s of weight streaming at
# decode shapes when measured in isolation (scripts/bench_gemm.py) and beats
# this kernel's simple 2-barrier structure; opt in for experiments.
SKINNY_GEMM_MAX_M = int(os.environ.get("PARALLAX_SKINNY_GEMM_MAX_M", "0"))
LT_GEMM = os.environ.get("PARALLAX_LT_GEMM", "1") != "0"


def linear(
    x: torch.Tensor, weight: torch.Tensor, bias: Optional[torch.Tensor] = None
) -> torch.Tensor:
    """F.linear with a custom streaming kernel for decode-sized M (hipBLASLt
    tile picks are 2-5x off the weight-streaming roofline at M <= 256)."""
    if (
        x.is_cuda
        and x.dim() == 2
        and x.dtype == torch.bfloat16
        and 0 < x.shape[0] <= SKINNY_GEMM_MAX_M
        and weight.shape[0] % 64 == 0
        and weight.shape[1] % 32 == 0
        and x.stride(1) == 1
    ):
        ext = _require_ext("skinny_gemm")
        out = torch.empty(
            x.shape[0], weight.shape[0], dtype=x.dtype, device=x.device
        )
        if ext.skinny_gemm(out, x, weight,
                           bias if bias is not None else x.new_empty(0)):
            return out
    if (
        LT_GEMM
        and x.is_cuda
        and x.dim() == 2
        and x.dtype == torch.bfloat16
        and weight.dtype == torch.bfloat16
        and x.is_contiguous()
        and weight.is_contiguous()
    ):
        # tuned hipBLASLt: the default heuristic's decode-shape tile picks
        # are ~2.5x off the weight-streaming roofline (profiles/README.md);
        # first call per shape times the heuristic candidates (engine warmup,
        # outside graph capture) and later calls replay the winner
        ext = _require_ext("lt_linear")
        y = ext.lt_linear(x, weight)
        return y if bias is None else y + bias
    return torch.nn.functional.linear(x, weight, bias)


# -- MoE ----------------------------------------------------------------------


def fused_moe_forward(
    x: torch.Tensor,          # [T, H] bf16
    w_gate_up: torch.Tensor,  # [E, 2I, H] bf16
    w_down: torch.Tensor,     # [E, H, I] bf16
    topk_ids: torch.Tensor,   # [T, k] long
    topk_weights: torch.Tensor,  # [T, k] float
    activation: str = "silu",
    limit: float = 0.0,
    bias_gate_up: Optional[torch.Tensor] = None,  # [E, 2I] bf16 (gpt-oss)
) -> torch.Tensor:
    """Grouped-GEMM MoE (device-side routing; graph-capture safe). Returns
    fp32 [T, H]."""
    assert x.is_cuda
    ext = _require_ext("moe_forward")
    T, H = x.shape
    E = w_gate_up.shape[0]
    k = topk_ids.shape[1]
    flat = topk_ids.reshape(-1)
    sorted_ids, perm = torch.sort(flat)
    seg = torch.searchsorted(
        sorted_ids, torch.arange(E + 1, device=x.device, dtype=sorted_ids.dtype)
    ).to(torch.int32)
    route_w = topk_weights.reshape(-1).float()[perm].contiguous()
    out = torch.zeros(T, H, dtype=torch.float32, device=x.device)
    ext.moe_forward(out, x.contiguous(), w_gate_up, w_down, perm.contiguous(),
                    seg.contiguous(), route_w, k, activation == "gelu", limit,
                    bias_gate_up if bias_gate_up is not None
                    else x.new_empty(0))
    return out


_FP8_LIN_BUFS = {}


def linear_fp8(x: torch.Tensor, w_q: torch.Tensor, w_scale: torch.Tensor,
               bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """W8A8 dense GEMM through hipBLASLt fp8: x quantized per-tensor on the
    fly into persistent per-shape buffers (graph-capture safe: the scale and
    xq pointers baked into a captured graph never move). CPU fallback
    dequantizes (plumbing tests)."""
    if not x.is_cuda:
        w = (w_q.float() * w_scale.float()).to(torch.float32)
        y = torch.nn.functional.linear(x.float(), w).to(x.dtype)
        return y if bias is None else y + bias
    ext = _require_ext("lt_linear_fp8")
    M, K = x.shape
    key = (M, K, x.device.index)
    bufs = _FP8_LIN_BUFS.get(key)
    if bufs is None:
        bufs = (
            torch.empty(M, K, dtype=torch.float8_e4m3fn, device=x.device),
            torch.ones(1, dtype=torch.float32, device=x.device),
        )
        _FP8_LIN_BUFS[key] = bufs
    xq, xs = bufs
    xf = x.float()
    xs.copy_((xf.abs().amax() / 448.0).clamp_min(1e-8).reshape(1))
    xq.copy_((xf / xs).clamp(-448.0, 448.0))
    y = ext.lt_linear_fp8(xq, w_q, xs, w_scale)
    return y if bias is None else y + bias


def quantize_fp8_weight(w: torch.Tensor):
    """Per-output-channel fp8-E4M3 quantization of a weight tensor whose rows
    are output channels along dim -2 ([..., N, K] -> fp8 [..., N, K] +
    fp32 dequant scale [..., N]). OCP e4m3fn — the gfx950-native encoding."""
    amax = w.float().abs().amax(dim=-1).clamp_min(1e-8)
    scale = amax / 448.0
    q = (w.float() / scale.unsqueeze(-1)).clamp(-448.0, 448.0)
    return q.to(torch.float8_e4m3fn), scale.contiguous()


def fused_moe_forward_fp8(
    x: torch.Tensor,             # [T, H] bf16
    w_gate_up: torch.Tensor,     # [E, 2I, H] fp8 e4m3
    w_gu_scale: torch.Tensor,    # [E, 2I] fp32
    w_down: torch.Tensor,        # [E, H, I] fp8 e4m3
    w_down_scale: torch.Tensor,  # [E, H] fp32
    topk_ids: torch.Tensor,      # [T, k] long
    topk_weights: torch.Tensor,  # [T, k] float
    activation: str = "silu",
    limit: float = 0.0,
) -> torch.Tensor:
    """W8A8 grouped-GEMM MoE on the fp8 MFMA path (activations quantized
    per-token on device). Returns fp32 [T, H]."""
    assert x.is_cuda
    ext = _require_ext("moe_forward_fp8")
    T, H = x.shape
    E = w_gate_up.shape[0]
    k = topk_ids.shape[1]
    flat = topk_ids.reshape(-1)
    sorted_ids, perm = torch.sort(flat)
    seg = torch.searchsorted(
        sorted_ids, torch.arange(E + 1, device=x.device, dtype=sorted_ids.dtype)
    ).to(torch.int32)
    route_w = topk_weights.reshape(-1).float()[perm].contiguous()
    out = torch.zeros(T, H, dtype=torch.float32, device=x.device)
    ext.moe_forward_fp8(out, x.contiguous(), w_gate_up, w_gu_scale, w_down,
                        w_down_scale, perm.contiguous(), seg.contiguous(),
                        route_w, k, activation == "gelu", limit)
    return out


# -- sampling (torch ops; GPU path uses torch's ROCm kernels — not a hot spot
#    relative to the model forward, custom kernel is a later optimization) -------

_SAMPLE_CALLS = 0


def sample_tokens(logits, temperatures, top_ps, top_ks, min_ps,
                  generator=None):
    """Batched sampling. GPU rows with no top-k/p/min-p filtering take the
    fused one-pass gumbel-max HIP kernel (the torch composition makes ~6
    passes over the [B, vocab] buffer); filtered rows fall back to the torch
    path. Seed = generator seed + a per-call counter so repeated calls draw
    fresh noise deterministically."""
    global _SAMPLE_CALLS
    if logits.is_cuda and has_extension():
        B, V = logits.shape
        temps = [float(t) for t in temperatures]
        greedy = [t <= 0.0 for t in temps]
        need_filter = any(
            (not g) and (0 < int(k) < V or float(p) < 1.0 or float(m) > 0.0)
            for g, k, p, m in zip(greedy, top_ks, top_ps, min_ps)
        )
        if not need_filter:
            ext = _require_ext("sample_gumbel")
            _SAMPLE_CALLS += 1
            base = generator.initial_seed() if generator is not None else 0x5eed
            inv_t = torch.tensor(
                [1.0 / max(t, 1e-6) for t in temps], dtype=torch.float32,
                device=logits.device,
            )
            gmask = torch.tensor(greedy, dtype=torch.uint8,
                                 device=logits.device)
            return ext.sample_gumbel(
                logits.float().contiguous(), inv_t, gmask,
                (base + _SAMPLE_CALLS) & 0x7FFFFFFFFFFF,
            )
    return ref.sample_tokens(logits, temperatures, top_ps, top_ks, min_ps,
                             generator)


apply_penalties = ref.apply_penalties
