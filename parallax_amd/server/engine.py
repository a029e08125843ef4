"""Single-host pipeline-parallel serving engine (the xGMI fast path).

Design: SPMD replicated scheduling. Every pipeline stage runs an identical
continuous-batching scheduler on identical request state; only hidden states
move point-to-point (RCCL send/recv over xGMI) and sampled tokens are broadcast
once per step. This removes the reference's per-hop serialize+RPC cost
(SURVEY.md §3.3 — decode pays a Lattica RPC of (1, hidden) per token per hop)
for the in-host path; the packet-driven peer executor (p2p/) remains the
multi-host path.

Replicated determinism: admission, batch formation, cache allocation and LRU
eviction are all single-threaded and driven by the same event sequence on every
rank, so per-rank block tables stay identical without any wire traffic.
"""

from __future__ import annotations

import os
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist

from .. import ops
from ..models import get_model_class
from ..models.config import ModelConfig
from ..models.forward_meta import ForwardMeta
from ..parallel.comm import CommContext, get_comm
from ..utils.logging_config import get_logger
from .cache_manager import CacheManager
from .kv_cache import (
    KVCacheSpec,
    MLAKVCache,
    PagedKVCache,
    build_block_table_tensor,
    slot_mapping_for_positions,
)
from .request import InitialRequest, new_request_id
from .sampler import Sampler
from .sampling_params import SamplingParams
from .scheduler import PrefillChunk, Scheduler

logger = get_logger("server.engine")

# async decode pipelining kill-switch (PARALLAX_ASYNC_DECODE=0 disables)
ASYNC_DECODE = os.environ.get("PARALLAX_ASYNC_DECODE", "1") != "0"


@dataclass
class StepOutput:
    rid: str
    token_id: int
    finished: bool
    finish_reason: Optional[str] = None
    logprob: Optional[float] = None


@dataclass
class EngineArgs:
    block_size: int = 32
    max_batch_size: int = 128
    max_num_tokens_per_batch: int = 16384
    prefill_chunk_size: int = 8192
    cache_memory_fraction: float = 0.80
    num_kv_blocks: Optional[int] = None  # override (tests / CPU)
    micro_batches: int = 1
    enable_prefix_cache: bool = True
    dtype: torch.dtype = torch.bfloat16
    seed: Optional[int] = 0
    request_timeout_s: float = 600.0
    start_layer: Optional[int] = None   # explicit layer range (decentralized mode)
    end_layer: Optional[int] = None
    max_model_len: int = 8192           # context ceiling (sizes graph buffers)
    enable_graphs: bool = True          # hipGraph-captured decode forward
    kv_cache_dtype: str = "auto"        # "auto" (= engine dtype) | "fp8"(e4m3)
    moe_weight_dtype: str = "auto"      # "auto" (= engine dtype) | "fp8"(W8A8)
    enable_routing_stats: bool = False  # per-expert MoE routing counters
    prefill_priority: bool = True       # prefills before decodes in a step
    linear_weight_dtype: str = "auto"   # dense GEMMs: "auto" | "fp8"(W8A8)


def partition_layers(num_layers: int, pp_size: int, pp_rank: int) -> Tuple[int, int]:
    """Contiguous near-even split; earlier stages get the remainder (they also
    host the embedding, but on 288 GB the difference is noise)."""
    base, rem = divmod(num_layers, pp_size)
    start = pp_rank * base + min(pp_rank, rem)
    return start, start + base + (1 if pp_rank < rem else 0)


class Engine:
    def __init__(
        self,
        cfg: ModelConfig,
        args: EngineArgs = EngineArgs(),
        comm: Optional[CommContext] = None,
        model_path: Optional[str] = None,
        random_weights: bool = False,
        lora_path: Optional[str] = None,
    ):
        self.cfg = cfg
        self.args = args
        self.comm = comm or get_comm()
        self.device = self.comm.device

        if args.start_layer is not None:
            start, end = args.start_layer, args.end_layer
        else:
            start, end = partition_layers(cfg.num_layers, self.comm.pp_size, self.comm.pp_rank)
        self.start_layer, self.end_layer = start, end

        model_cls = get_model_class(cfg.architecture)
        with torch.device("meta"):
            pass  # (meta-init then materialize is a later memory optimization)
        self.model = model_cls(cfg, start, end)
        if random_weights:
            self.model.init_random()
        elif model_path is not None:
            from .shard_loader import load_shard_weights

            load_shard_weights(self.model, model_path, lora_path=lora_path)
        if hasattr(self.model, "finalize_weights"):
            self.model.finalize_weights()
        self.model = self.model.to(device=self.device, dtype=args.dtype)
        # rope cache stays fp32
        self.model.rope_cache = self.model.rope_cache.float()
        self.model.eval()
        if args.moe_weight_dtype == "fp8":
            from ..models.moe import FusedMoE

            n_q = 0
            for m in self.model.modules():
                if isinstance(m, FusedMoE):
                    m.quantize_fp8()
                    n_q += 1
            logger.info("fp8 MoE: quantized %d expert blocks (W8A8)", n_q)
        if args.enable_routing_stats:
            from ..models.moe import FusedMoE

            for m in self.model.modules():
                if isinstance(m, FusedMoE):
                    m.enable_routing_stats()  # before any graph capture
        if args.linear_weight_dtype == "fp8":
            from ..parallel.layers import (
                ColumnParallelLinear, MergedColumnParallelLinear,
                RowParallelLinear,
            )

            n_q = 0
            for name, m in self.model.named_modules():
                if isinstance(m, (ColumnParallelLinear,
                                  MergedColumnParallelLinear,
                                  RowParallelLinear)) \
                        and "lm_head" not in name:
                    m.quantize_fp8()
                    n_q += 1
            logger.info("fp8 dense: quantized %d linears (W8A8)", n_q)

        # hybrid stacks: paged KV only for the full-attention layers
        self.local_layer_types = [cfg.layer_type(g) for g in range(start, end)]
        n_kv_layers = sum(
            1 for t in self.local_layer_types if t != "linear_attention"
        ) or 1
        self.linear_cache = None
        if cfg.has_linear_layers:
            from .kv_cache import LinearStateCache

            n_linear = sum(1 for t in self.local_layer_types if t == "linear_attention")
            conv_dim = (
                2 * cfg.linear_num_key_heads * cfg.linear_key_head_dim
                + cfg.linear_num_value_heads * cfg.linear_value_head_dim
            )
            self.linear_cache = LinearStateCache(
                max(1, n_linear),
                conv_state_shape=(conv_dim, cfg.linear_conv_kernel_dim - 1),
                recurrent_state_shape=(
                    cfg.linear_num_value_heads, cfg.linear_key_head_dim,
                    cfg.linear_value_head_dim,
                ),
                # +1: the last slot is a scratch slot for graph-capture pad
                # rows (DecodeGraphRunner.linear_scratch_slot); the cache
                # manager only hands out the first max_batch_size + 8
                num_slots=args.max_batch_size + 9,
                device=self.device,
                dtype=args.dtype if args.dtype != torch.float32 else torch.float32,
            )

        self.is_mla = cfg.is_mla
        if self.is_mla:
            mla_block_bytes = MLAKVCache.bytes_per_block(
                end - start, cfg.kv_lora_rank, cfg.qk_rope_head_dim,
                args.block_size, args.dtype,
                index_dim=cfg.index_head_dim if cfg.is_dsa else 0,
            )
            if args.num_kv_blocks:
                num_blocks = args.num_kv_blocks
            else:
                if self.device.type == "cuda":
                    free_b, _ = torch.cuda.mem_get_info(self.device)
                    budget = int(free_b * args.cache_memory_fraction) - (2 << 30)
                else:
                    budget = 1 << 30
                num_blocks = max(16, budget // mla_block_bytes)
            mla_dtype = args.dtype if args.dtype != torch.float32 else torch.float32
            if args.kv_cache_dtype == "fp8":
                assert self.device.type == "cuda", "fp8 KV needs the HIP kernels"
                mla_dtype = torch.float8_e4m3fn
            self.kv_cache = MLAKVCache(
                end - start, cfg.kv_lora_rank, cfg.qk_rope_head_dim,
                args.block_size, num_blocks, self.device, mla_dtype,
                index_dim=cfg.index_head_dim if cfg.is_dsa else 0,
            )
        else:
            kv_dtype = args.dtype
            if args.kv_cache_dtype == "fp8":
                assert self.device.type == "cuda", "fp8 KV needs the HIP kernels"
                kv_dtype = torch.float8_e4m3fn
            spec = KVCacheSpec(
                num_layers=n_kv_layers,
                num_kv_heads=max(1, cfg.num_kv_heads // self.comm.tp_size),
                head_dim=cfg.head_dim,
                block_size=args.block_size,
                dtype=kv_dtype,
                index_dim=cfg.index_head_dim if cfg.is_msa else 0,
            )
            num_blocks = args.num_kv_blocks or CacheManager.num_blocks_from_memory(
                spec, self.device, args.cache_memory_fraction
            )
            self.kv_cache = PagedKVCache(spec, num_blocks, self.device)
        self.cache_manager = CacheManager(
            args.block_size, num_blocks,
            enable_prefix_cache=args.enable_prefix_cache and not cfg.has_linear_layers,
            num_linear_slots=(args.max_batch_size + 8) if cfg.has_linear_layers else 0,
        )
        self.scheduler = Scheduler(
            self.cache_manager,
            max_batch_size=args.max_batch_size,
            max_num_tokens_per_batch=args.max_num_tokens_per_batch,
            prefill_chunk_size=args.prefill_chunk_size,
            request_timeout_s=args.request_timeout_s,
            eos_token_ids=cfg.eos_token_ids,
            prefill_priority=args.prefill_priority,
        )
        self.sampler = Sampler(self.device, args.seed)
        self.graph_runner = None
        if args.enable_graphs and self.device.type == "cuda":
            from .graph_runner import DecodeGraphRunner

            self.graph_runner = DecodeGraphRunner(
                self.model, self.kv_cache, self.device, args.dtype,
                max_batch=args.max_batch_size, max_model_len=args.max_model_len,
                block_size=args.block_size, hidden_size=cfg.hidden_size,
                is_first_stage=self.comm.is_first_stage,
                is_last_stage=self.comm.is_last_stage,
                is_mla=self.is_mla,
                linear_cache=self.linear_cache,
                linear_scratch_slot=args.max_batch_size + 8,
            )
        self._pending_adds: List[InitialRequest] = []
        self._inflight = None       # async decode pipelining state
        # double-buffered pinned staging: step N's D2H copy must not overwrite
        # the buffer step N-1's finalize has yet to read
        self._async_pinned = [None, None]
        self._async_pin_idx = 0
        self._pending_aborts: List[str] = []
        self.step_count = 0
        # device-tensor PP sync state (no host-object broadcasts in the step path)
        self._flag_buf: Optional[torch.Tensor] = None
        self._tok_bcast_buf: Optional[torch.Tensor] = None
        self._lp_bcast_buf: Optional[torch.Tensor] = None
        self.object_sync_count = 0  # payload broadcasts (adds/aborts only)
        self._deferred_free: List[InitialRequest] = []
        logger.info(
            "engine up: layers [%d,%d) of %d, %d KV blocks x %d tokens, mla=%s, device %s",
            start, end, cfg.num_layers, num_blocks, args.block_size,
            self.is_mla, self.device,
        )

    # -- public API (head rank) ----------------------------------------------------

    def submit(
        self,
        prompt_token_ids: List[int],
        sampling_params: Optional[SamplingParams] = None,
        rid: Optional[str] = None,
    ) -> str:
        sp = sampling_params or SamplingParams()
        limit = min(self.args.max_model_len, self.cfg.max_position_embeddings)
        if len(prompt_token_ids) + sp.max_new_tokens > limit:
            # out-of-range positions would read past the rope table / graph
            # buffers (a device fault, not an error message) — reject up front
            raise ValueError(
                f"request length {len(prompt_token_ids)} + "
                f"{sp.max_new_tokens} new tokens exceeds the engine context "
                f"limit {limit} (max_model_len / max_position_embeddings)"
            )
        rid = rid or new_request_id()
        # list() snapshots are GIL-atomic; the step thread mutates these
        # structures concurrently (lock-free submit)
        if rid in self.scheduler.running \
                or any(r.rid == rid for r in list(self._pending_adds)) \
                or any(r.rid == rid for r in list(self.scheduler.wait_queue)):
            # a duplicate rid would collide in the cache-state and running
            # maps and silently corrupt block accounting
            raise ValueError(f"duplicate request id {rid!r}")
        req = InitialRequest(
            rid=rid,
            prompt_token_ids=list(prompt_token_ids),
            sampling_params=sp,
        )
        self._pending_adds.append(req)
        return req.rid

    def abort(self, rid: str) -> None:
        self._pending_aborts.append(rid)

    def routing_stats(self) -> Optional[Dict[str, List[int]]]:
        """Per-expert routed-token counts per MoE layer (None if disabled).
        Reference parity: enable_return_routed_experts observability."""
        if not self.args.enable_routing_stats:
            return None
        from ..models.moe import FusedMoE

        out: Dict[str, List[int]] = {}
        for name, m in self.model.named_modules():
            if isinstance(m, FusedMoE) and m.routing_counts is not None:
                out[name] = m.routing_counts[:-1].tolist()
        return out

    def warmup_gemms(self, ms: List[int]) -> None:
        """Pre-tune the hipBLASLt algo picks for every 2-D weight shape at
        the given M values. Per-shape tuning is one-time but otherwise lands
        inside the first prefill and pollutes TTFT; a long-lived server calls
        this at startup (the bench does too, before the timed phases)."""
        if self.device.type != "cuda":
            return
        seen = set()
        with torch.inference_mode():
            for m in sorted({int(x) for x in ms if x and x > 0}):
                for name, p in self.model.named_parameters():
                    if p.dim() != 2 or "embed_tokens" in name \
                            or p.dtype != self.args.dtype:
                        continue
                    key = (m, p.shape[0], p.shape[1])
                    if key in seen:
                        continue
                    seen.add(key)
                    x = torch.zeros(m, p.shape[1], dtype=p.dtype,
                                    device=self.device)
                    ops.linear(x, p)
                # fp8-quantized linears hold buffers, not parameters
                for name, mod in self.model.named_modules():
                    w = getattr(mod, "weight_fp8", None)
                    if w is None or not getattr(mod, "fp8", False):
                        continue
                    key = (m, w.shape[0], w.shape[1], "fp8")
                    if key in seen:
                        continue
                    seen.add(key)
                    x = torch.zeros(m, w.shape[1], dtype=self.args.dtype,
                                    device=self.device)
                    ops.linear_fp8(x, w, mod.weight_scale)
        torch.cuda.synchronize()
        logger.info("gemm warmup: tuned %d shapes", len(seen))

    def warmup_serving(self) -> None:
        """Pre-tune the GEMM algo picks for EVERY decode graph bucket plus the
        prefill chunk shapes. Without this, serving pays a multi-second
        hipBLASLt tuning stall the first time each batch-size bucket appears,
        and a multi-second graph capture the first time each (batch, ctx)
        bucket pair appears (measured: 44 s p50 TTFT at request-rate 16 from
        exactly these two stalls compounding during the arrival burst)."""
        hot = [self.args.max_num_tokens_per_batch,
               self.args.prefill_chunk_size, self.args.max_batch_size]
        minor = (list(self.graph_runner.buckets)
                 if self.graph_runner is not None else [])
        prev = os.environ.get("PARALLAX_LT_TUNE_MS")
        self.warmup_gemms(hot)  # deep search for the steady-state shapes
        try:
            os.environ["PARALLAX_LT_TUNE_MS"] = "100"
            self.warmup_gemms(minor)  # quick picks for transient buckets
        finally:
            # steady-state serving: shapes not warmed here (arbitrary prefill
            # token totals under mixed arrivals) take the hipBLASLt heuristic
            # pick instantly instead of a ~1.5 s timing loop per new shape —
            # measured 270 ms p50 TPOT at request-rate 16 from those stalls
            os.environ["PARALLAX_LT_TUNE_MS"] = "0"
        if self.graph_runner is not None:
            self.graph_runner.capture_all()

    def set_grammar_vocab(self, vocab: List[str]) -> None:
        """Enable json_schema constrained decoding: vocab[i] is token i's
        text. Must be called on every rank that samples (in practice: all)."""
        self.sampler.grammar_vocab = vocab

    @property
    def has_work(self) -> bool:
        # an in-flight async step / deferred KV frees still need one more
        # step() to drain even when the scheduler itself is empty (the last
        # running request can finish one step late)
        return (self.scheduler.has_work or bool(self._pending_adds)
                or self._inflight is not None or bool(self._deferred_free))

    # -- the step ---------------------------------------------------------------------

    def step(self) -> List[StepOutput]:
        """One engine iteration on every rank. Returns newly sampled tokens
        (meaningful on the head rank; identical on all ranks).

        Async decode pipelining (single-rank engines): when every running
        request's finish condition is value-independent (ignore_eos, no stop
        tokens, no penalties/logprobs, >= 2 tokens remaining), the step
        ENQUEUES this iteration's forward+sampling and returns the PREVIOUS
        iteration's tokens — the next step's input ids are copied
        device-to-device from the sampler output, so the host-side scheduler
        work overlaps the GPU instead of gating it (~3 ms/step at batch 512).
        Steps outside that envelope (prefills, finishes, eos-sensitive
        requests) drain the in-flight step first and run synchronously."""
        self._sync_ingress()
        timeout_sweep = self.step_count % 64 == 0
        outputs: List[StepOutput] = []
        if self._inflight is not None and (
            self._pending_aborts or timeout_sweep
        ):
            outputs.extend(self._finalize_inflight())
        for req in self.scheduler.sweep_aborted():
            outputs.append(StepOutput(rid=req.rid, token_id=-1, finished=True,
                                      finish_reason=req.status.finish_reason))
        for req in self.scheduler.drain_aborted_waiting():
            outputs.append(StepOutput(rid=req.rid, token_id=-1, finished=True,
                                      finish_reason=req.status.finish_reason))
        # the single-host engine reports finishes via StepOutput; drain the
        # scheduler's bookkeeping list so a long-lived server doesn't grow it
        self.scheduler.drain_finished()
        if timeout_sweep:
            for req in self.scheduler.sweep_timeouts():
                outputs.append(StepOutput(rid=req.rid, token_id=-1,
                                          finished=True,
                                          finish_reason=req.status.finish_reason))
        self.scheduler.admit_requests()
        batch = self.scheduler.form_batch()
        self.step_count += 1
        if batch.is_empty:
            if self._inflight is not None:
                outputs.extend(self._finalize_inflight())
            return outputs

        if self._async_eligible(batch):
            prev = self._inflight
            if prev is not None and prev["rids"] == [r.rid for r in batch.decode_reqs]:
                self._enqueue_async(batch.decode_reqs, input_dev=prev["tokens_dev"])
                outputs.extend(self._finalize_inflight(prev))
                return outputs
            if prev is not None:  # membership changed: drain, then refill
                outputs.extend(self._finalize_inflight())
                # the drain can detect one-step-late finishes: requests the
                # formed batch still lists may have just been released
                batch.decode_reqs = [
                    r for r in batch.decode_reqs
                    if r.rid in self.scheduler.running
                ]
                if not batch.decode_reqs:
                    return outputs
            self._enqueue_async(batch.decode_reqs, input_dev=None)
            return outputs

        if self._inflight is not None:
            # the drain patches token values AND may finish requests
            # one step late — drop freed requests from the formed batch
            # before any forward touches their (released) cache state
            outputs.extend(self._finalize_inflight())
            batch.decode_reqs = [
                r for r in batch.decode_reqs
                if r.rid in self.scheduler.running
            ]
            if batch.is_empty:
                return outputs
        sample_reqs: List[InitialRequest] = []
        logits_parts: List[torch.Tensor] = []

        if batch.prefill_chunks:
            logits, samp = self._run_prefill(batch.prefill_chunks)
            if samp:
                sample_reqs.extend(samp)
                if logits is not None:
                    logits_parts.append(logits)
        if batch.decode_reqs:
            logits = self._run_decode(batch.decode_reqs)
            sample_reqs.extend(batch.decode_reqs)
            if logits is not None:
                logits_parts.append(logits)

        # bookkeeping that must precede token commit
        for chunk in batch.prefill_chunks:
            self.scheduler.complete_prefill_chunk(chunk)

        if sample_reqs:
            sampled = self._sample_and_broadcast(logits_parts, sample_reqs)
            for req, (tok, lp) in zip(sample_reqs, sampled):
                finished = self.scheduler.commit_token(req.rid, tok)
                outputs.append(
                    StepOutput(
                        rid=req.rid,
                        token_id=tok,
                        finished=finished is not None,
                        finish_reason=req.status.finish_reason,
                        logprob=lp,
                    )
                )
        return outputs

    # -- async decode pipelining ----------------------------------------------------

    PLACEHOLDER_TOKEN = -1

    def _async_eligible(self, batch) -> bool:
        """Safe envelope for one-step-late token commit: finishes must be
        value-independent and the request set stable through this step."""
        if not ASYNC_DECODE:
            return False
        if self.comm.world_size > 1 or batch.prefill_chunks or not batch.decode_reqs:
            return False
        if self._pending_adds or self._pending_aborts:
            return False
        for r in batch.decode_reqs:
            sp = r.sampling_params
            # eos/stop-token finishes are handled ONE STEP LATE by
            # _finalize_inflight (rollback: truncate the zombie placeholder,
            # defer the cache release until the in-flight step's event is
            # synced) — so the common serving case keeps the pipelining.
            # logprobs/json_schema/penalties stay sync: they need token
            # VALUES on the host before the next forward.
            if sp.logprobs or sp.json_schema or sp.logit_bias \
                    or sp.seed is not None:
                return False
            if (sp.repetition_penalty != 1.0 or sp.presence_penalty != 0.0
                    or sp.frequency_penalty != 0.0):
                return False
            # placeholder commit must not reach the length limit
            if r.num_output_tokens + 2 > sp.max_new_tokens:
                return False
        return True

    def _enqueue_async(self, reqs: List[InitialRequest],
                       input_dev: Optional[torch.Tensor]) -> None:
        """Launch this step's forward + sampling without waiting for tokens.
        input_dev: previous step's sampled-token tensor (device) feeding this
        step's input ids row-for-row; None = pipeline fill (host values)."""
        input_ids = [r.output_token_ids[-1] for r in reqs]
        if self.graph_runner is not None:
            positions, slots, btabs, seq_lens = [], [], [], []
            for r in reqs:
                state = self.cache_manager.get(r.rid)
                pos = r.total_len - 1
                positions.append(pos)
                slots.append(
                    state.block_table[pos // self.args.block_size]
                    * self.args.block_size + pos % self.args.block_size
                )
                btabs.append(state.block_table)
                seq_lens.append(r.total_len)
            lin_slots = None
            if self.linear_cache is not None:
                lin_slots = [
                    self.cache_manager.get(r.rid).linear_slot or 0 for r in reqs
                ]
            with torch.inference_mode():
                logits = self.graph_runner.run(
                    input_ids, positions, slots, btabs, seq_lens,
                    rids=[r.rid for r in reqs], linear_slots=lin_slots,
                    input_ids_dev=input_dev,
                )[: len(reqs)]
        else:
            meta, ids = self._build_decode_meta(reqs)
            if input_dev is not None:
                ids = input_dev.view(-1)[: len(reqs)].to(ids.device, ids.dtype)
            with torch.inference_mode():
                logits = self._pipeline_forward(meta, ids, meta.logits_indices)
        sp = [r.sampling_params for r in reqs]
        with torch.inference_mode():
            tokens_dev = ops.sample_tokens(
                logits,
                [s.temperature for s in sp],
                [s.top_p for s in sp],
                [s.top_k for s in sp],
                [s.min_p for s in sp],
                generator=self.sampler.generator,
            )
        if self.device.type == "cuda":
            self._async_pin_idx ^= 1
            pinned = self._async_pinned[self._async_pin_idx]
            if pinned is None or pinned.numel() < len(reqs):
                pinned = torch.empty(
                    max(len(reqs), self.args.max_batch_size),
                    dtype=tokens_dev.dtype, pin_memory=True,
                )
                self._async_pinned[self._async_pin_idx] = pinned
            pinned[: len(reqs)].copy_(tokens_dev, non_blocking=True)
            event = torch.cuda.Event()
            event.record()
        else:
            pinned, event = tokens_dev, None
        # placeholder commit: lengths advance now, values patch at finalize
        slots_idx = []
        for r in reqs:
            self.scheduler.commit_token(r.rid, self.PLACEHOLDER_TOKEN)
            slots_idx.append(len(r.output_token_ids) - 1)
        self._inflight = {
            "rids": [r.rid for r in reqs],
            "reqs": list(reqs),
            "tokens_dev": tokens_dev,
            "pinned": pinned,
            "event": event,
            "out_idx": slots_idx,
        }

    def _finalize_inflight(self, flight=None) -> List[StepOutput]:
        f = flight if flight is not None else self._inflight
        if f is None:
            return []
        drained = flight is None  # no newer flight exists after this one
        if drained:
            self._inflight = None
        if f["event"] is not None:
            f["event"].synchronize()
        # the event sync above covers the zombie KV writes that deferred
        # these frees (see release_keep_cache)
        for req in self._deferred_free:
            self.scheduler.free_cache(req)
        self._deferred_free.clear()
        toks = f["pinned"][: len(f["reqs"])].tolist()
        outputs = []
        for req, tok, idx in zip(f["reqs"], toks, f["out_idx"]):
            if idx >= len(req.output_token_ids):
                continue  # zombie: request finished before this step's token
            if req.output_token_ids[idx] == self.PLACEHOLDER_TOKEN:
                req.output_token_ids[idx] = tok
            finished = False
            if req.rid in self.scheduler.running:
                # one-step-late eos/stop/length detection: drop any newer
                # placeholders, test, restore them if the request continues
                tail = len(req.output_token_ids) - (idx + 1)
                del req.output_token_ids[idx + 1:]
                if req.check_finished():
                    finished = True
                    self.scheduler.release_keep_cache(req)
                    if drained:
                        self.scheduler.free_cache(req)
                    else:
                        self._deferred_free.append(req)
                else:
                    req.output_token_ids.extend(
                        [self.PLACEHOLDER_TOKEN] * tail
                    )
            outputs.append(
                StepOutput(rid=req.rid, token_id=tok, finished=finished,
                           finish_reason=req.status.finish_reason
                           if finished else None)
            )
        return outputs

    # -- ingress replication ----------------------------------------------------------

    def _sync_ingress(self) -> None:
        """Replicate pending adds/aborts to every rank.

        Steady-state decode pays exactly ONE tiny device-tensor broadcast (the
        payload-length flag, usually 0) — never an object broadcast. Only when
        requests actually arrive or abort does a second uint8-tensor broadcast
        carry the serialized payload (both are RCCL device collectives on GPU;
        the reference's per-hop pickle+RPC cost, p2p/server.py:628-755, has no
        analogue here). ``object_sync_count`` counts payload broadcasts so
        tests can assert the decode hot path stays object-free."""
        if self.comm.world_size > 1:
            import pickle

            if self._flag_buf is None:
                self._flag_buf = torch.zeros(
                    1, dtype=torch.int64, device=self.device
                )
            payload_bytes = b""
            if self.comm.rank == 0:
                # snapshot: the HTTP thread may append concurrently; ranks
                # must drain exactly what was serialized (SPMD determinism)
                n_adds = len(self._pending_adds)
                n_abrt = len(self._pending_aborts)
                if n_adds or n_abrt:
                    payload_bytes = pickle.dumps(
                        (
                            [
                                (r.rid, r.prompt_token_ids,
                                 r.sampling_params.to_dict())
                                for r in self._pending_adds[:n_adds]
                            ],
                            self._pending_aborts[:n_abrt],
                        )
                    )
                self._flag_buf.fill_(len(payload_bytes))
            dist.broadcast(self._flag_buf, src=0)
            n = int(self._flag_buf.item())
            if n > 0:
                self.object_sync_count += 1
                if self.comm.rank == 0:
                    buf = torch.frombuffer(
                        bytearray(payload_bytes), dtype=torch.uint8
                    ).to(self.device)
                else:
                    buf = torch.empty(n, dtype=torch.uint8, device=self.device)
                dist.broadcast(buf, src=0)
                if self.comm.rank != 0:
                    adds, aborts = pickle.loads(bytes(buf.cpu().numpy()))
                    self._pending_adds = [
                        InitialRequest(
                            rid=rid,
                            prompt_token_ids=toks,
                            sampling_params=SamplingParams.from_dict(sp),
                        )
                        for rid, toks, sp in adds
                    ]
                    self._pending_aborts = list(aborts)
        if self.comm.world_size > 1 and self.comm.rank == 0:
            adds = self._pending_adds[:n_adds]
            aborts = self._pending_aborts[:n_abrt]
            del self._pending_adds[:n_adds]
            del self._pending_aborts[:n_abrt]
        else:
            # lock-free handoff from the submit thread: snapshot the count,
            # consume exactly that many (appends racing in stay for next step)
            n_a, n_b = len(self._pending_adds), len(self._pending_aborts)
            adds = self._pending_adds[:n_a]
            aborts = self._pending_aborts[:n_b]
            del self._pending_adds[:n_a]
            del self._pending_aborts[:n_b]
        for req in adds:
            self.scheduler.add_request(req)
        for rid in aborts:
            self.scheduler.abort_request(rid)

    # -- forward passes ------------------------------------------------------------------

    def _build_prefill_meta(self, chunks: List[PrefillChunk]) -> Tuple[ForwardMeta, torch.Tensor]:
        # vectorized host build: a 16k-token chunk used to cost ~10 ms of
        # per-token Python loops per step — at 16 chunked-prefill steps that
        # was a visible TTFT slice
        bs = self.args.block_size
        pos_parts, slot_parts, input_ids = [], [], []
        block_tables, seq_lens, query_lens, logits_idx = [], [], [], []
        t = 0
        for c in chunks:
            state = self.cache_manager.get(c.req.rid)
            pos = torch.arange(c.start, c.start + c.num_tokens, dtype=torch.int64)
            bt = torch.tensor(state.block_table, dtype=torch.int64)
            slot_parts.append(bt[pos // bs] * bs + pos % bs)
            pos_parts.append(pos)
            input_ids.extend(c.req.prompt_token_ids[c.start : c.start + c.num_tokens])
            block_tables.append(state.block_table)
            seq_lens.append(c.start + c.num_tokens)
            query_lens.append(c.num_tokens)
            t += c.num_tokens
            logits_idx.append(t - 1)
        dev = self.device
        meta = ForwardMeta(
            is_prefill=True,
            positions=torch.cat(pos_parts).to(device=dev, dtype=torch.int32),
            slot_mapping=torch.cat(slot_parts).to(dev),
            block_tables=build_block_table_tensor(block_tables, dev),
            seq_lens=torch.tensor(seq_lens, dtype=torch.int32, device=dev),
            query_lens=torch.tensor(query_lens, dtype=torch.int32, device=dev),
            kv_cache=None if self.is_mla else self.kv_cache,
            mla_cache=self.kv_cache if self.is_mla else None,
            linear_cache=self.linear_cache,
            linear_slots=torch.tensor(
                [self.cache_manager.get(c.req.rid).linear_slot or 0 for c in chunks],
                dtype=torch.int64, device=dev,
            ) if self.linear_cache is not None else None,
            logits_indices=torch.tensor(logits_idx, dtype=torch.int64, device=dev),
            max_seq_len=max(seq_lens),
        )
        ids = torch.tensor(input_ids, dtype=torch.long, device=dev)
        return meta, ids

    def _build_decode_meta(self, reqs: List[InitialRequest]) -> Tuple[ForwardMeta, torch.Tensor]:
        positions, input_ids, slot_mapping, block_tables, seq_lens = [], [], [], [], []
        for r in reqs:
            state = self.cache_manager.get(r.rid)
            pos = r.total_len - 1  # position of the token being fed in
            positions.append(pos)
            input_ids.append(r.output_token_ids[-1])
            slot_mapping.extend(
                slot_mapping_for_positions(state.block_table, pos, 1, self.args.block_size)
            )
            block_tables.append(state.block_table)
            seq_lens.append(r.total_len)
        dev = self.device
        meta = ForwardMeta(
            is_prefill=False,
            positions=torch.tensor(positions, dtype=torch.int32, device=dev),
            slot_mapping=torch.tensor(slot_mapping, dtype=torch.int64, device=dev),
            block_tables=build_block_table_tensor(block_tables, dev),
            seq_lens=torch.tensor(seq_lens, dtype=torch.int32, device=dev),
            kv_cache=None if self.is_mla else self.kv_cache,
            mla_cache=self.kv_cache if self.is_mla else None,
            linear_cache=self.linear_cache,
            linear_slots=torch.tensor(
                [self.cache_manager.get(r.rid).linear_slot or 0 for r in reqs],
                dtype=torch.int64, device=dev,
            ) if self.linear_cache is not None else None,
            logits_indices=torch.arange(len(reqs), dtype=torch.int64, device=dev),
            max_seq_len=max(seq_lens),
        )
        ids = torch.tensor(input_ids, dtype=torch.long, device=dev)
        return meta, ids

    def _pipeline_forward(
        self, meta: ForwardMeta, input_ids: torch.Tensor, need_logits_at: torch.Tensor
    ) -> Optional[torch.Tensor]:
        """Run the shard; move hidden states along the pipeline. Returns logits
        on the last stage, None elsewhere."""
        comm = self.comm
        T = meta.num_tokens
        h = self.cfg.hidden_size
        if comm.is_first_stage:
            hidden = self.model.embed(input_ids).to(self.args.dtype)
        else:
            hidden = comm.pp_recv((T, h), self.args.dtype, comm.pp_rank - 1)
        hidden = self.model(hidden, meta)
        if not comm.is_last_stage:
            comm.pp_send(hidden, comm.pp_rank + 1)
            return None
        if need_logits_at.numel() == 0:
            return None
        return self.model.compute_logits(hidden[need_logits_at])

    def _run_prefill(self, chunks: List[PrefillChunk]):
        samp_reqs = [c.req for c in chunks if c.is_last_chunk]
        mb = max(1, self.args.micro_batches) if self.comm.pp_size > 1 else 1
        groups: List[List[PrefillChunk]] = (
            [list(g) for g in _split(chunks, mb)] if mb > 1 and len(chunks) >= mb
            else [chunks]
        )
        logits_parts: List[torch.Tensor] = []
        with torch.inference_mode():
            if len(groups) == 1:
                logits = self._prefill_group(chunks)
                return logits, samp_reqs
            # micro-batch pipelining: stage i computes group g while its
            # hidden states for group g-1 are in flight to stage i+1 (same
            # pre-posted-irecv + staged-isend pattern as _run_decode)
            comm = self.comm
            h = self.cfg.hidden_size
            recvs: List[tuple] = []
            if not comm.is_first_stage:
                for g in groups:
                    T = sum(c.num_tokens for c in g)
                    recvs.append(
                        comm.pp_irecv((T, h), self.args.dtype, comm.pp_rank - 1)
                    )
            for i, g in enumerate(groups):
                hidden_in = None
                if not comm.is_first_stage:
                    buf, work = recvs[i]
                    work.wait()
                    hidden_in = buf
                out = self._prefill_group(g, hidden_in=hidden_in,
                                          send_slot=i if not comm.is_last_stage
                                          else None)
                if comm.is_last_stage and out is not None:
                    logits_parts.append(out)
        if self.comm.is_last_stage and logits_parts:
            return torch.cat(logits_parts, dim=0), samp_reqs
        return None, samp_reqs

    def _prefill_group(self, chunks: List[PrefillChunk],
                       hidden_in: Optional[torch.Tensor] = None,
                       send_slot: Optional[int] = None):
        """Forward one prefill (micro-)batch through the local shard. When
        hidden_in/send_slot are given the PP transport is async (pipelined);
        otherwise the blocking single-group path runs."""
        meta, input_ids = self._build_prefill_meta(chunks)
        keep = torch.tensor(
            [i for i, c in enumerate(chunks) if c.is_last_chunk], dtype=torch.int64
        )
        need = meta.logits_indices.cpu()[keep].to(self.device) if keep.numel() else \
            torch.empty(0, dtype=torch.int64, device=self.device)
        comm = self.comm
        if hidden_in is None and send_slot is None:
            return self._pipeline_forward(meta, input_ids, need)
        if comm.is_first_stage:
            hidden = self.model.embed(input_ids).to(self.args.dtype)
        else:
            hidden = hidden_in
        hidden = self.model(hidden, meta)
        if not comm.is_last_stage:
            comm.pp_send_async(hidden, comm.pp_rank + 1, slot=1000 + (send_slot or 0))
            return None
        if need.numel() == 0:
            return None
        return self.model.compute_logits(hidden[need])

    def _decode_fwd(
        self, reqs: List[InitialRequest], hidden_in: Optional[torch.Tensor]
    ) -> torch.Tensor:
        """One decode forward for one (micro-)batch — compute only, no PP
        comm. Returns hidden states (intermediate stages) or logits (last
        stage); graph-captured on GPU."""
        if self.graph_runner is None:
            meta, ids = self._build_decode_meta(reqs)
            if self.comm.is_first_stage:
                hidden = self.model.embed(ids).to(self.args.dtype)
            else:
                hidden = hidden_in
            hidden = self.model(hidden, meta)
            if self.comm.is_last_stage:
                return self.model.compute_logits(hidden[meta.logits_indices])
            return hidden
        input_ids, positions, slots, btabs, seq_lens = [], [], [], [], []
        for r in reqs:
            state = self.cache_manager.get(r.rid)
            pos = r.total_len - 1
            positions.append(pos)
            input_ids.append(r.output_token_ids[-1])
            slots.append(
                state.block_table[pos // self.args.block_size] * self.args.block_size
                + pos % self.args.block_size
            )
            btabs.append(state.block_table)
            seq_lens.append(r.total_len)
        lin_slots = None
        if self.linear_cache is not None:
            lin_slots = [
                self.cache_manager.get(r.rid).linear_slot or 0 for r in reqs
            ]
        return self.graph_runner.run(input_ids, positions, slots, btabs, seq_lens,
                                     hidden_in=hidden_in,
                                     rids=[r.rid for r in reqs],
                                     linear_slots=lin_slots)

    def _decode_one(self, reqs: List[InitialRequest]) -> Optional[torch.Tensor]:
        """Single-micro-batch decode with blocking PP transport."""
        comm = self.comm
        hidden_in = None
        if not comm.is_first_stage:
            hidden_in = comm.pp_recv(
                (len(reqs), self.cfg.hidden_size), self.args.dtype, comm.pp_rank - 1
            )
        out = self._decode_fwd(reqs, hidden_in)
        if not comm.is_last_stage:
            comm.pp_send(out, comm.pp_rank + 1)
            return None
        return out

    def _run_decode(self, reqs: List[InitialRequest]) -> Optional[torch.Tensor]:
        mb = max(1, self.args.micro_batches) if self.comm.pp_size > 1 else 1
        with torch.inference_mode():
            if mb == 1 or len(reqs) < mb:
                return self._decode_one(reqs)
            # Micro-batch pipelining with comm/compute overlap: all receives
            # are pre-posted (they land in issue order per RCCL peer pair),
            # and each group's output is staged + isent on the comm stream so
            # this stage starts group i+1's forward while group i's hidden
            # states are still in flight over xGMI.
            groups: List[List[InitialRequest]] = [list(x) for x in _split(reqs, mb)]
            comm = self.comm
            h = self.cfg.hidden_size
            recvs: List[tuple] = []
            if not comm.is_first_stage:
                for g in groups:
                    recvs.append(
                        comm.pp_irecv((len(g), h), self.args.dtype, comm.pp_rank - 1)
                    )
            logits_parts = []
            for i, g in enumerate(groups):
                hidden_in = None
                if not comm.is_first_stage:
                    buf, work = recvs[i]
                    work.wait()  # stream-ordered on RCCL
                    hidden_in = buf
                out = self._decode_fwd(g, hidden_in)
                if not comm.is_last_stage:
                    comm.pp_send_async(out, comm.pp_rank + 1, slot=i)
                else:
                    logits_parts.append(out.clone())  # graph output buffer is reused
            if comm.is_last_stage:
                return torch.cat(logits_parts, dim=0)
            return None

    # -- sampling + commit ------------------------------------------------------------------

    def _sample_and_broadcast(
        self, logits_parts: List[torch.Tensor], sample_reqs: List[InitialRequest]
    ) -> List[tuple]:
        """Returns [(token_id, logprob-or-None)] for each sampled request.

        PP>1: the last stage samples on-device and the token ids travel as ONE
        int64 device-tensor RCCL broadcast (plus a float tensor only when a
        request asked for logprobs) — no pickled host objects on the decode
        critical path. Every rank's replicated scheduler state agrees on the
        request order and count, so shapes need no negotiation."""
        comm = self.comm
        if comm.pp_size == 1:
            logits = torch.cat(logits_parts, dim=0)
            return self.sampler.sample_with_logprobs(logits, sample_reqs)
        B = len(sample_reqs)
        need_lp = any(r.sampling_params.logprobs for r in sample_reqs)
        if self._tok_bcast_buf is None or self._tok_bcast_buf.numel() < B:
            cap = max(B, self.args.max_batch_size)
            self._tok_bcast_buf = torch.zeros(
                cap, dtype=torch.int64, device=self.device
            )
            self._lp_bcast_buf = torch.zeros(
                cap, dtype=torch.float32, device=self.device
            )
        if comm.is_last_stage and comm.tp_rank == 0:
            logits = torch.cat(logits_parts, dim=0)
            tokens_dev, lp_dev = self.sampler.sample_device(
                logits, sample_reqs, want_logprobs=need_lp
            )
            self._tok_bcast_buf[:B].copy_(tokens_dev)
            if need_lp:
                self._lp_bcast_buf[:B].copy_(lp_dev)
        src = comm.stage_rank(comm.pp_size - 1) - comm.tp_rank  # tp_rank 0 of last stage
        dist.broadcast(self._tok_bcast_buf[:B], src=src)
        if need_lp:
            dist.broadcast(self._lp_bcast_buf[:B], src=src)
        toks = self._tok_bcast_buf[:B].tolist()
        if not need_lp:
            return [(t, None) for t in toks]
        lps = self._lp_bcast_buf[:B].tolist()
        return [
            (t, lps[i] if sample_reqs[i].sampling_params.logprobs else None)
            for i, t in enumerate(toks)
        ]

    # -- weight refit (runtime weight update, reference §3.5) ---------------------------------

    def update_weights_from_disk(self, model_path: str) -> int:
        """Reload this shard's weights from a (new) checkpoint directory and
        invalidate the prefix cache (cached KV was computed with old weights).
        Reference analogue: model_runner.update_weights_from_disk
        (sglang/model_runner.py:406-422)."""
        from .shard_loader import load_shard_weights

        was_gpu = self.device.type == "cuda"
        model_cpu = self.model.to("cpu") if was_gpu else self.model
        n = load_shard_weights(model_cpu, model_path)
        if hasattr(model_cpu, "finalize_weights"):
            model_cpu.finalize_weights()
        self.model = model_cpu.to(device=self.device, dtype=self.args.dtype)
        self.model.rope_cache = self.model.rope_cache.float()
        if self.graph_runner is not None:
            self.graph_runner.model = self.model
            # weights moved: captured graphs reference stale parameter storage
            self.graph_runner._graphs.clear()
            self.graph_runner._outputs.clear()
        self.cache_manager.reset_prefix_cache()
        logger.info("weight refit: reloaded %d tensors from %s", n, model_path)
        return n

    def update_weights_from_tensors(self, named_tensors) -> int:
        """In-place update from (name, tensor) pairs (RL weight push)."""
        n = 0
        for name, t in named_tensors:
            if self.model.load_hf_weight(name, t.cpu()):
                n += 1
        if hasattr(self.model, "finalize_weights"):
            self.model.finalize_weights()
        self.cache_manager.reset_prefix_cache()
        return n

    # -- convenience: synchronous generation (tests, chat CLI) --------------------------------

    def generate(
        self,
        prompts: List[List[int]],
        sampling_params: Optional[List[SamplingParams]] = None,
        max_steps: int = 100000,
    ) -> Dict[str, List[int]]:
        rids = []
        for i, p in enumerate(prompts):
            sp = sampling_params[i] if sampling_params else SamplingParams()
            rids.append(self.submit(p, sp))
        done: Dict[str, List[int]] = {}
        outputs: Dict[str, List[int]] = {rid: [] for rid in rids}
        for _ in range(max_steps):
            for out in self.step():
                if out.rid in outputs:
                    if out.token_id >= 0:
                        outputs[out.rid].append(out.token_id)
                    if out.finished:
                        done[out.rid] = outputs[out.rid]
            # check AFTER the step: step() syncs ingress across ranks, so the
            # loop exits on the same iteration on every pipeline stage
            if not self.has_work or len(done) == len(rids):
                break
        return {rid: outputs[rid] for rid in rids}


def _split(xs, n):
    k, m = divmod(len(xs), n)
    out, i = [], 0
    for j in range(n):
        size = k + (1 if j < m else 0)
        if size:
            out.append(xs[i : i + size])
        i += size
    return out
