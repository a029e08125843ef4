"""hipGraph-captured decode forward.

No reference analogue: the reference delegates decode batching to SGLang/vLLM
(their CUDA-graph runners, sglang/model_runner.py); this is the MI355X-native
equivalent designed for this engine's SPMD stage loop.

The decode step of a 32-layer shard launches ~350 kernels; at ~10-20 us of
launch+gap each that dominates the ~3 ms of real work. This runner captures the
whole per-stage decode forward (embedding -> layers -> final norm -> lm_head)
into a hipGraph per batch-size bucket and replays it with inputs written into
persistent device buffers. Sampling and PP send/recv stay eager (outside the
graph), so the same runner serves every pipeline stage.

Capture-shape invariants:
- batch padded up to the bucket; pad rows get seq_len=1, slot -1 (no KV write),
  block-table row 0 (reads one garbage token — confined to the pad row).
- block_tables buffer is [max_batch, max_blocks(max_model_len)]; real tables
  are copied into the leading columns.
- the attention partition count is a function of meta.max_seq_len only (see
  bindings.cpp), so kernel grids are capture-stable per (batch, ctx) bucket.
- graphs are additionally keyed by a context-length bucket (1k, 2k, ...,
  max_model_len): short contexts replay a graph whose attention grid has few
  flash-decoding partitions instead of carrying max_model_len's empty ones
  (~0.5 us per 1k empty workgroups per call adds up over 32-61 layers).
"""

from __future__ import annotations

import time
from typing import Dict, List, Optional, Tuple

import torch

from ..models.forward_meta import ForwardMeta
from ..utils.logging_config import get_logger

logger = get_logger("server.graph_runner")

DEFAULT_BUCKETS = [1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128, 160, 192, 224, 256,
                   384, 512, 768]


class DecodeGraphRunner:
    def __init__(
        self,
        model,
        kv_cache,
        device: torch.device,
        dtype: torch.dtype,
        max_batch: int,
        max_model_len: int,
        block_size: int,
        hidden_size: int,
        is_first_stage: bool,
        is_last_stage: bool,
        buckets: Optional[List[int]] = None,
        is_mla: bool = False,
        linear_cache=None,
        linear_scratch_slot: int = 0,
    ):
        self.is_mla = is_mla
        # hybrid stacks (deltanet/lightning): per-request state slots; pad
        # rows write into a dedicated scratch slot so capture/warmup replays
        # never corrupt live state
        self.linear_cache = linear_cache
        self.linear_scratch_slot = linear_scratch_slot
        self.model = model
        self.kv_cache = kv_cache
        self.device = device
        self.dtype = dtype
        self.max_model_len = max_model_len
        self.block_size = block_size
        self.is_first = is_first_stage
        self.is_last = is_last_stage
        self.max_blocks = (max_model_len + block_size - 1) // block_size
        self.buckets = sorted(b for b in (buckets or DEFAULT_BUCKETS) if b <= max_batch)
        if not self.buckets or self.buckets[-1] < max_batch:
            self.buckets.append(max_batch)
        B = self.buckets[-1]

        dev = device
        self.input_ids = torch.zeros(B, dtype=torch.long, device=dev)
        self.hidden_in = torch.zeros(B, hidden_size, dtype=dtype, device=dev)
        self.positions = torch.zeros(B, dtype=torch.int32, device=dev)
        self.slot_mapping = torch.full((B,), -1, dtype=torch.int64, device=dev)
        self.block_tables = torch.zeros(B, self.max_blocks, dtype=torch.int32, device=dev)
        self.seq_lens = torch.ones(B, dtype=torch.int32, device=dev)
        # pinned host staging (one copy per step)
        self.h_input_ids = torch.zeros(B, dtype=torch.long, pin_memory=True)
        self.h_positions = torch.zeros(B, dtype=torch.int32, pin_memory=True)
        self.h_slot_mapping = torch.full((B,), -1, dtype=torch.int64, pin_memory=True)
        self.h_block_tables = torch.zeros(B, self.max_blocks, dtype=torch.int32,
                                          pin_memory=True)
        self.h_seq_lens = torch.ones(B, dtype=torch.int32, pin_memory=True)
        if linear_cache is not None:
            self.linear_slots = torch.full((B,), linear_scratch_slot,
                                           dtype=torch.int64, device=dev)
            self.h_linear_slots = torch.full((B,), linear_scratch_slot,
                                             dtype=torch.int64, pin_memory=True)

        self.ctx_buckets: List[int] = []
        c = 1024
        while c < max_model_len:
            self.ctx_buckets.append(c)
            c *= 2
        self.ctx_buckets.append(max_model_len)

        self._graphs: Dict[Tuple[int, int], torch.cuda.CUDAGraph] = {}
        self._outputs: Dict[Tuple[int, int], torch.Tensor] = {}
        self._pool = None
        self._warmed: set = set()
        # block-table row cache: rows only change when a request crosses a
        # block boundary or the batch composition changes
        self._cached_rids: List[str] = []
        self._cached_btlens: List[int] = []

    def _meta(self, bucket: int, ctx_bucket: Optional[int] = None) -> ForwardMeta:
        return ForwardMeta(
            is_prefill=False,
            positions=self.positions[:bucket],
            slot_mapping=self.slot_mapping[:bucket],
            block_tables=self.block_tables[:bucket],
            seq_lens=self.seq_lens[:bucket],
            kv_cache=None if self.is_mla else self.kv_cache,
            mla_cache=self.kv_cache if self.is_mla else None,
            logits_indices=None,
            max_seq_len=ctx_bucket or self.max_model_len,
            linear_cache=self.linear_cache,
            linear_slots=self.linear_slots[:bucket]
            if self.linear_cache is not None else None,
        )

    def _forward(self, bucket: int, ctx_bucket: Optional[int] = None) -> torch.Tensor:
        meta = self._meta(bucket, ctx_bucket)
        if self.is_first:
            hidden = self.model.embed(self.input_ids[:bucket]).to(self.dtype)
        else:
            hidden = self.hidden_in[:bucket]
        hidden = self.model(hidden, meta)
        if self.is_last:
            return self.model.compute_logits(hidden)
        return hidden

    def _capture(self, bucket: int, ctx_bucket: int) -> None:
        logger.info("capturing decode graph for batch bucket %d ctx %d",
                    bucket, ctx_bucket)
        # neutralize buffers: warmup/capture must not write into the live KV
        # cache (slot -1 = skip) or read past block-table row 0
        self.input_ids.zero_()
        self.positions.zero_()
        self.slot_mapping.fill_(-1)
        self.block_tables.zero_()
        self.seq_lens.fill_(1)
        if self.linear_cache is not None:
            self.linear_slots.fill_(self.linear_scratch_slot)
        torch.cuda.synchronize()
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            # warmup (caching allocator, rocBLAS heuristics) — per batch
            # bucket only: GEMM shapes don't depend on the ctx bucket
            n_warm = 0 if bucket in self._warmed else 2
            for _ in range(n_warm):
                self._forward(bucket, ctx_bucket)
        self._warmed.add(bucket)
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph, pool=self._pool):
            out = self._forward(bucket, ctx_bucket)
        if self._pool is None:
            self._pool = graph.pool()
        self._graphs[(bucket, ctx_bucket)] = graph
        self._outputs[(bucket, ctx_bucket)] = out

    def capture_all(self) -> None:
        """Pre-capture every (batch bucket, ctx bucket) graph so serving never
        stalls on an on-demand capture (a capture costs seconds; during a
        request burst those stalls compound into tens of seconds of TTFT).
        Large batches first: their allocations establish the memory pool's
        high-water mark so later captures reuse it."""
        t0 = time.monotonic()
        for b in sorted(self.buckets, reverse=True):
            for c in self.ctx_buckets:
                if (b, c) not in self._graphs:
                    self._capture(b, c)
        logger.info("pre-captured %d decode graphs in %.1fs",
                    len(self._graphs), time.monotonic() - t0)

    def bucket_for(self, batch: int) -> int:
        for b in self.buckets:
            if b >= batch:
                return b
        return self.buckets[-1]

    def ctx_bucket_for(self, max_ctx: int) -> int:
        for c in self.ctx_buckets:
            if c >= max_ctx:
                return c
        return self.ctx_buckets[-1]

    def run(
        self,
        input_ids: List[int],
        positions: List[int],
        slot_mapping: List[int],
        block_tables: List[List[int]],
        seq_lens: List[int],
        hidden_in: Optional[torch.Tensor] = None,
        rids: Optional[List[str]] = None,
        linear_slots: Optional[List[int]] = None,
        input_ids_dev: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        """Returns logits[:B] (last stage) or hidden[:B] (other stages)."""
        B = len(seq_lens)
        bucket = self.bucket_for(B)
        ctx_bucket = self.ctx_bucket_for(max(seq_lens))
        key = (bucket, ctx_bucket)
        if key not in self._graphs:
            self._capture(bucket, ctx_bucket)
            self._cached_rids = []

        self.h_input_ids[:B] = torch.tensor(input_ids, dtype=torch.long)
        self.h_positions[:B] = torch.tensor(positions, dtype=torch.int32)
        self.h_slot_mapping[:B] = torch.tensor(slot_mapping, dtype=torch.int64)
        self.h_slot_mapping[B:bucket] = -1
        self.h_seq_lens[:B] = torch.tensor(seq_lens, dtype=torch.int32)
        self.h_seq_lens[B:bucket] = 1

        btlens = [len(bt) for bt in block_tables]
        same_batch = rids is not None and rids == self._cached_rids
        if same_batch:
            # steady-state decode: copy only rows whose table grew
            for i, bt in enumerate(block_tables):
                if btlens[i] != self._cached_btlens[i]:
                    self.h_block_tables[i, : btlens[i]] = torch.tensor(
                        bt, dtype=torch.int32
                    )
                    self.block_tables[i, : btlens[i]].copy_(
                        self.h_block_tables[i, : btlens[i]], non_blocking=True
                    )
        else:
            self.h_block_tables[:B].zero_()
            for i, bt in enumerate(block_tables):
                self.h_block_tables[i, : btlens[i]] = torch.tensor(bt, dtype=torch.int32)
            self.block_tables[:bucket].copy_(
                self.h_block_tables[:bucket], non_blocking=True
            )
        self._cached_rids = list(rids) if rids is not None else []
        self._cached_btlens = btlens

        if input_ids_dev is None:
            self.input_ids[:bucket].copy_(self.h_input_ids[:bucket],
                                          non_blocking=True)
        else:
            # async decode: previous step's sampled tokens feed this step's
            # input ids without a host round-trip (device-to-device, ordered
            # after the sampler kernels on the same stream)
            self.input_ids[:B].copy_(input_ids_dev.view(-1)[:B])
            if bucket > B:
                self.input_ids[B:bucket].copy_(self.h_input_ids[B:bucket],
                                               non_blocking=True)
        self.positions[:bucket].copy_(self.h_positions[:bucket], non_blocking=True)
        self.slot_mapping[:bucket].copy_(self.h_slot_mapping[:bucket], non_blocking=True)
        self.seq_lens[:bucket].copy_(self.h_seq_lens[:bucket], non_blocking=True)
        if self.linear_cache is not None:
            self.h_linear_slots[:B] = torch.tensor(
                linear_slots or [self.linear_scratch_slot] * B, dtype=torch.int64
            )
            self.h_linear_slots[B:bucket] = self.linear_scratch_slot
            self.linear_slots[:bucket].copy_(self.h_linear_slots[:bucket],
                                             non_blocking=True)
        if hidden_in is not None:
            self.hidden_in[:B].copy_(hidden_in)

        self._graphs[key].replay()
        return self._outputs[key][:B]
