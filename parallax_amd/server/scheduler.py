"""Continuous-batching in-node scheduler.

Capability parity with the reference's src/parallax/server/scheduler.py:42 —
two-phase operation: ``admit_requests`` moves the wait queue into the running
set under KV-budget and batch-size limits (with prefix-cache matching), and
``form_batch`` builds one engine iteration: chunked prefills first, then ready
decodes, bounded by ``max_num_tokens_per_batch`` and ``micro_batch_size``.
EOS/stop/length finish checks live on the request (request.py); the timeout
sweep and release bookkeeping live here. Fresh design.
"""

from __future__ import annotations

import time
from collections import deque
from dataclasses import dataclass, field
from typing import Deque, Dict, List, Optional, Tuple

from ..utils.logging_config import get_logger
from .allocator import OutOfBlocksError
from .cache_manager import CacheManager
from .request import InitialRequest, RequestStatus

logger = get_logger("server.scheduler")


@dataclass
class PrefillChunk:
    req: InitialRequest
    start: int          # first prompt position in this chunk
    num_tokens: int     # chunk length

    @property
    def is_last_chunk(self) -> bool:
        return self.start + self.num_tokens >= self.req.prompt_len


@dataclass
class ScheduledBatch:
    prefill_chunks: List[PrefillChunk] = field(default_factory=list)
    decode_reqs: List[InitialRequest] = field(default_factory=list)

    @property
    def num_tokens(self) -> int:
        return sum(c.num_tokens for c in self.prefill_chunks) + len(self.decode_reqs)

    @property
    def is_empty(self) -> bool:
        return not self.prefill_chunks and not self.decode_reqs


class Scheduler:
    def __init__(
        self,
        cache_manager: CacheManager,
        max_batch_size: int = 128,
        max_num_tokens_per_batch: int = 16384,
        prefill_chunk_size: int = 8192,
        request_timeout_s: float = 600.0,
        eos_token_ids: Optional[List[int]] = None,
        prefill_priority: bool = True,
    ):
        self.cache = cache_manager
        self.max_batch_size = max_batch_size
        self.max_num_tokens_per_batch = max_num_tokens_per_batch
        # page-align the prefill chunk so chunk boundaries land on KV blocks
        bs = cache_manager.block_size
        self.prefill_chunk_size = max(bs, (prefill_chunk_size // bs) * bs)
        self.request_timeout_s = request_timeout_s
        self.eos_token_ids = eos_token_ids or []
        # reference --prefill-priority: prefills before decodes (default) or
        # decodes first (lower TPOT under arrival bursts at the cost of TTFT)
        self.prefill_priority = prefill_priority

        self.wait_queue: Deque[InitialRequest] = deque()
        self.running: Dict[str, InitialRequest] = {}
        self.finished_reqs: List[InitialRequest] = []
        # aborts that hit the wait queue never pass through the running-set
        # sweeps, so the engine reports them to consumers from this list
        self.aborted_waiting: List[InitialRequest] = []

    # -- ingress ---------------------------------------------------------------

    def add_request(self, req: InitialRequest) -> None:
        if not req.eos_token_ids:
            req.eos_token_ids = list(self.eos_token_ids)
        self.wait_queue.append(req)

    def abort_request(self, rid: str) -> bool:
        req = self.running.get(rid)
        if req is not None:
            req.abort_requested = True
            return True
        for i, r in enumerate(self.wait_queue):
            if r.rid == rid:
                r.abort_requested = True
                r.status = RequestStatus.FINISHED_ABORT
                del self.wait_queue[i]
                self.finished_reqs.append(r)
                self.aborted_waiting.append(r)
                return True
        return False

    # -- phase 1: admission -------------------------------------------------------

    def admit_requests(self) -> int:
        """Wait queue -> running, allocating KV (with prefix match). FCFS."""
        admitted = 0
        while self.wait_queue and len(self.running) < self.max_batch_size:
            req = self.wait_queue[0]
            # Budget: the whole prompt plus one decode slack block.
            need_tokens = req.prompt_len + self.cache.block_size
            if not self.cache.can_allocate(need_tokens, req.prompt_token_ids):
                break
            self.wait_queue.popleft()
            try:
                state = self.cache.allocate_request(
                    req.rid, req.prompt_token_ids, req.sampling_params.max_new_tokens
                )
            except OutOfBlocksError:
                self.wait_queue.appendleft(req)
                break
            req.num_prefilled_tokens = state.num_cached_tokens
            req.status = RequestStatus.PREFILLING
            self.running[req.rid] = req
            admitted += 1
        return admitted

    # -- phase 2: batch formation ---------------------------------------------------

    def form_batch(self) -> ScheduledBatch:
        batch = ScheduledBatch()
        token_budget = self.max_num_tokens_per_batch
        if not self.prefill_priority:
            token_budget -= self._form_decodes(batch, token_budget)

        # prefills (possibly chunked)
        for req in self.running.values():
            if token_budget <= 0:
                break
            if req.status is RequestStatus.PREFILLING and not req.is_finished:
                remaining = req.prompt_len - req.num_prefilled_tokens
                if remaining <= 0:
                    continue
                chunk = min(remaining, self.prefill_chunk_size, token_budget)
                if chunk < remaining:
                    # keep chunk page-aligned unless it is the final chunk
                    chunk = (chunk // self.cache.block_size) * self.cache.block_size
                    if chunk == 0:
                        continue
                batch.prefill_chunks.append(
                    PrefillChunk(req, req.num_prefilled_tokens, chunk)
                )
                token_budget -= chunk

        if self.prefill_priority:
            self._form_decodes(batch, token_budget)
        return batch

    def _form_decodes(self, batch: ScheduledBatch, token_budget: int) -> int:
        used = 0
        for req in self.running.values():
            if used >= token_budget \
                    or len(batch.decode_reqs) >= self.max_batch_size:
                break
            if req.status is RequestStatus.DECODING and not req.is_finished:
                try:
                    self.cache.append_tokens(req.rid, req.total_len + 1)
                except OutOfBlocksError:
                    # KV exhausted: abort the youngest request (reference aborts
                    # and signals the client on KV OOM, sglang_executor.py:527)
                    logger.warning("KV OOM growing %s; aborting", req.rid)
                    req.abort_requested = True
                    continue
                batch.decode_reqs.append(req)
                used += 1
        return used

    # -- step completion -----------------------------------------------------------

    def complete_prefill_chunk(self, chunk: PrefillChunk) -> None:
        req = chunk.req
        req.num_prefilled_tokens += chunk.num_tokens
        if req.prefill_done:
            self.cache.publish_prefill(req.rid, req.prompt_token_ids)

    def commit_token(self, rid: str, token_id: int) -> Optional[InitialRequest]:
        """Commit a sampled token on the head peer; transitions PREFILLING→DECODING
        and releases the request when finished. Returns the request if finished."""
        req = self.running.get(rid)
        if req is None:
            return None
        req.commit_new_token(token_id)
        if req.status is RequestStatus.PREFILLING:
            req.status = RequestStatus.DECODING
        if req.check_finished():
            self._release(req)
            return req
        return None

    def sweep_timeouts(self) -> List[InitialRequest]:
        """Abort requests that exceeded the per-request timeout."""
        now = time.monotonic()
        timed_out = []
        for req in list(self.running.values()):
            if now - req.arrival_time > self.request_timeout_s:
                req.abort_requested = True
                req.check_finished()
                self._release(req)
                timed_out.append(req)
        for req in timed_out:
            logger.warning("request %s timed out after %.0fs", req.rid, self.request_timeout_s)
        return timed_out

    def sweep_aborted(self) -> List[InitialRequest]:
        out = []
        for req in list(self.running.values()):
            if req.abort_requested and req.check_finished():
                self._release(req)
                out.append(req)
        return out

    def _release(self, req: InitialRequest) -> None:
        self.running.pop(req.rid, None)
        self.free_cache(req)
        self.finished_reqs.append(req)

    def release_keep_cache(self, req: InitialRequest) -> None:
        """Remove from the running set (so the next batch excludes it) but
        keep its KV blocks allocated: the async-pipelined step already in
        flight still WRITES this request's last KV slot — freeing now would
        let a concurrent allocation adopt the block mid-write. The engine
        calls free_cache() after that step's completion event is synced."""
        self.running.pop(req.rid, None)
        self.finished_reqs.append(req)

    def free_cache(self, req: InitialRequest) -> None:
        # Publish only tokens whose KV was actually computed (the reference
        # bounds insertion by context_len, cache_manager.insert_full_blocks_to_cache):
        # the final sampled token is never forwarded so its KV slot is unwritten,
        # and a request aborted mid-prefill has no complete KV at all.
        if req.prefill_done:
            computed = req.num_prefilled_tokens + max(0, req.num_output_tokens - 1)
            self.cache.free_request(req.rid, req.all_token_ids[:computed])
        else:
            self.cache.free_request(req.rid, None)

    # -- introspection ----------------------------------------------------------------

    @property
    def has_work(self) -> bool:
        return bool(self.wait_queue or self.running)

    @property
    def num_running(self) -> int:
        return len(self.running)

    def drain_aborted_waiting(self) -> List[InitialRequest]:
        out, self.aborted_waiting = self.aborted_waiting, []
        return out

    def drain_finished(self) -> List[InitialRequest]:
        out, self.finished_reqs = self.finished_reqs, []
        return out
