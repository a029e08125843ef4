"""Distributed communication context: RCCL over xGMI (single host), gloo for
CPU tests.

MI355X-first design: one process per GPU; the world is laid out
as pp_size x tp_size with rank = pp_rank * tp_size + tp_rank. Pipeline
hidden-state transport is torch.distributed send/recv (RCCL point-to-point over
xGMI; ~153 GB/s per link); the async variants (pp_irecv/pp_send_async) ride the
process group's internal comm stream so micro-batch sends overlap the next
group's compute. TP all-reduce uses the per-stage group.
Replaces the reference's NCCL-inside-SGLang init (sglang/model_runner.py:97-218)
and its Lattica hidden-state RPC for the in-host path.
"""

from __future__ import annotations

import datetime
import os
from dataclasses import dataclass
from typing import List, Optional

import torch
import torch.distributed as dist

from ..utils.logging_config import get_logger

logger = get_logger("parallel.comm")

_CTX: Optional["CommContext"] = None


@dataclass
class CommContext:
    world_size: int
    rank: int
    pp_size: int
    tp_size: int
    pp_rank: int
    tp_rank: int
    device: torch.device
    tp_group: Optional[dist.ProcessGroup] = None
    pp_group: Optional[dist.ProcessGroup] = None  # group over same-tp-rank stages

    @property
    def is_first_stage(self) -> bool:
        return self.pp_rank == 0

    @property
    def is_last_stage(self) -> bool:
        return self.pp_rank == self.pp_size - 1

    def stage_rank(self, pp_rank: int) -> int:
        """Global rank of the given pipeline stage at our tp_rank."""
        return pp_rank * self.tp_size + self.tp_rank

    @property
    def next_stage_rank(self) -> int:
        return self.stage_rank((self.pp_rank + 1) % self.pp_size)

    @property
    def prev_stage_rank(self) -> int:
        return self.stage_rank((self.pp_rank - 1) % self.pp_size)

    # -- TP collectives -----------------------------------------------------

    def tp_all_reduce(self, t: torch.Tensor) -> torch.Tensor:
        if self.tp_size > 1:
            dist.all_reduce(t, group=self.tp_group)
        return t

    def tp_all_gather(self, t: torch.Tensor, dim: int = -1) -> torch.Tensor:
        if self.tp_size == 1:
            return t
        parts = [torch.empty_like(t) for _ in range(self.tp_size)]
        dist.all_gather(parts, t, group=self.tp_group)
        return torch.cat(parts, dim=dim)

    # -- PP point-to-point ---------------------------------------------------

    def pp_send(self, t: torch.Tensor, dst_pp_rank: int) -> None:
        dist.send(t.contiguous(), self.stage_rank(dst_pp_rank))

    def pp_recv(self, shape, dtype: torch.dtype, src_pp_rank: int) -> torch.Tensor:
        t = torch.empty(shape, dtype=dtype, device=self.device)
        dist.recv(t, self.stage_rank(src_pp_rank))
        return t

    # -- PP async point-to-point (micro-batch pipelining) ---------------------
    #
    # With the RCCL backend, isend/irecv enqueue on the process group's
    # internal comm stream (ordered after the caller's current stream), and
    # Work.wait() inserts a stream dependency rather than blocking the host —
    # so a stage can compute micro-batch i while micro-batch i-1 is in flight
    # over xGMI and micro-batch i+1's recv is already posted.

    def pp_irecv(self, shape, dtype: torch.dtype, src_pp_rank: int):
        """Post a receive; returns (buffer, work). Call work.wait() before
        reading the buffer (stream-ordered on RCCL, host-blocking on gloo)."""
        t = torch.empty(shape, dtype=dtype, device=self.device)
        work = dist.irecv(t, self.stage_rank(src_pp_rank))
        return t, work

    def pp_send_async(self, t: torch.Tensor, dst_pp_rank: int, slot: int = 0) -> None:
        """Copy `t` into a per-slot staging buffer and isend it. The caller's
        stream only pays for the staging copy; the send itself overlaps the
        next micro-batch's compute. `slot` keys the staging buffer (use the
        micro-batch index) so a graph-output buffer can be reused immediately
        after this returns."""
        if not hasattr(self, "_send_slots"):
            self._send_slots = {}
        t = t.contiguous()
        key = (dst_pp_rank, slot)
        prev = self._send_slots.get(key)
        buf = None
        if prev is not None:
            pwork, pbuf = prev
            if pwork is not None:
                pwork.wait()  # slot reuse: previous send must drain first
            if pbuf.shape == t.shape and pbuf.dtype == t.dtype:
                buf = pbuf
        if buf is None:
            buf = torch.empty_like(t)
        buf.copy_(t)
        work = dist.isend(buf, self.stage_rank(dst_pp_rank))
        self._send_slots[key] = (work, buf)

    def pp_flush_sends(self) -> None:
        """Wait (stream-ordered) for all outstanding async sends."""
        for key, (work, _buf) in list(getattr(self, "_send_slots", {}).items()):
            if work is not None:
                work.wait()
                w, b = self._send_slots[key]
                self._send_slots[key] = (None, b)

    def barrier(self) -> None:
        if dist.is_initialized():
            dist.barrier()


def init_distributed(
    pp_size: int = 1,
    tp_size: int = 1,
    rank: Optional[int] = None,
    world_size: Optional[int] = None,
    backend: Optional[str] = None,
    device: Optional[torch.device] = None,
    timeout_s: float = 600.0,
) -> CommContext:
    """Initialize torch.distributed (backend 'nccl' IS RCCL on ROCm) and build
    the TP/PP groups. Reads RANK/WORLD_SIZE/MASTER_* from env when launched via
    torch.distributed.run; single-process worlds skip init entirely."""
    global _CTX
    rank = rank if rank is not None else int(os.environ.get("RANK", "0"))
    world_size = (
        world_size if world_size is not None else int(os.environ.get("WORLD_SIZE", "1"))
    )
    assert world_size == pp_size * tp_size, (
        f"world_size {world_size} != pp_size {pp_size} * tp_size {tp_size}"
    )
    use_gpu = torch.cuda.is_available()
    if device is None:
        if use_gpu:
            local_rank = int(os.environ.get("LOCAL_RANK", rank % max(1, torch.cuda.device_count())))
            torch.cuda.set_device(local_rank)
            device = torch.device("cuda", local_rank)
        else:
            device = torch.device("cpu")
    if backend is None:
        backend = "nccl" if use_gpu else "gloo"

    if world_size > 1 and not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29511")
        dist.init_process_group(
            backend=backend,
            rank=rank,
            world_size=world_size,
            timeout=datetime.timedelta(seconds=timeout_s),
        )

    pp_rank, tp_rank = divmod(rank, tp_size)
    tp_group = pp_group = None
    if world_size > 1:
        # build all groups on every rank (torch requires global participation)
        for p in range(pp_size):
            ranks = list(range(p * tp_size, (p + 1) * tp_size))
            g = dist.new_group(ranks) if tp_size > 1 else None
            if p == pp_rank:
                tp_group = g
        for t in range(tp_size):
            ranks = [p * tp_size + t for p in range(pp_size)]
            g = dist.new_group(ranks) if pp_size > 1 else None
            if t == tp_rank:
                pp_group = g

    _CTX = CommContext(
        world_size=world_size,
        rank=rank,
        pp_size=pp_size,
        tp_size=tp_size,
        pp_rank=pp_rank,
        tp_rank=tp_rank,
        device=device,
        tp_group=tp_group,
        pp_group=pp_group,
    )
    logger.info(
        "comm init: rank %d/%d pp %d/%d tp %d/%d backend=%s device=%s",
        rank, world_size, pp_rank, pp_size, tp_rank, tp_size, backend, device,
    )
    return _CTX


def get_comm() -> CommContext:
    global _CTX
    if _CTX is None:
        _CTX = CommContext(
            world_size=1, rank=0, pp_size=1, tp_size=1, pp_rank=0, tp_rank=0,
            device=torch.device("cuda", torch.cuda.current_device())
            if torch.cuda.is_available() else torch.device("cpu"),
        )
    return _CTX


def set_comm(ctx: CommContext) -> None:
    global _CTX
    _CTX = ctx


def destroy() -> None:
    global _CTX
    if dist.is_initialized():
        dist.destroy_process_group()
    _CTX = None
