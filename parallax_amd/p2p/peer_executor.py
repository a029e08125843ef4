"""Packet-driven per-stage executor for the decentralized (multi-host) path.

Reference analogue: server/executor/base_executor.py run_loop (:634-769) — the
head peer owns full request state and admission; intermediate peers process
IntermediateRequest packets as they arrive; the last peer samples and loops the
token back to the head (routing table wrap-around). Cache state on non-head
peers is length-driven (no radix) and freed on 'release'/'abort' control
packets. Transport is pluggable (loopback for tests, TCP for real hosts).
"""

from __future__ import annotations

import time
from collections import defaultdict
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import torch

from ..models import get_model_class
from ..models.config import ModelConfig
from ..models.forward_meta import ForwardMeta
from ..server.cache_manager import CacheManager
from ..server.kv_cache import (
    KVCacheSpec,
    MLAKVCache,
    PagedKVCache,
    build_block_table_tensor,
    slot_mapping_for_positions,
)
from ..server.request import InitialRequest, IntermediateRequest
from ..server.sampler import Sampler
from ..server.sampling_params import SamplingParams
from ..server.scheduler import Scheduler
from ..utils.logging_config import get_logger
from . import codec
from .transport import Transport

logger = get_logger("p2p.peer_executor")


@dataclass
class PeerOutput:
    rid: str
    token_id: int
    finished: bool
    finish_reason: Optional[str] = None
    logprob: Optional[float] = None


class PeerExecutor:
    def __init__(
        self,
        cfg: ModelConfig,
        start_layer: int,
        end_layer: int,
        peer_id: str,
        transport: Transport,
        device: Optional[torch.device] = None,
        dtype: torch.dtype = torch.float32,
        num_kv_blocks: int = 1024,
        block_size: int = 16,
        random_weights: bool = False,
        model_path: Optional[str] = None,
        seed: int = 0,
        refit_dir: Optional[str] = None,
    ):
        self.cfg = cfg
        self.peer_id = peer_id
        self.transport = transport
        # version dirs for pushed weight refits (None = manifest pushes ignored)
        self.refit_dir = refit_dir
        self.device = device or torch.device("cpu")
        self.dtype = dtype
        self.block_size = block_size
        self.is_head = start_layer == 0
        self.is_last = end_layer == cfg.num_layers

        self.model = get_model_class(cfg.architecture)(cfg, start_layer, end_layer)
        if random_weights:
            self.model.init_random()
        elif model_path:
            from ..server.shard_loader import load_shard_weights

            load_shard_weights(self.model, model_path)
        if hasattr(self.model, "finalize_weights"):
            self.model.finalize_weights()
        self.model = self.model.to(device=self.device, dtype=dtype).eval()
        self.model.rope_cache = self.model.rope_cache.float()

        self.is_mla = cfg.is_mla
        if self.is_mla:
            self.kv_cache = MLAKVCache(
                end_layer - start_layer, cfg.kv_lora_rank, cfg.qk_rope_head_dim,
                block_size, num_kv_blocks, self.device,
                dtype if dtype != torch.float32 else torch.float32,
                index_dim=cfg.index_head_dim if cfg.is_dsa else 0,
            )
        else:
            spec = KVCacheSpec(
                num_layers=end_layer - start_layer, num_kv_heads=cfg.num_kv_heads,
                head_dim=cfg.head_dim, block_size=block_size, dtype=dtype,
                index_dim=cfg.index_head_dim if cfg.is_msa else 0,
            )
            self.kv_cache = PagedKVCache(spec, num_kv_blocks, self.device)
        # hybrid stacks: per-request conv/recurrent state slots
        self.linear_cache = None
        num_linear_slots = 0
        if cfg.has_linear_layers:
            from ..server.kv_cache import LinearStateCache

            local_types = [cfg.layer_type(g) for g in range(start_layer, end_layer)]
            n_linear = sum(1 for t in local_types if t == "linear_attention")
            conv_dim = (
                2 * cfg.linear_num_key_heads * cfg.linear_key_head_dim
                + cfg.linear_num_value_heads * cfg.linear_value_head_dim
            )
            num_linear_slots = 64
            self.linear_cache = LinearStateCache(
                max(1, n_linear),
                conv_state_shape=(conv_dim, cfg.linear_conv_kernel_dim - 1),
                recurrent_state_shape=(
                    cfg.linear_num_value_heads, cfg.linear_key_head_dim,
                    cfg.linear_value_head_dim,
                ),
                num_slots=num_linear_slots + 1,
                device=self.device,
                dtype=dtype if dtype != torch.float32 else torch.float32,
            )
        # non-head peers never see token content: radix prefix match disabled
        # (and linear-state slots do not compose with prefix reuse)
        self.cache_manager = CacheManager(
            block_size, num_kv_blocks,
            enable_prefix_cache=self.is_head and not cfg.has_linear_layers,
            num_linear_slots=num_linear_slots,
        )
        self.scheduler = (
            Scheduler(self.cache_manager, eos_token_ids=cfg.eos_token_ids)
            if self.is_head else None
        )
        self.sampler = Sampler(self.device, seed) if self.is_last else None
        # last-stage sampling context: rid -> (params, prompt_ids, output_ids)
        self._sampling_ctx: Dict[str, Tuple[SamplingParams, List[int], List[int]]] = {}
        # non-head cache position bookkeeping: rid -> tokens cached so far
        self._peer_positions: Dict[str, int] = {}
        self.finished_outputs: List[PeerOutput] = []
        # packets that raced a blocking refit fetch; drained before transport
        self._deferred_packets: List[bytes] = []
        self._pending_refit_manifest: Optional[tuple] = None

    def _recv(self, timeout: float) -> Optional[bytes]:
        if self._deferred_packets:
            return self._deferred_packets.pop(0)
        return self.transport.recv(timeout=timeout)

    # -- head API ------------------------------------------------------------------

    def submit(
        self,
        prompt_token_ids: List[int],
        sampling_params: SamplingParams,
        routing_table: List[str],
        rid: Optional[str] = None,
    ) -> str:
        assert self.is_head
        from ..server.request import new_request_id

        rid = rid or new_request_id()
        if rid in self.scheduler.running \
                or any(r.rid == rid for r in list(self.scheduler.wait_queue)):
            raise ValueError(f"duplicate request id {rid!r}")
        req = InitialRequest(
            rid=rid,
            prompt_token_ids=list(prompt_token_ids),
            sampling_params=sampling_params,
            routing_table=list(routing_table),
        )
        self.scheduler.add_request(req)
        return req.rid

    def abort(self, rid: str) -> None:
        """Client-requested abort (head only): swept on the next step."""
        req = self.scheduler.running.get(rid) if self.scheduler else None
        if req is not None:
            req.abort_requested = True

    @property
    def has_work(self) -> bool:
        if self.is_head:
            return self.scheduler.has_work
        return bool(self._peer_positions)

    def drain_outputs(self) -> List[PeerOutput]:
        out, self.finished_outputs = self.finished_outputs, []
        return out

    # -- the step ----------------------------------------------------------------------

    def step(self, recv_timeout: float = 0.01) -> None:
        if self._pending_refit_manifest is not None and self.refit_dir:
            manifest, publisher = self._pending_refit_manifest
            self._pending_refit_manifest = None
            try:
                self.refit_from_peer(publisher, manifest, self.refit_dir)
            except Exception as e:  # keep serving on a failed refit
                logger.error("pushed refit failed: %s", e)
        if self.is_head:
            self._head_step(recv_timeout)
        else:
            self._peer_step(recv_timeout)

    # -- head ------------------------------------------------------------------------

    def _head_step(self, recv_timeout: float) -> None:
        # 1. drain token packets from the last stage
        while True:
            data = self._recv(recv_timeout)
            if data is None:
                break
            msg = codec.decode(data)
            if msg["kind"].startswith("refit_"):
                self._handle_refit(msg)
            elif msg["kind"] == "token":
                for entry in msg["tokens"]:
                    rid, tok = entry[0], entry[1]
                    lp = entry[2] if len(entry) > 2 else None
                    finished = self.scheduler.commit_token(rid, tok)
                    req = finished or self.scheduler.running.get(rid)
                    if req is not None:
                        self.finished_outputs.append(
                            PeerOutput(rid, tok, finished is not None,
                                       req.status.finish_reason, logprob=lp)
                        )
                    if finished is not None:
                        self._broadcast_control("release", [rid],
                                                finished.routing_table)
            recv_timeout = 0.0  # only block on the first recv

        # 2. timeout sweep: requests stranded by a dead downstream peer
        # terminate with a timeout finish instead of hanging forever
        self._sweep_counter = getattr(self, "_sweep_counter", 0) + 1
        if self._sweep_counter % 64 == 0:
            for req in self.scheduler.sweep_timeouts():
                self.finished_outputs.append(
                    PeerOutput(req.rid, -1, True, req.status.finish_reason)
                )
                self._broadcast_control("release", [req.rid],
                                        req.routing_table)
        # abort sweep: release downstream state and terminate streams
        for req in self.scheduler.sweep_aborted():
            self.finished_outputs.append(
                PeerOutput(req.rid, -1, True, req.status.finish_reason)
            )
            self._broadcast_control("release", [req.rid], req.routing_table)

        # 3. schedule local work
        self.scheduler.admit_requests()
        batch = self.scheduler.form_batch()
        if batch.is_empty:
            return

        if batch.prefill_chunks:
            chunks = batch.prefill_chunks
            meta, input_ids, _ = self._head_prefill_meta(chunks)
            with torch.inference_mode():
                hidden = self.model.embed(input_ids).to(self.dtype)
                hidden = self.model(hidden, meta)
            packets = []
            t = 0
            for c in chunks:
                h = hidden[t : t + c.num_tokens]
                pkt = IntermediateRequest.from_initial(
                    c.req, h, is_prefill=True, position=c.start,
                    num_new_tokens=c.num_tokens,
                )
                pkt.input_ids = c.req.prompt_token_ids[c.start : c.start + c.num_tokens]
                # convention: mid-prompt chunks must not sample downstream
                pkt.next_token_id = None if c.is_last_chunk else -1
                packets.append(pkt)
                t += c.num_tokens
                self.scheduler.complete_prefill_chunk(c)
            self._send_forward(packets)

        if batch.decode_reqs:
            reqs = batch.decode_reqs
            meta, input_ids = self._head_decode_meta(reqs)
            with torch.inference_mode():
                hidden = self.model.embed(input_ids).to(self.dtype)
                hidden = self.model(hidden, meta)
            packets = []
            for i, r in enumerate(reqs):
                pkt = IntermediateRequest.from_initial(
                    r, hidden[i : i + 1], is_prefill=False,
                    position=r.total_len - 1, num_new_tokens=1,
                )
                packets.append(pkt)
            self._send_forward(packets)

    def _head_prefill_meta(self, chunks):
        positions, input_ids, slots, btabs, seq_lens, qlens = [], [], [], [], [], []
        for c in chunks:
            state = self.cache_manager.get(c.req.rid)
            positions.extend(range(c.start, c.start + c.num_tokens))
            input_ids.extend(c.req.prompt_token_ids[c.start : c.start + c.num_tokens])
            slots.extend(slot_mapping_for_positions(
                state.block_table, c.start, c.num_tokens, self.block_size))
            btabs.append(state.block_table)
            seq_lens.append(c.start + c.num_tokens)
            qlens.append(c.num_tokens)
        meta = self._meta(True, positions, slots, btabs, seq_lens, qlens,
                          rids=[c.req.rid for c in chunks])
        return meta, torch.tensor(input_ids, dtype=torch.long, device=self.device), None

    def _head_decode_meta(self, reqs):
        positions, input_ids, slots, btabs, seq_lens = [], [], [], [], []
        for r in reqs:
            state = self.cache_manager.get(r.rid)
            pos = r.total_len - 1
            positions.append(pos)
            input_ids.append(r.output_token_ids[-1])
            slots.extend(slot_mapping_for_positions(
                state.block_table, pos, 1, self.block_size))
            btabs.append(state.block_table)
            seq_lens.append(r.total_len)
        meta = self._meta(False, positions, slots, btabs, seq_lens, None,
                          rids=[r.rid for r in reqs])
        return meta, torch.tensor(input_ids, dtype=torch.long, device=self.device)

    # -- non-head -------------------------------------------------------------------------

    def _peer_step(self, recv_timeout: float) -> None:
        packets: List[IntermediateRequest] = []
        data = self._recv(recv_timeout)
        if data is None:
            return
        while data is not None:
            msg = codec.decode(data)
            if msg["kind"] == "forward":
                packets.extend(msg["reqs"])
            elif msg["kind"].startswith("refit_"):
                self._handle_refit(msg)
            elif msg["kind"] in ("release", "abort"):
                for rid in msg["rids"]:
                    self.cache_manager.free_request(rid)
                    self._peer_positions.pop(rid, None)
                    self._sampling_ctx.pop(rid, None)
            data = self._recv(0.0)
        if not packets:
            return
        prefills = [p for p in packets if p.is_prefill]
        decodes = [p for p in packets if not p.is_prefill]
        if prefills:
            self._process_packets(prefills, is_prefill=True)
        if decodes:
            self._process_packets(decodes, is_prefill=False)

    def _process_packets(self, pkts: List[IntermediateRequest], is_prefill: bool) -> None:
        positions, slots, btabs, seq_lens, qlens = [], [], [], [], []
        hiddens = []
        for p in pkts:
            total = p.current_position + p.num_new_tokens
            if p.rid not in self._peer_positions:
                self.cache_manager.allocate_request(p.rid, [0] * total)
            else:
                self.cache_manager.append_tokens(p.rid, total)
            self._peer_positions[p.rid] = total
            state = self.cache_manager.get(p.rid)
            positions.extend(range(p.current_position, total))
            slots.extend(slot_mapping_for_positions(
                state.block_table, p.current_position, p.num_new_tokens,
                self.block_size))
            btabs.append(state.block_table)
            seq_lens.append(total)
            qlens.append(p.num_new_tokens)
            hiddens.append(p.hidden_states.to(self.device, self.dtype))
            if self.is_last and p.sampling_params is not None:
                ctx = self._sampling_ctx.setdefault(
                    p.rid, (p.sampling_params, [], [])
                )
                if is_prefill and p.input_ids:
                    ctx[1].extend(p.input_ids)
        meta = self._meta(is_prefill, positions, slots, btabs, seq_lens,
                          qlens if is_prefill else None,
                          rids=[p.rid for p in pkts])
        hidden = torch.cat(hiddens, dim=0)
        with torch.inference_mode():
            hidden = self.model(hidden, meta)

        if self.is_last:
            # sample only for final prefill chunks / decodes
            idx, sample_pkts = [], []
            t = 0
            for p, ql in zip(pkts, qlens):
                t += ql
                # a prefill chunk samples only when it completes the prompt —
                # the head marks that by setting num_new_tokens to reach the
                # prompt end; mid-chunks carry no sampling duty. The head only
                # expects a token when this was the final chunk; it encodes
                # that via p.return_logprob?  Convention: head sets
                # p.next_token_id = -1 on chunks that must NOT sample.
                if p.next_token_id == -1:
                    continue
                idx.append(t - 1)
                sample_pkts.append(p)
            if idx:
                logits = self.model.compute_logits(
                    hidden[torch.tensor(idx, dtype=torch.long, device=self.device)]
                )
                fake_reqs = []
                for p in sample_pkts:
                    sp, prompt_ids, out_ids = self._sampling_ctx.get(
                        p.rid, (p.sampling_params or SamplingParams(), [], [])
                    )
                    fr = InitialRequest(rid=p.rid, prompt_token_ids=prompt_ids,
                                        sampling_params=sp)
                    fr.output_token_ids = out_ids
                    fake_reqs.append(fr)
                sampled = self.sampler.sample_with_logprobs(logits, fake_reqs)
                for p, (tok, _) in zip(sample_pkts, sampled):
                    if p.rid in self._sampling_ctx:
                        self._sampling_ctx[p.rid][2].append(tok)
                # token goes to the head = first entry of the routing table
                head = sample_pkts[0].routing_table[0]
                self.transport.send(
                    head, codec.encode_tokens(
                        [(p.rid, int(t), lp)
                         for p, (t, lp) in zip(sample_pkts, sampled)]
                    ),
                )
        else:
            out_pkts = []
            t = 0
            for p, ql in zip(pkts, qlens):
                p.hidden_states = hidden[t : t + ql]
                t += ql
                out_pkts.append(p)
            self._send_forward(out_pkts)

    # -- shared ---------------------------------------------------------------------------

    def _meta(self, is_prefill, positions, slots, btabs, seq_lens, qlens,
              rids=None) -> ForwardMeta:
        dev = self.device
        linear_slots = None
        if self.linear_cache is not None and rids is not None:
            linear_slots = torch.tensor(
                [self.cache_manager.get(r).linear_slot or 0 for r in rids],
                dtype=torch.int64, device=dev,
            )
        return ForwardMeta(
            is_prefill=is_prefill,
            positions=torch.tensor(positions, dtype=torch.int32, device=dev),
            slot_mapping=torch.tensor(slots, dtype=torch.int64, device=dev),
            block_tables=build_block_table_tensor(btabs, dev),
            seq_lens=torch.tensor(seq_lens, dtype=torch.int32, device=dev),
            query_lens=torch.tensor(qlens, dtype=torch.int32, device=dev)
            if qlens else None,
            kv_cache=None if self.is_mla else self.kv_cache,
            mla_cache=self.kv_cache if self.is_mla else None,
            linear_cache=self.linear_cache,
            linear_slots=linear_slots,
            max_seq_len=max(seq_lens),
        )

    def _send_forward(self, pkts: List[IntermediateRequest]) -> None:
        """Group packets by next hop and ship them (reference start_node_sender
        groups by peer, p2p/server.py:687)."""
        by_peer: Dict[str, List[IntermediateRequest]] = defaultdict(list)
        for p in pkts:
            nxt = p.next_hop(self.peer_id)
            if nxt is None or nxt == self.peer_id:
                continue
            by_peer[nxt].append(p)
        for peer, group in by_peer.items():
            try:
                self.transport.send(peer, codec.encode_forward(group))
            except (ConnectionError, OSError, KeyError) as e:
                # dead/unknown next hop: abort the affected requests so the
                # head terminates their streams (reference: batch error ->
                # abort + ERROR to client, sglang_executor.py:505-546) and
                # keep serving everything else
                logger.error("send to %s failed (%s): aborting %d reqs",
                             peer, e, len(group))
                if self.is_head:
                    for pkt in group:
                        self.scheduler.abort_request(pkt.rid)

    # -- weight refit over the transport (reference p2p/server.py:224-338) --------

    def set_refit_publisher(self, publisher) -> None:
        """Serve weight chunks to peers (the trainer/origin side)."""
        self._refit_publisher = publisher

    def _handle_refit(self, msg: dict) -> None:
        from . import refit as refit_mod

        if msg["kind"] == "refit_get":
            pub = getattr(self, "_refit_publisher", None)
            if pub is None:
                logger.warning("refit_get but no publisher attached")
                return
            refit_mod.answer_refit_get(pub, self.transport, msg)
        elif msg["kind"] == "refit_chunk":
            logger.debug("stray refit_chunk (no fetch in progress)")
        elif msg["kind"] == "refit_manifest":
            # a pushed manifest: fetch our shard's files, then hot-reload
            self._pending_refit_manifest = (
                msg["manifest"], msg.get("publisher", "")
            )

    def refit_from_peer(
        self, publisher_peer: str, manifest: dict, dest_root: str,
        reload_weights: bool = True, timeout: float = 60.0,
    ) -> str:
        """Pull the manifest's chunks covering this peer's layer range over
        the transport, verify CIDs, reassemble, GC old versions and hot-reload
        the shard weights. Chunk replies are consumed by the step loop (or
        drained here when no loop is running)."""
        import msgpack

        from . import refit as refit_mod

        fetcher = refit_mod.RefitFetcher(dest_root)

        def get(name: str, idx: int) -> bytes:
            self.transport.send(
                publisher_peer,
                msgpack.packb(
                    {"kind": "refit_get", "name": name, "idx": idx,
                     "reply_to": self.peer_id},
                    use_bin_type=True,
                ),
            )
            deadline = time.monotonic() + timeout
            while time.monotonic() < deadline:
                data = self.transport.recv(timeout=0.05)
                if data is None:
                    continue
                msg = codec.decode(data)
                if msg["kind"] == "refit_chunk":
                    if msg.get("name") == name and msg.get("idx") == idx:
                        return msg["data"]
                    continue  # stale reply from a retried request
                if msg["kind"] == "refit_get":
                    self._handle_refit(msg)
                else:
                    # pipeline packet raced the refit: step() drains these first
                    self._deferred_packets.append(data)
            raise refit_mod.RefitError(
                f"timeout fetching {name}[{idx}] from {publisher_peer}"
            )

        vdir = fetcher.fetch(
            manifest, get,
            layer_range=(self.model.start_layer, self.model.end_layer),
        )
        if reload_weights:
            from ..server.shard_loader import load_shard_weights

            n = load_shard_weights(self.model, vdir)
            if hasattr(self.model, "finalize_weights"):
                self.model.finalize_weights()
            logger.info("refit: reloaded %d tensors from %s", n, vdir)
        return vdir

    def _broadcast_control(self, kind: str, rids: List[str], routing_table: List[str]) -> None:
        for peer in routing_table:
            if peer != self.peer_id:
                try:
                    self.transport.send(peer, codec.encode_control(kind, rids))
                except (ConnectionError, OSError, KeyError) as e:
                    # a dead peer cannot receive the release; its state is
                    # reclaimed when it rejoins (fresh executor) or expires
                    logger.warning("control %s to %s failed: %s",
                                   kind, peer, e)
