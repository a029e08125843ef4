"""Node model: hardware profile, layer capacity, latency estimation.

Behavior parity with the reference's scheduling/node.py:58-163,212-324
(get_decoder_layer_capacity, max_requests from the KV budget, roofline
layer_latency overridden by heartbeat EWMA); fresh implementation with MI355X
as the canonical profile."""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from .model_info import ModelInfo

# memory split defaults (reference p2p/server.py:368-369)
# reference defaults (p2p/server.py:368-369); overridable per node so a
# deployment can trade parameter share against KV share (--param-mem-ratio /
# --kvcache-mem-ratio in the reference's server_args)
PARAM_MEM_RATIO = 0.65
KVCACHE_MEM_RATIO = 0.25


@dataclass
class NodeHardware:
    name: str = "MI355X"
    num_gpus: int = 1
    memory_gb: float = 288.0
    tflops_bf16: float = 2500.0       # dense MFMA peak
    memory_bandwidth_gbps: float = 8000.0

    @property
    def total_memory_bytes(self) -> int:
        return int(self.num_gpus * self.memory_gb * (1 << 30))


MI355X = NodeHardware()
MI355X_x8 = NodeHardware(name="MI355X x8", num_gpus=8)


def detect_hardware() -> NodeHardware:
    """Probe the local machine (reference server_info.HardwareInfo.detect):
    on a ROCm GPU read name/VRAM from torch and assume MI355X-class compute
    ratios scaled by memory; CPU fallback is a tiny test profile."""
    import torch

    if torch.cuda.is_available():
        props = torch.cuda.get_device_properties(0)
        mem_gb = props.total_memory / (1 << 30)
        scale = mem_gb / 288.0
        return NodeHardware(
            name=props.name or "ROCm GPU",
            num_gpus=torch.cuda.device_count(),
            memory_gb=mem_gb,
            tflops_bf16=2500.0 * max(scale, 0.05),
            memory_bandwidth_gbps=8000.0 * max(scale, 0.05),
        )
    return NodeHardware(name="cpu-test", num_gpus=1, memory_gb=16.0,
                        tflops_bf16=1.0, memory_bandwidth_gbps=50.0)


@dataclass
class Node:
    node_id: str
    hardware: NodeHardware = field(default_factory=lambda: MI355X)
    model: Optional[ModelInfo] = None

    # assigned layer range (set by the allocator)
    start_layer: int = -1
    end_layer: int = -1

    # dynamic state from heartbeats
    current_requests: int = 0
    # memory split (overridable per node; reference --param-mem-ratio /
    # --kvcache-mem-ratio)
    param_mem_ratio: float = PARAM_MEM_RATIO
    kvcache_mem_ratio: float = KVCACHE_MEM_RATIO
    measured_layer_latency_ms: Optional[float] = None  # EWMA from node_update
    rtt_ms: Dict[str, float] = field(default_factory=dict)  # peer -> RTT
    last_heartbeat: float = field(default_factory=time.monotonic)
    last_refit_time: float = 0.0
    is_active: bool = True

    _latency_ewma_alpha: float = 0.3

    # -- assignment ----------------------------------------------------------

    @property
    def has_assignment(self) -> bool:
        return self.start_layer >= 0 and self.end_layer > self.start_layer

    @property
    def num_layers_hosted(self) -> int:
        return max(0, self.end_layer - self.start_layer)

    @property
    def is_first(self) -> bool:
        return self.start_layer == 0

    def is_last_for(self, model: ModelInfo) -> bool:
        return self.end_layer == model.num_layers

    def clear_assignment(self) -> None:
        self.start_layer = self.end_layer = -1

    # -- capacity ------------------------------------------------------------

    def decoder_layer_capacity(
        self, model: ModelInfo, is_first: bool = False, is_last: bool = False
    ) -> int:
        """Max decoder layers this node can host (reference node.py:274-307)."""
        budget = self.hardware.total_memory_bytes * self.param_mem_ratio
        if is_first:
            budget -= model.embedding_io_bytes()
        if is_last and not model.cfg.tie_word_embeddings:
            budget -= model.lm_head_io_bytes()
        per_layer = model.decoder_layer_param_bytes()
        return max(0, int(budget // max(1, per_layer)))

    def kv_budget_bytes(self) -> int:
        return int(self.hardware.total_memory_bytes * self.kvcache_mem_ratio)

    def max_requests(self, model: ModelInfo, avg_context: int = 2048) -> int:
        """KV-budget-bounded concurrent batch size (reference node.py:212-246)."""
        if not self.has_assignment:
            return 0
        per_req = (
            model.kv_bytes_per_token_per_layer() * avg_context * self.num_layers_hosted
        )
        return max(1, int(self.kv_budget_bytes() // max(1, per_req)))

    def per_decoder_layer_kv_cache_memory(self) -> int:
        if not self.has_assignment:
            return 0
        return self.kv_budget_bytes() // self.num_layers_hosted

    # -- latency -------------------------------------------------------------

    def roofline_layer_latency_ms(self, model: ModelInfo, context_len: int = 1024) -> float:
        """max(compute, memory) per layer per token (reference node.py:58-163)."""
        flops = model.decoder_layer_flops(context_len)
        io = model.decoder_layer_io_bytes() + model.kv_bytes_per_token_per_layer() * context_len
        t_compute = flops / (self.hardware.tflops_bf16 * 1e12)
        t_memory = io / (self.hardware.memory_bandwidth_gbps * 1e9)
        return max(t_compute, t_memory) * 1e3

    def layer_latency_ms(self, model: Optional[ModelInfo] = None) -> float:
        if self.measured_layer_latency_ms is not None:
            return self.measured_layer_latency_ms
        if model is None:
            model = self.model
        if model is None:
            return 1.0
        return self.roofline_layer_latency_ms(model)

    def set_layer_latency_ms(self, measured: float) -> None:
        """EWMA update from heartbeat metrics (reference node.py:350-388)."""
        if self.measured_layer_latency_ms is None:
            self.measured_layer_latency_ms = measured
        else:
            a = self._latency_ewma_alpha
            self.measured_layer_latency_ms = (
                a * measured + (1 - a) * self.measured_layer_latency_ms
            )

    def node_latency_ms(self, model: Optional[ModelInfo] = None) -> float:
        if not self.has_assignment:
            return float("inf")
        lat = self.layer_latency_ms(model) * self.num_layers_hosted
        m = model or self.model
        if m is not None:
            bw = self.hardware.memory_bandwidth_gbps * 1e9
            if self.is_first:
                lat += m.embedding_io_bytes() / bw * 1e3 * 0.01  # gather, not full read
            if self.is_last_for(m):
                lat += m.lm_head_io_bytes() / bw * 1e3
        return lat

    # -- scheduling power (water-filling weight) --------------------------------

    def power(self) -> float:
        """Relative capability used by the water-filling rebalance: bandwidth
        for memory-bound decode (reference uses TFLOPS or bandwidth)."""
        return self.hardware.num_gpus * self.hardware.memory_bandwidth_gbps

    def heartbeat(self) -> None:
        self.last_heartbeat = time.monotonic()

    def is_stale(self, timeout_s: float = 30.0) -> bool:
        return time.monotonic() - self.last_heartbeat > timeout_s
