"""Layer-range allocation across heterogeneous nodes.

Behavior parity with the reference's scheduling/layer_allocation.py:
- greedy allocator (capacity-sorted, look-ahead close, :582-755)
- dynamic-programming allocator scoring pipeline count k by
  Z(k) = k^alpha / (T_comp + (stages/k) * RTT)  (:758-965)
- water-filling rebalance: binary-search lambda with
  sum_i min(cap_i, lambda * power_i) = L, floor + largest-remainder (:278-400)
- dynamic join onto the lightest layers via a per-layer load heap (:35-68,193)
Fresh implementation.
"""

from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence, Tuple

from ..utils.logging_config import get_logger
from .model_info import ModelInfo
from .node import Node

logger = get_logger("scheduling.layer_allocation")


@dataclass
class Pipeline:
    """An ordered set of nodes covering layers [0, L) contiguously."""

    nodes: List[Node] = field(default_factory=list)

    def covers(self, num_layers: int) -> bool:
        if not self.nodes:
            return False
        pos = 0
        for n in self.nodes:
            if n.start_layer != pos:
                return False
            pos = n.end_layer
        return pos == num_layers

    @property
    def node_ids(self) -> List[str]:
        return [n.node_id for n in self.nodes]

    def latency_ms(self, model: ModelInfo) -> float:
        """Sum of node latencies + inter-node RTTs (reference
        request_routing.py:60-84); inf if any hop is unknown."""
        total = 0.0
        for i, n in enumerate(self.nodes):
            lat = n.node_latency_ms(model)
            if math.isinf(lat):
                return float("inf")
            total += lat
            if i + 1 < len(self.nodes):
                rtt = n.rtt_ms.get(self.nodes[i + 1].node_id)
                if rtt is None:
                    return float("inf")
                total += rtt
        return total

    def detach_on_member_leave(self, node_id: str) -> None:
        self.nodes = [n for n in self.nodes if n.node_id != node_id]


# ---------------------------------------------------------------------------
# water-filling rebalance
# ---------------------------------------------------------------------------


def water_fill_layers(
    nodes: Sequence[Node], model: ModelInfo, num_layers: int
) -> Optional[List[Tuple[int, int]]]:
    """Assign contiguous layer counts proportional to node power, capped by
    capacity: binary-search lambda s.t. sum_i min(cap_i, lambda*power_i) = L,
    then floor + largest-remainder integerization. Returns [(start, end)] in
    node order, or None if total capacity < L."""
    caps = [
        n.decoder_layer_capacity(model, is_first=(i == 0), is_last=(i == len(nodes) - 1))
        for i, n in enumerate(nodes)
    ]
    if sum(caps) < num_layers:
        return None
    powers = [max(1e-9, n.power()) for n in nodes]

    lo, hi = 0.0, num_layers / min(powers) + 1.0
    for _ in range(64):
        lam = (lo + hi) / 2
        total = sum(min(c, lam * p) for c, p in zip(caps, powers))
        if total < num_layers:
            lo = lam
        else:
            hi = lam
    lam = hi
    raw = [min(c, lam * p) for c, p in zip(caps, powers)]
    floors = [int(x) for x in raw]
    deficit = num_layers - sum(floors)
    # largest remainder, respecting caps
    order = sorted(
        range(len(nodes)), key=lambda i: raw[i] - floors[i], reverse=True
    )
    for i in order:
        if deficit <= 0:
            break
        if floors[i] < caps[i]:
            floors[i] += 1
            deficit -= 1
    # spill any remaining deficit into whoever has cap headroom
    i = 0
    while deficit > 0 and i < len(nodes):
        room = caps[i] - floors[i]
        take = min(room, deficit)
        floors[i] += take
        deficit -= take
        i += 1
    if deficit > 0:
        return None
    # drop zero-layer nodes is NOT done here; caller may prune
    spans, pos = [], 0
    for cnt in floors:
        spans.append((pos, pos + cnt))
        pos += cnt
    return spans


def apply_spans(nodes: Sequence[Node], spans: Sequence[Tuple[int, int]]) -> None:
    for n, (s, e) in zip(nodes, spans):
        n.start_layer, n.end_layer = s, e


# ---------------------------------------------------------------------------
# greedy allocator
# ---------------------------------------------------------------------------


class GreedyLayerAllocator:
    """Capacity-sorted greedy pipeline construction with a look-ahead close:
    when the remaining pool can still build another full pipeline, prefer the
    SMALLEST node able to close the current one (reference :582-755)."""

    def __init__(self, model: ModelInfo):
        self.model = model

    def allocate_from_standby(self, standby: List[Node]) -> List[Pipeline]:
        L = self.model.num_layers
        pool = sorted(
            standby, key=lambda n: n.decoder_layer_capacity(self.model), reverse=True
        )
        pipelines: List[Pipeline] = []
        while pool:
            pipe_nodes: List[Node] = []
            covered = 0
            while covered < L and pool:
                remaining = L - covered
                # can the rest of the pool still build one more full pipeline
                # after we take a closer?
                closers = [
                    n for n in pool
                    if n.decoder_layer_capacity(
                        self.model, is_first=covered == 0, is_last=True
                    ) >= remaining
                ]
                take: Node
                if closers:
                    rest_cap_after_smallest = sum(
                        n.decoder_layer_capacity(self.model) for n in pool
                    ) - min(
                        n.decoder_layer_capacity(self.model) for n in closers
                    )
                    if rest_cap_after_smallest >= L:
                        # close with the smallest sufficient node
                        take = min(
                            closers,
                            key=lambda n: n.decoder_layer_capacity(self.model),
                        )
                    else:
                        take = pool[0]
                else:
                    take = pool[0]
                pool.remove(take)
                cap = take.decoder_layer_capacity(
                    self.model, is_first=covered == 0,
                    is_last=covered + 1 >= remaining,
                )
                if cap <= 0:
                    continue
                span = min(cap, remaining)
                take.start_layer, take.end_layer = covered, covered + span
                take.model = self.model
                covered += span
                pipe_nodes.append(take)
            if covered == L:
                # rebalance the spans inside the pipeline by power
                spans = water_fill_layers(pipe_nodes, self.model, L)
                if spans:
                    apply_spans(pipe_nodes, spans)
                    pipe_nodes = [n for n in pipe_nodes if n.num_layers_hosted > 0]
                pipelines.append(Pipeline(pipe_nodes))
            else:
                # incomplete: return nodes to standby state
                for n in pipe_nodes:
                    n.clear_assignment()
                break
        return pipelines

    # -- dynamic join ---------------------------------------------------------

    def dynamic_join(self, node: Node, pipelines: List[Pipeline]) -> Optional[Pipeline]:
        """Place a new node over the lightest layers (min aggregate power per
        layer across pipelines — reference LayerLoad heap :35-68) and rebalance
        that pipeline with the newcomer inserted."""
        L = self.model.num_layers
        if not pipelines:
            return None
        load = [0.0] * L
        for p in pipelines:
            for n in p.nodes:
                for l in range(n.start_layer, n.end_layer):
                    load[l] += n.power() / max(1, n.num_layers_hosted)
        lightest = min(range(L), key=lambda l: load[l])
        # pick the pipeline whose node hosting `lightest` is weakest
        best_pipe, host = None, None
        for p in pipelines:
            for n in p.nodes:
                if n.start_layer <= lightest < n.end_layer:
                    if host is None or n.power() < host.power():
                        best_pipe, host = p, n
        if best_pipe is None:
            return None
        idx = best_pipe.nodes.index(host)
        new_order = best_pipe.nodes[: idx + 1] + [node] + best_pipe.nodes[idx + 1 :]
        spans = water_fill_layers(new_order, self.model, L)
        if spans is None:
            return None
        apply_spans(new_order, spans)
        node.model = self.model
        best_pipe.nodes = [n for n in new_order if n.num_layers_hosted > 0]
        return best_pipe


# ---------------------------------------------------------------------------
# dynamic-programming allocator
# ---------------------------------------------------------------------------


class DynamicProgrammingLayerAllocator:
    """Chooses the number of pipelines k maximizing
    Z(k) = k^alpha / (T_comp + (stages/k) * RTT), alpha = 2 (reference :758-965:
    throughput grows with replicas, per-request latency grows with stage count),
    using a DP over capacity-sorted nodes to find the minimum-stage partition
    into k complete pipelines."""

    alpha = 2.0
    default_rtt_ms = 5.0

    def __init__(self, model: ModelInfo):
        self.model = model

    def allocate_from_standby(self, standby: List[Node]) -> List[Pipeline]:
        best: Tuple[float, List[Pipeline]] = (-1.0, [])
        greedy = GreedyLayerAllocator(self.model)
        max_k = max(1, len(standby))
        for k in range(1, max_k + 1):
            # try to build exactly k pipelines from a fresh copy of assignments
            for n in standby:
                n.clear_assignment()
            pipes = self._build_k(list(standby), k)
            if pipes is None:
                continue
            score = self._score(pipes, k)
            if score > best[0]:
                best = (score, pipes)
        if best[1]:
            # re-apply the winning assignment (nodes were mutated per k-trial)
            for n in standby:
                n.clear_assignment()
            winning = self._build_k(list(standby), len(best[1]))
            return winning or []
        for n in standby:
            n.clear_assignment()
        return greedy.allocate_from_standby(standby)

    def _build_k(self, pool: List[Node], k: int) -> Optional[List[Pipeline]]:
        """Split the capacity-sorted pool round-robin into k groups, then
        water-fill each group; DP-style fallback shrinks groups that fail."""
        L = self.model.num_layers
        pool = sorted(pool, key=lambda n: n.decoder_layer_capacity(self.model), reverse=True)
        groups: List[List[Node]] = [[] for _ in range(k)]
        for i, n in enumerate(pool):
            groups[i % k].append(n)
        pipes = []
        for g in groups:
            spans = water_fill_layers(g, self.model, L)
            if spans is None:
                return None
            apply_spans(g, spans)
            for n in g:
                n.model = self.model
            pipes.append(Pipeline([n for n in g if n.num_layers_hosted > 0]))
        return pipes

    def _score(self, pipes: List[Pipeline], k: int) -> float:
        stages = sum(len(p.nodes) for p in pipes)
        t_comp = max(
            sum(n.node_latency_ms(self.model) for n in p.nodes) for p in pipes
        )
        return (k ** self.alpha) / (t_comp + (stages / k) * self.default_rtt_ms)
