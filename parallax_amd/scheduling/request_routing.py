"""Request routing across registered pipelines.

Behavior parity with the reference's scheduling/request_routing.py:180-852:
bootstrap enumerates complete pipelines, scores them by estimated latency and
registers node-disjoint ones; dispatch round-robins (or samples / picks the
latency-optimal path) over registered pipelines, skipping overloaded,
not-ready and stale-weight nodes (the weight-version gate :797-851).
Fresh implementation."""

from __future__ import annotations

import random
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from ..utils.logging_config import get_logger
from .layer_allocation import Pipeline
from .model_info import ModelInfo
from .node import Node

logger = get_logger("scheduling.request_routing")


@dataclass
class RoutingDecision:
    pipeline: Pipeline

    @property
    def routing_table(self) -> List[str]:
        return self.pipeline.node_ids


class RequestRouter:
    """strategy: 'round_robin' | 'random' | 'latency'."""

    def __init__(self, model: ModelInfo, strategy: str = "round_robin"):
        assert strategy in ("round_robin", "random", "latency")
        self.model = model
        self.strategy = strategy
        self.pipelines: List[Pipeline] = []
        self._rr_index = 0
        self._rng = random.Random(0)

    # -- registration ---------------------------------------------------------

    def bootstrap(self, pipelines: List[Pipeline]) -> int:
        """Register complete, node-disjoint pipelines, best latency first."""
        complete = [p for p in pipelines if p.covers(self.model.num_layers)]
        complete.sort(key=lambda p: p.latency_ms(self.model))
        seen: set = set()
        self.pipelines = []
        for p in complete:
            if any(nid in seen for nid in p.node_ids):
                continue
            seen.update(p.node_ids)
            self.pipelines.append(p)
        logger.info("router registered %d pipeline(s)", len(self.pipelines))
        return len(self.pipelines)

    def expand_pipelines(self, pipeline: Pipeline) -> None:
        if pipeline not in self.pipelines and pipeline.covers(self.model.num_layers):
            self.pipelines.append(pipeline)

    def remove_node(self, node_id: str) -> List[Pipeline]:
        """Detach a departed node; returns pipelines that became incomplete."""
        broken = []
        for p in self.pipelines:
            if node_id in p.node_ids:
                p.detach_on_member_leave(node_id)
                if not p.covers(self.model.num_layers):
                    broken.append(p)
        self.pipelines = [p for p in self.pipelines if p not in broken]
        return broken

    # -- dispatch ----------------------------------------------------------------

    def _eligible(self, scheduler_refit_time: float = 0.0) -> List[Pipeline]:
        out = []
        for p in self.pipelines:
            ok = True
            for n in p.nodes:
                if not n.is_active or n.is_stale():
                    ok = False
                    break
                if n.current_requests >= n.max_requests(self.model):
                    ok = False
                    break
                # weight-version gate: skip nodes serving stale weights
                if n.last_refit_time < scheduler_refit_time:
                    ok = False
                    break
            if ok:
                out.append(p)
        return out

    def find_optimal_path(
        self, scheduler_refit_time: float = 0.0
    ) -> Optional[RoutingDecision]:
        candidates = self._eligible(scheduler_refit_time)
        if not candidates:
            return None
        if self.strategy == "random":
            return RoutingDecision(self._rng.choice(candidates))
        if self.strategy == "latency":
            # latency estimate + current load pressure
            def cost(p: Pipeline) -> float:
                lat = p.latency_ms(self.model)
                load = sum(n.current_requests for n in p.nodes)
                return lat * (1.0 + 0.1 * load)

            return RoutingDecision(min(candidates, key=cost))
        # round robin
        self._rr_index = (self._rr_index + 1) % len(candidates)
        return RoutingDecision(candidates[self._rr_index])

    @property
    def num_pipelines(self) -> int:
        return len(self.pipelines)


def dijkstra_route(
    layer_ranges: Dict[str, Tuple[int, int]],
    num_layers: int,
    latency: Optional[Dict[Tuple[str, str], float]] = None,
) -> Optional[List[str]]:
    """Schedulerless routing (reference p2p/server.py:592-626): given the
    layer ranges each peer announced over the DHT, find the cheapest chain of
    peers whose ranges concatenate to [0, num_layers). Edges connect node A
    (covering [s, m)) to node B covering [m, e); edge cost = RTT estimate
    (default 1 per hop, so min-hop). Returns node ids in pipeline order, or
    None if the layer space cannot be covered."""
    import heapq

    starts: Dict[int, List[str]] = {}
    for nid, (s, e) in layer_ranges.items():
        if e > s:
            starts.setdefault(s, []).append(nid)
    # state = layer index reached; start = 0, goal = num_layers
    best: Dict[int, float] = {0: 0.0}
    prev: Dict[int, Tuple[int, str]] = {}
    heap = [(0.0, 0, None)]
    while heap:
        cost, layer, from_nid = heapq.heappop(heap)
        if layer >= num_layers:
            break
        if cost > best.get(layer, float("inf")):
            continue
        for nid in starts.get(layer, []):
            _, e = layer_ranges[nid]
            hop = 1.0
            if latency is not None and from_nid is not None:
                hop = latency.get((from_nid, nid), 1.0)
            nxt = min(e, num_layers)
            c = cost + hop
            if c < best.get(nxt, float("inf")):
                best[nxt] = c
                prev[nxt] = (layer, nid)
                heapq.heappush(heap, (c, nxt, nid))
    if num_layers not in prev and num_layers not in best:
        return None
    if best.get(num_layers) is None:
        return None
    path = []
    at = num_layers
    while at != 0:
        layer, nid = prev[at]
        path.append(nid)
        at = layer
    return list(reversed(path))
