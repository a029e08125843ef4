"""Cluster scheduler: node lifecycle + bootstrap + dispatch.

Behavior parity with the reference's scheduling/scheduler.py: join/leave/update
event handling, bootstrap once min_nodes capacity is reached, heartbeat expiry
(default 30 s), dispatch bookkeeping with per-node in-flight counters, dynamic
join/rebalance after bootstrap, weight-refit versioning. Pure logic — the
service layer (backend/) feeds it events and reads decisions."""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from ..utils.logging_config import get_logger
from .layer_allocation import (
    DynamicProgrammingLayerAllocator,
    GreedyLayerAllocator,
    Pipeline,
)
from .model_info import ModelInfo
from .node import Node
from .request_routing import RequestRouter, RoutingDecision

logger = get_logger("scheduling.scheduler")


@dataclass
class LayerAssignment:
    node_id: str
    start_layer: int
    end_layer: int
    model_name: str


class ClusterScheduler:
    def __init__(
        self,
        model: ModelInfo,
        min_nodes_bootstrapping: int = 1,
        allocator: str = "greedy",
        routing_strategy: str = "round_robin",
        heartbeat_timeout_s: float = 30.0,
    ):
        self.model = model
        self.min_nodes = min_nodes_bootstrapping
        self.heartbeat_timeout_s = heartbeat_timeout_s
        self.nodes: Dict[str, Node] = {}
        self.standby: List[Node] = []
        self.pipelines: List[Pipeline] = []
        self.allocator = (
            DynamicProgrammingLayerAllocator(model)
            if allocator == "dp"
            else GreedyLayerAllocator(model)
        )
        self.router = RequestRouter(model, routing_strategy)
        self.bootstrapped = False
        self.last_refit_time = 0.0
        self.dispatch_count = 0

    # -- node lifecycle ---------------------------------------------------------

    def node_join(self, node: Node) -> Optional[LayerAssignment]:
        self.nodes[node.node_id] = node
        node.heartbeat()
        node.model = self.model
        if not self.bootstrapped:
            self.standby.append(node)
            self._try_bootstrap()
        else:
            pipe = None
            if isinstance(self.allocator, (GreedyLayerAllocator,)):
                pipe = self.allocator.dynamic_join(node, self.pipelines)
            else:
                pipe = GreedyLayerAllocator(self.model).dynamic_join(node, self.pipelines)
            if pipe is None:
                # couldn't slot into an existing pipeline: hold in standby and
                # try to build a fresh pipeline from standby nodes
                self.standby.append(node)
                self._try_expand()
        if node.has_assignment:
            return self._assignment(node)
        return None

    def node_leave(self, node_id: str) -> None:
        node = self.nodes.pop(node_id, None)
        if node is None:
            return
        self.standby = [n for n in self.standby if n.node_id != node_id]
        broken = self.router.remove_node(node_id)
        for p in broken:
            # survivors of a broken pipeline: rebalance if they can still cover
            from .layer_allocation import apply_spans, water_fill_layers

            spans = water_fill_layers(p.nodes, self.model, self.model.num_layers)
            if spans is not None:
                apply_spans(p.nodes, spans)
                p.nodes = [n for n in p.nodes if n.num_layers_hosted > 0]
                self.router.expand_pipelines(p)
            else:
                for n in p.nodes:
                    n.clear_assignment()
                    self.standby.append(n)
        self.pipelines = self.router.pipelines
        if not self.pipelines:
            self.bootstrapped = False
            self._try_bootstrap()

    def node_update(
        self,
        node_id: str,
        layer_latency_ms: Optional[float] = None,
        current_requests: Optional[int] = None,
        rtt_ms: Optional[Dict[str, float]] = None,
        last_refit_time: Optional[float] = None,
    ) -> Optional[LayerAssignment]:
        """Heartbeat: returns the node's current assignment so the worker can
        detect re-allocation (reference p2p/server.py:786-815 mismatch check)."""
        node = self.nodes.get(node_id)
        if node is None:
            return None
        node.heartbeat()
        if layer_latency_ms is not None:
            node.set_layer_latency_ms(layer_latency_ms)
        if current_requests is not None:
            node.current_requests = current_requests
        if rtt_ms:
            node.rtt_ms.update(rtt_ms)
        if last_refit_time is not None:
            node.last_refit_time = last_refit_time
        return self._assignment(node) if node.has_assignment else None

    def sweep_heartbeats(self) -> List[str]:
        """Expel nodes silent longer than the timeout (reference :269-277)."""
        expired = [
            nid for nid, n in self.nodes.items()
            if n.is_stale(self.heartbeat_timeout_s)
        ]
        for nid in expired:
            logger.warning("node %s heartbeat expired; removing", nid)
            self.node_leave(nid)
        return expired

    # -- bootstrap / expansion -----------------------------------------------------

    def _try_bootstrap(self) -> bool:
        if self.bootstrapped or len(self.standby) < self.min_nodes:
            return False
        total_cap = sum(n.decoder_layer_capacity(self.model) for n in self.standby)
        if total_cap < self.model.num_layers:
            return False
        pipelines = self.allocator.allocate_from_standby(list(self.standby))
        if not pipelines:
            return False
        assigned = {n.node_id for p in pipelines for n in p.nodes}
        self.standby = [n for n in self.standby if n.node_id not in assigned]
        self.pipelines = pipelines
        self.router.bootstrap(pipelines)
        self.bootstrapped = True
        logger.info(
            "bootstrap: %d pipeline(s) over %d node(s)", len(pipelines), len(assigned)
        )
        return True

    def _try_expand(self) -> None:
        if not self.standby:
            return
        extra = GreedyLayerAllocator(self.model).allocate_from_standby(list(self.standby))
        for p in extra:
            assigned = {n.node_id for n in p.nodes}
            self.standby = [n for n in self.standby if n.node_id not in assigned]
            self.pipelines.append(p)
            self.router.expand_pipelines(p)

    # -- dispatch ----------------------------------------------------------------------

    def dispatch_next_request(self) -> Optional[RoutingDecision]:
        decision = self.router.find_optimal_path(self.last_refit_time)
        if decision is None:
            return None
        for n in decision.pipeline.nodes:
            n.current_requests += 1
        self.dispatch_count += 1
        return decision

    def complete_request(self, routing_table: List[str]) -> None:
        for nid in routing_table:
            n = self.nodes.get(nid)
            if n is not None and n.current_requests > 0:
                n.current_requests -= 1

    # -- weight refit -------------------------------------------------------------------

    def update_last_refit_time(self) -> float:
        self.last_refit_time = time.time()
        return self.last_refit_time

    # -- introspection ---------------------------------------------------------------------

    def _assignment(self, node: Node) -> LayerAssignment:
        return LayerAssignment(
            node_id=node.node_id,
            start_layer=node.start_layer,
            end_layer=node.end_layer,
            model_name=self.model.name,
        )

    def cluster_status(self) -> dict:
        return {
            "model": self.model.name,
            "bootstrapped": self.bootstrapped,
            "num_nodes": len(self.nodes),
            "num_pipelines": len(self.pipelines),
            "num_layers": self.model.num_layers,
            "last_refit_time": self.last_refit_time,
            "nodes": [
                {
                    "node_id": n.node_id,
                    "hardware": n.hardware.name,
                    "start_layer": n.start_layer,
                    "end_layer": n.end_layer,
                    "current_requests": n.current_requests,
                    "layer_latency_ms": n.layer_latency_ms(self.model),
                    "active": n.is_active and not n.is_stale(self.heartbeat_timeout_s),
                }
                for n in self.nodes.values()
            ],
            "pipelines": [p.node_ids for p in self.pipelines],
        }
