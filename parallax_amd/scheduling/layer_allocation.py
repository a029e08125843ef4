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
    """Memoized state-space DP over capacity-sorted nodes (behavior parity
    with the reference's dp(i, open_residuals, finished) search,
    layer_allocation.py:758-965): for each target pipeline count k it finds
    the MINIMUM-STAGE partition of the pool into k complete pipelines —
    interleaving pipeline construction, so heterogeneous pools like
    capacities (40,30,20 | 35,30,25) over 90 layers partition into two
    3-stage pipelines where a round-robin grouping fails — then scores each
    feasible k with Z(k) = k^alpha / (T_comp + (stages/k) * RTT) and
    backtracks the winner's assignment.

    State: (i = next node index, open_residuals = sorted tuple of layers
    still needed by open pipelines, finished). Transitions: skip node /
    assign node to an open pipeline (closing it when the node can also
    afford the lm_head) / start a new pipeline (node hosts the embedding).
    Suffix-capacity pruning bounds the search."""

    alpha = 2.0
    default_rtt_ms = 5.0

    def __init__(self, model: ModelInfo):
        self.model = model

    def allocate_from_standby(self, standby: List[Node]) -> List[Pipeline]:
        L = self.model.num_layers
        nodes = sorted(
            standby, key=lambda n: n.decoder_layer_capacity(self.model),
            reverse=True,
        )
        N = len(nodes)
        cap_mid = [n.decoder_layer_capacity(self.model) for n in nodes]
        cap_first = [
            n.decoder_layer_capacity(self.model, is_first=True) for n in nodes
        ]
        cap_last = [
            n.decoder_layer_capacity(self.model, is_last=True) for n in nodes
        ]
        cap_solo = [
            n.decoder_layer_capacity(self.model, is_first=True, is_last=True)
            for n in nodes
        ]
        total_cap = sum(cap_mid)
        greedy = GreedyLayerAllocator(self.model)
        if N == 0 or L <= 0 or total_cap < L:
            return greedy.allocate_from_standby(standby)
        suffix = [0] * (N + 1)
        for i in range(N - 1, -1, -1):
            suffix[i] = suffix[i + 1] + cap_mid[i]
        max_k = min(N, total_cap // L)

        best_score, best_groups = -1.0, None
        for k in range(1, max_k + 1):
            groups = self._solve_k(
                nodes, cap_mid, cap_first, cap_last, cap_solo, suffix, L, k
            )
            if groups is None:
                continue
            pipes = self._apply_groups(groups, L)
            if pipes is None:
                continue
            score = self._score(pipes, k)
            if score > best_score:
                best_score, best_groups = score, groups
        for n in nodes:
            n.clear_assignment()
        if best_groups is None:
            return greedy.allocate_from_standby(standby)
        pipes = self._apply_groups(best_groups, L)
        return pipes if pipes is not None else greedy.allocate_from_standby(standby)

    def _apply_groups(
        self, groups: List[List[Node]], L: int
    ) -> Optional[List[Pipeline]]:
        """Water-fill spans within each DP-chosen node group."""
        for g in groups:
            for n in g:
                n.clear_assignment()
        pipes: List[Pipeline] = []
        for g in groups:
            spans = water_fill_layers(g, self.model, L)
            if spans is None:
                return None
            apply_spans(g, spans)
            for n in g:
                n.model = self.model
            pipes.append(Pipeline([n for n in g if n.num_layers_hosted > 0]))
        return pipes

    def _solve_k(
        self,
        nodes: List[Node],
        cap_mid: List[int],
        cap_first: List[int],
        cap_last: List[int],
        cap_solo: List[int],
        suffix: List[int],
        L: int,
        k: int,
    ) -> Optional[List[List[Node]]]:
        """Min-stage partition into exactly k pipelines, or None."""
        N = len(nodes)
        INF = float("inf")
        memo: Dict[Tuple[int, Tuple[int, ...], int], float] = {}
        action: Dict[Tuple[int, Tuple[int, ...], int], Tuple] = {}

        def dp(i: int, open_res: Tuple[int, ...], finished: int) -> float:
            if finished == k and not open_res:
                return 0.0
            if i == N:
                return INF
            key = (i, open_res, finished)
            if key in memo:
                return memo[key]
            new_needed = k - finished - len(open_res)
            # pruning: overshot target / not enough capacity or nodes left
            if (
                new_needed < 0
                or suffix[i] < sum(open_res) + max(0, new_needed) * L
                or finished + len(open_res) + (N - i) < k
            ):
                memo[key] = INF
                return INF

            best_cost = dp(i + 1, open_res, finished)
            best_act: Tuple = ("skip",)

            # assign node i to open pipeline j
            for j, rj in enumerate(open_res):
                if rj <= cap_last[i]:
                    # node covers the remaining layers AND the lm_head: close
                    lst = list(open_res)
                    lst.pop(j)
                    c = 1 + dp(i + 1, tuple(lst), finished + 1)
                    if c < best_cost:
                        best_cost, best_act = c, ("close", j)
                else:
                    # keep open; if the node could cover the layers but not
                    # the lm_head, it must leave >= 1 layer for a closer
                    r_after = max(1, rj - cap_mid[i])
                    lst = list(open_res)
                    lst[j] = r_after
                    lst.sort()
                    c = 1 + dp(i + 1, tuple(lst), finished)
                    if c < best_cost:
                        best_cost, best_act = c, ("assign", j)

            # start a new pipeline with node i as the embedding host
            if new_needed > 0:
                if L <= cap_solo[i]:
                    c = 1 + dp(i + 1, open_res, finished + 1)
                    if c < best_cost:
                        best_cost, best_act = c, ("solo",)
                else:
                    r_new = max(1, L - cap_first[i])
                    lst = sorted(list(open_res) + [r_new])
                    c = 1 + dp(i + 1, tuple(lst), finished)
                    if c < best_cost:
                        best_cost, best_act = c, ("start", r_new)

            memo[key] = best_cost
            action[key] = best_act
            return best_cost

        if dp(0, (), 0) == INF:
            return None

        # backtrack: replay the recorded decisions
        groups: List[List[Node]] = []
        open_list: List[Tuple[int, List[Node]]] = []  # (residual, group) sorted
        i, finished = 0, 0
        while not (finished == k and not open_list):
            if i >= N:  # inconsistent path (should not happen)
                return None
            key = (i, tuple(r for r, _ in open_list), finished)
            act = action.get(key, ("skip",))
            if act[0] == "skip":
                pass
            elif act[0] == "close":
                _, g = open_list.pop(act[1])
                g.append(nodes[i])
                groups.append(g)
                finished += 1
            elif act[0] == "assign":
                rj, g = open_list.pop(act[1])
                g.append(nodes[i])
                open_list.append((max(1, rj - cap_mid[i]), g))
                open_list.sort(key=lambda t: t[0])
            elif act[0] == "solo":
                groups.append([nodes[i]])
                finished += 1
            elif act[0] == "start":
                open_list.append((act[1], [nodes[i]]))
                open_list.sort(key=lambda t: t[0])
            i += 1
        return groups

    def _score(self, pipes: List[Pipeline], k: int) -> float:
        stages = sum(len(p.nodes) for p in pipes)
        t_comp = max(
            sum(n.node_latency_ms(self.model) for n in p.nodes) for p in pipes
        )
        return (k ** self.alpha) / (t_comp + (stages / k) * self.default_rtt_ms)
