"""WAN topology: ingress gateways, directed latency/capacity/cost graph.

Capability parity: reference simcore/network.py (Ingress/Edge/Graph with
Dijkstra ``shortest_path_latency``).  The graph is static for a scenario, so
engines consume precomputed all-pairs (ingress x DC) tables built once at init
by ``dijkstra_tables`` — the batched HIP engine keeps them as constant-like
device arrays.
"""
import heapq
import math
from dataclasses import dataclass, field
from typing import Dict, List, Tuple


@dataclass(frozen=True)
class IngressSpec:
    name: str
    region: str


@dataclass
class _Edge:
    to: str
    latency_ms: float
    capacity_gbps: float = math.inf
    cost_per_gb: float = 0.0


@dataclass
class WanGraph:
    adj: Dict[str, List[_Edge]] = field(default_factory=dict)

    def add_edge(self, u: str, v: str, latency_ms: float,
                 capacity_gbps: float = math.inf, cost_per_gb: float = 0.0) -> None:
        self.adj.setdefault(u, []).append(_Edge(v, latency_ms, capacity_gbps, cost_per_gb))

    def shortest_path(self, src: str, dst: str) -> Tuple[float, List[str], float, float]:
        """Dijkstra by latency.  Returns (latency_s, path, bottleneck_gbps,
        sum_cost_per_gb); bottleneck 0.0 means 'unconstrained' (all-inf edges),
        matching the reference's convention (simcore/network.py:33-62)."""
        dist: Dict[str, float] = {src: 0.0}
        prev: Dict[str, Tuple[str, _Edge]] = {}
        pq: List[Tuple[float, str]] = [(0.0, src)]
        while pq:
            d, u = heapq.heappop(pq)
            if u == dst:
                break
            if d > dist.get(u, math.inf):
                continue
            for e in self.adj.get(u, []):
                nd = d + e.latency_ms
                if nd < dist.get(e.to, math.inf):
                    dist[e.to] = nd
                    prev[e.to] = (u, e)
                    heapq.heappush(pq, (nd, e.to))
        if dst not in dist:
            return math.inf, [], 0.0, math.inf
        path = [dst]
        bottleneck = math.inf
        cost_sum = 0.0
        cur = dst
        while cur != src:
            pu, e = prev[cur]
            path.append(pu)
            bottleneck = min(bottleneck, e.capacity_gbps)
            cost_sum += e.cost_per_gb
            cur = pu
        path.reverse()
        return dist[dst] / 1000.0, path, (0.0 if bottleneck is math.inf else bottleneck), cost_sum


def dijkstra_tables(graph: WanGraph, ingress_names: List[str], dc_names: List[str]):
    """Precompute (latency_s, bottleneck_gbps, cost_per_gb, path) for every
    ingress->DC pair.  Returns three dense [n_ing][n_dc] float lists and a path
    dict; identical numbers to per-arrival Dijkstra since the graph is static."""
    n_i, n_d = len(ingress_names), len(dc_names)
    lat = [[math.inf] * n_d for _ in range(n_i)]
    bw = [[0.0] * n_d for _ in range(n_i)]
    cost = [[math.inf] * n_d for _ in range(n_i)]
    paths = {}
    for i, ing in enumerate(ingress_names):
        for d, dc in enumerate(dc_names):
            L, path, B, C = graph.shortest_path(ing, dc)
            lat[i][d], bw[i][d], cost[i][d] = L, B, C
            paths[(ing, dc)] = path
    return lat, bw, cost, paths
