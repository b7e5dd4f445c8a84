"""WAN graph / scenario-table tests."""
import math

import numpy as np
import pytest

from distributed_cluster_gpus_amd.configs.paper import paper_scenario, single_dc_scenario
from distributed_cluster_gpus_amd.models.wan import WanGraph, dijkstra_tables


def test_dijkstra_direct_and_multihop():
    g = WanGraph()
    g.add_edge("a", "b", 10)
    g.add_edge("b", "c", 5)
    g.add_edge("a", "c", 100)
    L, path, bw, cost = g.shortest_path("a", "c")
    assert L == pytest.approx(0.015)
    assert path == ["a", "b", "c"]
    assert bw == 0.0   # all edges unconstrained -> 0.0 convention
    assert cost == 0.0


def test_dijkstra_unreachable():
    g = WanGraph()
    g.add_edge("a", "b", 1)
    L, path, bw, cost = g.shortest_path("a", "zzz")
    assert L == math.inf and path == [] and cost == math.inf


def test_dijkstra_bottleneck_and_cost():
    g = WanGraph()
    g.add_edge("a", "b", 1, capacity_gbps=10.0, cost_per_gb=0.1)
    g.add_edge("b", "c", 1, capacity_gbps=2.0, cost_per_gb=0.3)
    L, path, bw, cost = g.shortest_path("a", "c")
    assert bw == pytest.approx(2.0)
    assert cost == pytest.approx(0.4)


def test_paper_scenario_shape(paper_sc):
    sc = paper_sc
    assert sc.n_dc == 8 and sc.n_ing == 8 and sc.n_freq == 8
    assert int(sc.total_gpus.sum()) == 1488  # 16+32+256+16+128+16+512+512
    assert sc.power_coeffs.shape == (8, 2, 3)
    assert sc.latency_coeffs.shape == (8, 2, 3)
    # gateway->own-DC latencies (reference paper_config.py topology)
    i = sc.ingress_names.index("gw-us-west")
    d = sc.dc_names.index("us-west")
    assert sc.wan_latency_s[i][d] == pytest.approx(0.012)
    # multihop: gw-us-west -> eu-west has no direct edge; shortest is via
    # eu-central (110) -> gw-eu-west (20) -> eu-west (10) = 140 ms
    d2 = sc.dc_names.index("eu-west")
    assert sc.wan_latency_s[i][d2] == pytest.approx(0.140)
    assert sc.wan_paths[("gw-us-west", "eu-west")] == \
        ["gw-us-west", "eu-central", "gw-eu-west", "eu-west"]
    # carbon only for 3 DCs
    cv = sc.carbon_vec()
    assert (cv > 0).sum() == 3
    # hourly tariff
    pv = sc.price_vec24()
    assert pv[0] == 0.12 and pv[12] == 0.20 and pv[20] == 0.16


def test_paper_scenario_exact_coeffs(paper_sc):
    sc = paper_sc
    d = sc.dc_names.index("us-west")
    assert tuple(sc.power_coeffs[d, 1, :]) == (75.0, 80.0, 110.0)    # training
    assert tuple(sc.latency_coeffs[d, 0, :]) == (0.0090, 0.0018, 0.0007)  # inference


def test_all_ingress_reach_all_dcs(paper_sc):
    assert np.isfinite(paper_sc.wan_latency_s).all()


def test_single_dc_scenario():
    sc = single_dc_scenario()
    assert sc.n_dc == 1 and sc.n_ing == 1
    assert int(sc.total_gpus[0]) == 128
    assert sc.wan_latency_s[0][0] == pytest.approx(0.012)


def test_make_dc_states(paper_sc):
    dcs = paper_sc.make_dc_states()
    assert list(dcs) == paper_sc.dc_names
    dc = dcs["sa-east"]
    assert dc.total_gpus == 512 and dc.current_freq == 1.0 and dc.free_gpus == 512
