"""Utility-layer and loader coverage."""
import os
import time

import pytest
import torch

from distributed_cluster_gpus_amd.utils.csvlog import (CLUSTER_COLUMNS,
                                                       ClusterLogWriter,
                                                       JOB_COLUMNS,
                                                       JobLogWriter)
from distributed_cluster_gpus_amd.utils.logging import get_logger
from distributed_cluster_gpus_amd.utils.timers import ThroughputMeter


def test_cluster_writer_formats(tmp_path):
    p = str(tmp_path / "c.csv")
    w = ClusterLogWriter(p)
    w.row(12.3456, "us-west", 0.6, 3, 13, 3, 2, 1, 0, 4,
          0.1875, 0.123456, 7.89123, 1234.567, 98765.4321)
    w.close()
    lines = open(p, "rb").read().split(b"\r\n")
    assert lines[0].decode().split(",") == CLUSTER_COLUMNS
    row = lines[1].decode().split(",")
    assert row[0] == "12.346"      # time %.3f
    assert row[2] == "0.60"        # freq %.2f
    assert row[10] == "0.1875"     # util %.4f
    assert row[12] == "7.8912"     # acc %.4f
    assert row[13] == "1234.57"    # power %.2f
    assert row[14] == "98.7654"    # energy kJ %.4f


def test_job_writer_formats(tmp_path):
    p = str(tmp_path / "j.csv")
    w = JobLogWriter(p)
    w.row(7, "gw-us-east", "inference", 1.23456, "us-east", 0.8, 2,
          0.070123, 1.000000125, 2.5, 0, 0.0123456, 99.619, 1.204999)
    w.close()
    lines = open(p, "rb").read().split(b"\r\n")
    assert lines[0].decode().split(",") == JOB_COLUMNS
    row = lines[1].decode().split(",")
    assert row[3] == "1.2346"      # size %.4f
    assert row[5] == "0.800"       # f %.3f
    assert row[7] == "0.0701"      # net lat %.4f
    assert row[8] == "1.000000"    # start %.6f
    assert row[10] == "1.500000"   # latency = finish-start %.6f
    assert row[13] == "99.62"      # P %.2f


def test_logger_rotating_and_idempotent(tmp_path):
    d1 = str(tmp_path / "a")
    lg = get_logger(d1)
    lg.info("hello")
    for h in lg.handlers:
        h.flush()
    assert os.path.exists(os.path.join(d1, "project.log"))
    assert "hello" in open(os.path.join(d1, "project.log")).read()
    # second call with the same dir adds no duplicate handler
    n = len(lg.handlers)
    get_logger(d1)
    assert len(lg.handlers) == n
    # a different dir gets its own file
    d2 = str(tmp_path / "b")
    get_logger(d2)
    lg.info("world")
    for h in lg.handlers:
        h.flush()
    assert "world" in open(os.path.join(d2, "project.log")).read()


def test_throughput_meter():
    m = ThroughputMeter().start()
    m.add(10)
    time.sleep(0.01)
    m.stop()
    assert m.count == 10
    assert m.per_sec > 0
    assert m.elapsed_s >= 0.01


def test_loaders_fail_loudly(monkeypatch, tmp_path):
    from distributed_cluster_gpus_amd import ops
    with pytest.raises(ImportError, match="not built"):
        ops.load_sim_hip("_definitely_missing")


def test_graphed_requires_gpu():
    from distributed_cluster_gpus_amd.rl.agent import CHSACAgent, CHSACAgentConfig
    from distributed_cluster_gpus_amd.rl.graphed import GraphedSACStep
    from distributed_cluster_gpus_amd.rl.replay import ReplayRing
    if torch.cuda.is_available():
        pytest.skip("CPU-only assertion")
    agent = CHSACAgent(CHSACAgentConfig(obs_dim=4, n_dc=2, n_g_choices=2,
                                        constraints={}, device="cpu"))
    ring = ReplayRing(capacity=8, obs_dim=4, n_costs=0, cost_names=[],
                      n_dc=2, n_g=2)
    with pytest.raises(AssertionError, match="needs a GPU"):
        GraphedSACStep(agent, ring, 4)


def test_batched_engine_requires_gpu():
    if torch.cuda.is_available():
        pytest.skip("CPU-only assertion")
    from distributed_cluster_gpus_amd.configs.paper import (build_arrivals,
                                                            paper_scenario)
    from distributed_cluster_gpus_amd.engine.batched import BatchedEngine
    sc = paper_scenario()
    inf, trn = build_arrivals()
    with pytest.raises(RuntimeError, match="requires a ROCm GPU"):
        BatchedEngine(sc, inf, trn, replicas=4, duration=1.0)
