"""Native C++ DES core: CPython-RNG bitwise compat + oracle log parity
(SURVEY §4 strategy (b): golden-log tests; the C++ core must be
byte-identical to the Python oracle, which the goldens pin)."""
import os
import random

import pytest

from distributed_cluster_gpus_amd.configs.paper import build_arrivals, paper_scenario
from distributed_cluster_gpus_amd.engine.native import NativeEngine
from distributed_cluster_gpus_amd.engine.oracle import OracleEngine
from distributed_cluster_gpus_amd.ops import have_des_core, load_des_core

pytestmark = pytest.mark.skipif(not have_des_core(),
                                reason="_des_core not built (python setup.py build_ext --inplace)")

GOLD = os.path.join(os.path.dirname(__file__), "golden")


@pytest.mark.parametrize("seed", [0, 1, 123, 987654321, 2**40 + 7])
def test_pyrandom_bitwise_vs_cpython(seed):
    m = load_des_core()
    pr, r = m.PyRandom(seed), random.Random(seed)
    assert all(pr.random() == r.random() for _ in range(3000))
    pr, r = m.PyRandom(seed), random.Random(seed)
    assert all(pr.expovariate(2.5) == r.expovariate(2.5) for _ in range(1000))
    pr, r = m.PyRandom(seed), random.Random(seed)
    assert all(pr.lognormvariate(10.8, 0.4) == r.lognormvariate(10.8, 0.4)
               for _ in range(1000))
    pr, r = m.PyRandom(seed), random.Random(seed)
    assert all(int(pr.randbelow(8)) == r._randbelow(8) for _ in range(1000))
    pr, r = m.PyRandom(seed), random.Random(seed)
    assert all(int(pr.getrandbits(32)) == r.getrandbits(32) for _ in range(1000))


def _run_native(tmp_path, algo, duration=60.0, **kw):
    sc = paper_scenario()
    inf, trn = build_arrivals()
    out = str(tmp_path / f"native_{algo}")
    eng = NativeEngine(sc, inf, trn, algo=algo, duration=duration,
                       log_interval=5.0, out_dir=out, seed=123, **kw)
    stats = eng.run()
    return stats, out


@pytest.mark.parametrize("algo,gold_prefix", [
    ("default_policy", "default_policy"),
    ("joint_nf", "joint_nf"),
])
def test_native_matches_golden(tmp_path, algo, gold_prefix):
    _, out = _run_native(tmp_path, algo)
    for produced, gold in (("cluster_log.csv", f"{gold_prefix}_cluster.csv"),
                           ("job_log.csv", f"{gold_prefix}_job.csv")):
        with open(os.path.join(out, produced), "rb") as f1, \
             open(os.path.join(GOLD, gold), "rb") as f2:
            assert f1.read() == f2.read(), f"native {algo}/{produced} != golden"


@pytest.mark.parametrize("algo,kw", [
    ("bandit", {}),
    ("carbon_cost", {}),
    ("eco_route", {}),
    ("debug", {"num_fixed_gpus": 2}),
    ("cap_greedy", {"power_cap": 30000.0}),
    ("cap_uniform", {"power_cap": 30000.0}),
])
def test_native_matches_oracle(tmp_path, algo, kw):
    """For algorithms without committed goldens: run oracle and native on the
    same seed/config and require byte-identical CSVs."""
    sc = paper_scenario()
    inf, trn = build_arrivals()
    out_o = str(tmp_path / "oracle")
    OracleEngine(sc, inf, trn, algo=algo, duration=60.0, log_interval=5.0,
                 out_dir=out_o, seed=123, **kw).run()
    _, out_n = _run_native(tmp_path, algo, duration=60.0, **kw)
    for f in ("cluster_log.csv", "job_log.csv"):
        with open(os.path.join(out_o, f), "rb") as f1, \
             open(os.path.join(out_n, f), "rb") as f2:
            assert f1.read() == f2.read(), f"native {algo}/{f} != oracle"


def test_native_speedup_vs_oracle(tmp_path):
    """The native core must be at least 10x the oracle's events/sec on the
    same workload (it measures ~50-100x; 10x is the regression floor)."""
    sc = paper_scenario()
    inf, trn = build_arrivals()
    o = OracleEngine(sc, inf, trn, algo="default_policy", duration=60.0,
                     log_interval=5.0, out_dir=str(tmp_path / "o"), seed=5)
    so = o.run()
    sn, _ = _run_native(tmp_path, "default_policy", duration=60.0)
    assert sn["events"] > 0
    assert sn["events_per_sec"] > 10 * so["events_per_sec"]


def test_native_chsac_falls_back_to_torch_path(tmp_path):
    sc = paper_scenario()
    from distributed_cluster_gpus_amd.models.arrivals import ArrivalProcess
    inf = ArrivalProcess(mode="off", rate=0.0)
    trn = ArrivalProcess(mode="poisson", rate=0.5)
    eng = NativeEngine(sc, inf, trn, algo="chsac_af", duration=40.0,
                       log_interval=5.0, out_dir=str(tmp_path / "rl"), seed=1,
                       rl_device="cpu", rl_batch=8, rl_warmup=5)
    assert eng.rl is not None
    stats = eng.run()
    assert stats["events"] > 0
