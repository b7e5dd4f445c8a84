"""CSV log writers — the normative output contract.

Column sets, ordering and number formats reproduce the reference exactly
(cluster schema: simulator_paper_multi.py:414-418 with row format :944-948;
job schema: :419-421 with row format :814-823; semantics documented in the
reference's docs/"log values" file).  Any engine (oracle / native / batched)
emits through these writers so logs are engine-independent.
"""
import csv
from typing import IO, List, Optional

CLUSTER_COLUMNS = ["time_s", "dc", "freq", "busy", "free",
                   "run_total", "run_inf", "run_train",
                   "q_inf", "q_train",
                   "util_inst", "util_avg", "acc_job_unit",
                   "power_W", "energy_kJ"]

JOB_COLUMNS = ["jid", "ingress", "type", "size", "dc", "f_used", "n_gpus",
               "net_lat_s", "start_s", "finish_s", "latency_s", "preempt_count",
               "T_pred", "P_pred", "E_pred"]


class _CsvBase:
    def __init__(self, path: str, columns: List[str]):
        self.path = path
        with open(path, "w", newline="") as f:
            csv.writer(f).writerow(columns)
        self._fh: Optional[IO] = None

    def _open(self) -> IO:
        # keep the file open across rows for speed; reopen-per-row (the
        # reference's pattern) costs ~30% of its event loop.
        if self._fh is None:
            self._fh = open(self.path, "a", newline="")
        return self._fh

    def flush(self):
        if self._fh is not None:
            self._fh.flush()

    def close(self):
        if self._fh is not None:
            self._fh.close()
            self._fh = None


class ClusterLogWriter(_CsvBase):
    def __init__(self, path: str):
        super().__init__(path, CLUSTER_COLUMNS)

    def row(self, now, dc_name, freq, busy, free, run_total, run_inf, run_trn,
            q_inf, q_train, util_inst, util_avg, acc_job_unit, power_w, energy_j):
        csv.writer(self._open()).writerow([
            f"{now:.3f}", dc_name, f"{freq:.2f}",
            busy, free, run_total, run_inf, run_trn,
            q_inf, q_train,
            f"{util_inst:.4f}", f"{util_avg:.4f}", f"{acc_job_unit:.4f}",
            f"{power_w:.2f}", f"{energy_j / 1000.0:.4f}"])


class JobLogWriter(_CsvBase):
    def __init__(self, path: str):
        super().__init__(path, JOB_COLUMNS)

    def row(self, jid, ingress, jtype, size, dc_name, f_used, n_gpus,
            net_lat_s, start_s, finish_s, preempt_count, t_pred, p_pred, e_pred):
        csv.writer(self._open()).writerow([
            jid, ingress, jtype, f"{size:.4f}", dc_name,
            f"{f_used:.3f}", n_gpus, f"{net_lat_s:.4f}",
            f"{start_s:.6f}", f"{finish_s:.6f}",
            f"{finish_s - start_s:.6f}",
            f"{preempt_count}",
            f"{t_pred:.6f}", f"{p_pred:.2f}", f"{e_pred:.2f}"])
