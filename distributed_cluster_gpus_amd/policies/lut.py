"""Measured (f, batch) -> (time, energy) lookup for 1-GPU inference.

Behavioral spec (reference simcore/inference_lut.py — dead code there, a
usable component here): a job of l requests on n GPUs at per-GPU batch b runs
ceil(l / (n*b)) sequential batches; each batch costs the measured (t1, e1) of
its (f, b) cell, energy scaled by n.  Stored as parallel key->row arrays so
whole-ladder queries vectorize.
"""
import math
from typing import Dict, Tuple

import numpy as np


class InferenceLUT:
    def __init__(self, t1_table: Dict[Tuple[float, int], float],
                 e1_table: Dict[Tuple[float, int], float]):
        keys = sorted(set(t1_table) | set(e1_table))
        self._row = {(float(f), int(b)): i for i, (f, b) in enumerate(keys)}
        nan = float("nan")
        self._t1 = np.array([t1_table.get(k, nan) for k in keys])
        self._e1 = np.array([e1_table.get(k, nan) for k in keys])

    def cells(self):
        """All populated (f, b) cells with finite measurements."""
        return [k for k, i in self._row.items()
                if math.isfinite(self._t1[i]) and math.isfinite(self._e1[i])]

    def time_and_energy(self, n: int, f: float, b: int, l: int):
        i = self._row.get((float(f), int(b)))
        if i is None or not (math.isfinite(self._t1[i])
                             and math.isfinite(self._e1[i])):
            raise KeyError(f"No LUT entry for f={f}, b={b}")
        batches = math.ceil(max(0, int(l)) / max(1, int(n * b)))
        return batches * float(self._t1[i]), batches * n * float(self._e1[i])
