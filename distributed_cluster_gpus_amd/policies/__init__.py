from .heuristic import heuristic_allocate
from .gridsearch import best_energy_freq, best_nf_grid, energy_tuple, freq_for_perf_expand
from .bandit import UCB1DVFS
from .powercap import DVFSAtom, RunningTask, atoms_for_task, aggregate_atoms
from .lut import InferenceLUT

__all__ = [
    "heuristic_allocate", "best_energy_freq", "best_nf_grid", "energy_tuple",
    "freq_for_perf_expand", "UCB1DVFS", "DVFSAtom", "RunningTask",
    "atoms_for_task", "aggregate_atoms", "InferenceLUT",
]
