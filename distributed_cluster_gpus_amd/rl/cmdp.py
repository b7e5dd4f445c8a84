"""PID-Lagrangian constrained-MDP wrapper.

Semantics parity with the reference LagrangianCMDP / PIDConfig / ConstraintSpec
(simcore/rl/cmdp_wrapper.py:6-57): r_eff = r - sum_i lambda_i * (c_i - target_i)+,
lambda updated by PID on the mean constraint excess, clamped to [0, 10].

MI355X-native detail: all PID state (lambda, error integral, previous error)
lives in device tensors and ``update_`` is completely synchronization-free, so
the whole SAC train step can run without host round trips (and is
hipGraph-capturable); ``update`` additionally materializes float stats.
For data-parallel training, ``cost_reduce_hook`` (a tensor -> tensor mean
reduction across ranks) keeps the PID state identical on every replica.
"""
from dataclasses import dataclass, field
from typing import Dict

import torch


@dataclass
class PIDGains:
    kp: float = 0.05
    ki: float = 0.01
    kd: float = 0.0
    clamp_min: float = 0.0
    clamp_max: float = 10.0


@dataclass
class Constraint:
    name: str
    target: float
    pid: PIDGains = field(default_factory=PIDGains)


class PIDLagrangian:
    def __init__(self, constraints: Dict[str, Constraint], device="cpu"):
        self.constraints = constraints
        self.device = torch.device(device)
        z = lambda: torch.zeros((), dtype=torch.float32, device=self.device)
        self.lmbda = {k: z() for k in constraints}
        self.err_int = {k: z() for k in constraints}
        self.err_prev = {k: z() for k in constraints}
        # DP hook: averages a scalar cost-mean TENSOR across ranks so the PID
        # state (and hence lambda) stays identical on every DP replica
        self.cost_reduce_hook = None

    def effective_reward(self, r: torch.Tensor,
                         cost_dict: Dict[str, torch.Tensor]) -> torch.Tensor:
        r_eff = r.clone()
        for name, spec in self.constraints.items():
            if name not in cost_dict:
                continue
            e = (cost_dict[name] - spec.target).clamp(min=0.0)
            r_eff = r_eff - self.lmbda[name].to(r.device) * e
        return r_eff

    @torch.no_grad()
    def update_(self, cost_dict: Dict[str, torch.Tensor]):
        """Sync-free PID update from cost tensors (no host round trip)."""
        for name, spec in self.constraints.items():
            if name not in cost_dict:
                continue
            c_mean = cost_dict[name].float().mean()
            if self.cost_reduce_hook is not None:
                c_mean = self.cost_reduce_hook(c_mean)
            c_mean = c_mean.to(self.device)
            e = (c_mean - spec.target).clamp(min=0.0)
            self.err_int[name] += e
            d = e - self.err_prev[name]
            self.err_prev[name].copy_(e)
            u = spec.pid.kp * e + spec.pid.ki * self.err_int[name] + spec.pid.kd * d
            self.lmbda[name].copy_(
                (self.lmbda[name] + u).clamp(spec.pid.clamp_min, spec.pid.clamp_max))

    def update(self, cost_dict: Dict[str, torch.Tensor]) -> Dict[str, float]:
        """PID update + float stats (synchronizes; use update_ on hot paths)."""
        self.update_(cost_dict)
        stats = {}
        for name in self.constraints:
            if name not in cost_dict:
                continue
            stats[f"lambda_{name}"] = float(self.lmbda[name].item())
            stats[f"cost_{name}"] = float(cost_dict[name].float().mean().item())
        return stats

    def to(self, device):
        self.device = torch.device(device)
        for d in (self.lmbda, self.err_int, self.err_prev):
            for k in d:
                d[k] = d[k].to(self.device)
        return self

    def state_dict(self):
        return {"lmbda": {k: float(v.item()) for k, v in self.lmbda.items()},
                "err_int": {k: float(v.item()) for k, v in self.err_int.items()},
                "err_prev": {k: float(v.item()) for k, v in self.err_prev.items()}}

    def load_state_dict(self, st):
        for attr in ("lmbda", "err_int", "err_prev"):
            d = getattr(self, attr)
            for k, v in st.get(attr, {}).items():
                if k in d:
                    d[k].fill_(float(v))
