"""PID-Lagrangian constrained-MDP wrapper.

Semantics parity with the reference LagrangianCMDP / PIDConfig / ConstraintSpec
(simcore/rl/cmdp_wrapper.py:6-57): r_eff = r - sum_i lambda_i * (c_i - target_i)+,
lambda updated by PID on the mean constraint excess, clamped to [0, 10].
Host-side scalar math; lambdas broadcast to all DP ranks by the trainer.
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
    def __init__(self, constraints: Dict[str, Constraint]):
        self.constraints = constraints
        self.lmbda = {k: torch.tensor(0.0) for k in constraints}
        self.err_int = {k: 0.0 for k in constraints}
        self.err_prev = {k: 0.0 for k in constraints}
        # data-parallel hook: averages a scalar cost mean across ranks so the
        # PID state (and hence lambda) stays identical on every DP replica
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

    def update(self, cost_dict: Dict[str, torch.Tensor]) -> Dict[str, float]:
        stats = {}
        for name, spec in self.constraints.items():
            if name not in cost_dict:
                continue
            c_mean = float(cost_dict[name].mean().item())
            if self.cost_reduce_hook is not None:
                c_mean = self.cost_reduce_hook(c_mean)
            e = max(0.0, c_mean - spec.target)
            self.err_int[name] += e
            d = e - self.err_prev[name]
            self.err_prev[name] = e
            u = spec.pid.kp * e + spec.pid.ki * self.err_int[name] + spec.pid.kd * d
            lam = float(self.lmbda[name].item()) + u
            lam = max(spec.pid.clamp_min, min(spec.pid.clamp_max, lam))
            self.lmbda[name] = torch.tensor(lam)
            stats[f"lambda_{name}"] = lam
            stats[f"cost_{name}"] = c_mean
        return stats

    def state_dict(self):
        return {"lmbda": {k: float(v.item()) for k, v in self.lmbda.items()},
                "err_int": dict(self.err_int), "err_prev": dict(self.err_prev)}

    def load_state_dict(self, st):
        for k, v in st.get("lmbda", {}).items():
            if k in self.lmbda:
                self.lmbda[k] = torch.tensor(float(v))
        self.err_int.update(st.get("err_int", {}))
        self.err_prev.update(st.get("err_prev", {}))
