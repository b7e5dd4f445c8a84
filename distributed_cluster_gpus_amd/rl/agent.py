"""CHSAC-AF agent facade: encoder + hybrid actor + quantile critic + CMDP.

Capability parity with the reference CHSAC_AF
(simcore/rl/rl_energy_agent_adv_upgrade.py:10-53): select_action returns
{dc, g} ints; train_step computes r_eff = r - sum lambda*(c-target)+ on the
active constraints, runs the SAC update, then PID-updates the lambdas.

Extensions over the reference (it has none of these):
* full checkpoint save/load (weights, optimizers, temperature, lambda/PID
  state, torch RNG) — ``save`` / ``load``;
* data-parallel training over RCCL: ``enable_ddp`` installs a fused flat
  gradient all-reduce after each backward (parallel/dist.py).
"""
from dataclasses import dataclass, field
from typing import Dict, Optional

import numpy as np
import torch

from .cmdp import Constraint, PIDLagrangian
from .networks import HybridActor, StateEncoder, TwinQuantileCritic
from .sac import MaskedHybridSAC


@dataclass
class CHSACAgentConfig:
    obs_dim: int
    n_dc: int
    n_g_choices: int
    constraints: Dict[str, float] = field(default_factory=dict)
    device: str = "cpu"
    latent_dim: int = 256
    n_quantiles: int = 32
    alpha: float = 0.2
    actor_lr: float = 3e-4
    critic_lr: float = 3e-4
    graph_capturable: bool = False  # Adam capturable state (hipGraph path)


class CHSACAgent:
    def __init__(self, cfg: CHSACAgentConfig):
        self.cfg = cfg
        self.device = torch.device(cfg.device)
        self.encoder = StateEncoder(cfg.obs_dim, out_dim=cfg.latent_dim).to(self.device)
        self.actor = HybridActor(cfg.latent_dim, cfg.n_dc, cfg.n_g_choices).to(self.device)
        self.critic = TwinQuantileCritic(cfg.latent_dim, cfg.n_dc, cfg.n_g_choices,
                                         cfg.n_quantiles).to(self.device)
        self.algo = MaskedHybridSAC(self.encoder, self.actor, self.critic,
                                    n_quantiles=cfg.n_quantiles, alpha=cfg.alpha,
                                    actor_lr=cfg.actor_lr, critic_lr=cfg.critic_lr,
                                    device=self.device,
                                    capturable=cfg.graph_capturable)
        self.cmdp = PIDLagrangian(
            {k: Constraint(name=k, target=v) for k, v in cfg.constraints.items()},
            device=cfg.device)

    # ---- acting ----
    def select_action(self, obs: np.ndarray, mask_dc: Optional[np.ndarray],
                      mask_g: Optional[np.ndarray], deterministic=False) -> Dict[str, int]:
        obs_t = torch.from_numpy(np.asarray(obs)).float().unsqueeze(0)
        m_dc = torch.from_numpy(np.asarray(mask_dc)).unsqueeze(0).bool() if mask_dc is not None else None
        m_g = torch.from_numpy(np.asarray(mask_g)).unsqueeze(0).bool() if mask_g is not None else None
        a = self.algo.act(obs_t, m_dc, m_g, deterministic=deterministic)
        return {"dc": int(a["dc"].item()), "g": int(a["g"].item())}

    def select_action_batch(self, obs: torch.Tensor, mask_dc: Optional[torch.Tensor],
                            mask_g: Optional[torch.Tensor], deterministic=False):
        """Batched action selection for the replica engine: obs [B, obs_dim]
        already on device; returns {dc: LongTensor[B], g: LongTensor[B]}."""
        with torch.no_grad():
            return self.algo.act(obs, mask_dc, mask_g, deterministic=deterministic)

    def constraint_target(self, name: str) -> Optional[float]:
        spec = self.cmdp.constraints.get(name)
        return None if spec is None else spec.target

    # ---- training ----
    def train_step(self, batch: Dict[str, torch.Tensor],
                   compute_stats: bool = True) -> Dict[str, float]:
        r = batch["r"]
        costs = batch.get("costs", {})
        r_eff = r.clone()
        active = [k for k in costs.keys() if k in self.cmdp.constraints]
        for k in active:
            spec = self.cmdp.constraints[k]
            e = (costs[k] - spec.target).clamp(min=0.0)
            r_eff = r_eff - self.cmdp.lmbda[k].to(r.device) * e
        stats = self.algo.update({**batch, "r_eff": r_eff},
                                 compute_stats=compute_stats)
        if compute_stats:
            stats.update(self.cmdp.update({k: costs[k] for k in active}))
        else:
            self.cmdp.update_({k: costs[k] for k in active})
        return stats

    def enable_ddp(self):
        """Install data-parallel hooks (call after
        torch.distributed.init_process_group; see parallel/dist.py):
        fused flat gradient all-reduce after each backward, and cross-rank
        averaging of constraint-cost means so the PID-lambda state stays
        identical on every replica."""
        from ..parallel.dist import allreduce_gradients, world_size
        self.algo.allreduce_hook = allreduce_gradients
        w = world_size()
        from ..parallel.dist import allreduce_tensor_sum
        self.cmdp.cost_reduce_hook = (
            lambda t: allreduce_tensor_sum(t.clone()) / w)

    # ---- checkpointing (capability extension; the reference has none) ----
    def state_dict(self) -> Dict:
        return {
            "cfg": self.cfg.__dict__,
            "encoder": self.encoder.state_dict(),
            "actor": self.actor.state_dict(),
            "critic": self.critic.state_dict(),
            "target_critic": self.algo.target_critic.state_dict(),
            "log_alpha": self.algo.log_alpha.detach().cpu(),
            "actor_opt": self.algo.actor_opt.state_dict(),
            "critic_opt": self.algo.critic_opt.state_dict(),
            "cmdp": self.cmdp.state_dict(),
        }

    def load_state_dict(self, st: Dict):
        self.encoder.load_state_dict(st["encoder"])
        self.actor.load_state_dict(st["actor"])
        self.critic.load_state_dict(st["critic"])
        self.algo.target_critic.load_state_dict(st["target_critic"])
        with torch.no_grad():
            self.algo.log_alpha.copy_(st["log_alpha"])
        self.algo.actor_opt.load_state_dict(st["actor_opt"])
        self.algo.critic_opt.load_state_dict(st["critic_opt"])
        self.cmdp.load_state_dict(st["cmdp"])

    def save(self, path: str):
        torch.save(self.state_dict(), path)

    def load(self, path: str):
        self.load_state_dict(torch.load(path, map_location=self.device,
                                        weights_only=False))


def make_agent(cfg: CHSACAgentConfig) -> CHSACAgent:
    return CHSACAgent(cfg)
