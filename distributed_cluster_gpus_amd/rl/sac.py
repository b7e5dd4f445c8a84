"""Masked hybrid-discrete SAC with quantile (QR) critics.

Algorithm parity with the reference HybridSAC (simcore/rl/hybrid_sac.py:96-244):
quantile-Huber critic loss against entropy-regularized target quantiles,
actor loss alpha*logp - Q, learned temperature against a target entropy of
-3, polyak tau=0.005, grad-clip 5.0, Adam 3e-4.

MI355X-native notes: parameters stay fp32 (the nets total ~0.6 M params —
latency, not bandwidth, bound); ``allreduce_hook`` supports data-parallel
training where each rank all-reduces the flattened gradient over RCCL/xGMI
after backward (see parallel/dist.py) — at ~2.3 MB fp32 a single fused
all-reduce per optimizer step is the right shape for the xGMI mesh
(latency-bound, not ring-bandwidth-bound).
"""
import math
from typing import Callable, Dict, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .masking import sample_categorical
from .networks import HybridActor, StateEncoder, TwinQuantileCritic


def quantile_huber_loss(pred: torch.Tensor, target: torch.Tensor,
                        taus: torch.Tensor) -> torch.Tensor:
    """QR quantile-Huber loss (pred/target [B, N], taus [N]).

    Built from the standard decomposition: a kappa=1 Huber on every
    (target_j, pred_i) pair — smooth_l1 with beta=1 IS that Huber — weighted
    by the asymmetric pinball factor |tau - 1{u < 0}| on the undershoot
    indicator of u = target - pred (spec: the reference's QR critic update,
    simcore/rl/hybrid_sac.py:83-93)."""
    pred_pairs = pred.unsqueeze(1).expand(-1, target.shape[1], -1)
    tgt_pairs = target.unsqueeze(2).expand_as(pred_pairs)    # [B, N_tgt, N_pred]
    huber = F.smooth_l1_loss(pred_pairs, tgt_pairs, reduction="none", beta=1.0)
    undershoot = (tgt_pairs.detach() < pred_pairs.detach()).float()
    pinball = (taus.view(1, -1, 1) - undershoot).abs()
    return (pinball * huber).mean()


class MaskedHybridSAC(nn.Module):
    def __init__(self, encoder: StateEncoder, actor: HybridActor,
                 critic: TwinQuantileCritic, n_quantiles: int = 32,
                 alpha: float = 0.2, actor_lr: float = 3e-4, critic_lr: float = 3e-4,
                 device: Optional[torch.device] = None, capturable: bool = False):
        super().__init__()
        self.encoder, self.actor, self.critic = encoder, actor, critic
        self.target_critic = TwinQuantileCritic(encoder.out_dim, actor.n_dc,
                                                actor.n_g, n_quantiles)
        self.target_critic.load_state_dict(self.critic.state_dict())
        self.nq = n_quantiles
        # buffer so Module.to() moves it with the model (a plain attribute
        # would stay on CPU and force a pageable H2D copy per train step —
        # slow, and illegal inside hipGraph capture)
        self.register_buffer("taus", torch.linspace(
            1.0 / (2 * n_quantiles), 1 - 1.0 / (2 * n_quantiles), n_quantiles))
        self.device = device or torch.device("cpu")
        self.log_alpha = torch.tensor(math.log(alpha), requires_grad=True,
                                      device=self.device)
        self.actor_opt = torch.optim.Adam(
            list(self.encoder.parameters()) + list(self.actor.parameters()) + [self.log_alpha],
            lr=actor_lr, capturable=capturable)
        self.critic_opt = torch.optim.Adam(self.critic.parameters(), lr=critic_lr,
                                           capturable=capturable)
        self.to(self.device)
        # optional distributed-gradient hook, called after each backward with
        # the parameter list whose grads must be averaged across ranks.
        self.allreduce_hook: Optional[Callable] = None

    @property
    def alpha(self) -> torch.Tensor:
        return self.log_alpha.exp()

    def act(self, state: torch.Tensor, mask_dc, mask_g, deterministic=False):
        state = state.to(self.device)
        h = self.encoder(state)
        if deterministic:
            return self.actor.greedy(h)
        if mask_dc is not None:
            mask_dc = mask_dc.to(self.device).bool()
        if mask_g is not None:
            mask_g = mask_g.to(self.device).bool()
        a, logp = self.actor.sample(h, mask_dc, mask_g)
        a["logp"] = logp
        return a

    @torch.no_grad()
    def _target_quantiles(self, h_next, mask_dc_n, mask_g_n):
        logits_dc, logits_g = self.actor(h_next)
        if mask_dc_n is not None:
            mask_dc_n = mask_dc_n.to(self.device).bool()
        if mask_g_n is not None:
            mask_g_n = mask_g_n.to(self.device).bool()
        a_dc_n, logp_dc_n = sample_categorical(logits_dc, mask_dc_n)
        a_g_n, logp_g_n = sample_categorical(logits_g, mask_g_n)
        logp_n = logp_dc_n + logp_g_n
        q1_t, q2_t = self.target_critic(h_next, a_dc_n, a_g_n)
        q_min = torch.min(q1_t, q2_t)
        q_min = q_min - self.alpha.detach().to(self.device).view(1,) * logp_n.view(-1, 1)
        return q_min, logp_n

    def _optimize(self, opt, loss, sync_params, clip_params=None):
        """backward -> (optional DP grad all-reduce over sync_params) ->
        clip(5.0) over clip_params (defaults to sync_params; the temperature
        scalar is synced but never clipped) -> step."""
        opt.zero_grad(set_to_none=True)
        loss.backward()
        if self.allreduce_hook is not None:
            self.allreduce_hook(sync_params)
        nn.utils.clip_grad_norm_(clip_params or sync_params, 5.0)
        opt.step()

    def _critic_phase(self, s, s_next, a_dc, a_g, r_eff, done, gamma,
                      mask_dc_n, mask_g_n):
        """QR-Huber TD loss against entropy-regularized target quantiles."""
        h = self.encoder(s)
        h_next = self.encoder(s_next).detach()
        with torch.no_grad():
            q_next, _ = self._target_quantiles(h_next, mask_dc_n, mask_g_n)
            target = (r_eff + (1 - done) * gamma * q_next).detach()
        q1, q2 = self.critic(h, a_dc, a_g)
        taus = self.taus.to(self.device)
        loss = quantile_huber_loss(q1, target, taus) + \
            quantile_huber_loss(q2, target, taus)
        self._optimize(self.critic_opt, loss, list(self.critic.parameters()))
        return loss

    def _actor_phase(self, s, mask_dc, mask_g, target_entropy):
        """-Q + alpha*logp actor loss joint with the learned-temperature
        objective (one backward over both)."""
        h_pi = self.encoder(s)
        logits_dc, logits_g = self.actor(h_pi)
        a_dc_pi, logp_dc = sample_categorical(logits_dc, mask_dc)
        a_g_pi, logp_g = sample_categorical(logits_g, mask_g)
        logp = logp_dc + logp_g
        q1_pi, q2_pi = self.critic(h_pi, a_dc_pi, a_g_pi)
        q_pi = torch.min(q1_pi, q2_pi).mean(dim=-1)
        actor_loss = (self.alpha.detach() * logp - q_pi).mean()
        temp_loss = -(self.log_alpha * (logp.detach() + target_entropy)).mean()
        shared = list(self.encoder.parameters()) + \
            list(self.actor.parameters())
        self._optimize(self.actor_opt, actor_loss + temp_loss,
                       shared + [self.log_alpha], clip_params=shared)
        return actor_loss, temp_loss

    @torch.no_grad()
    def _polyak(self, tau):
        for tp, p in zip(self.target_critic.parameters(),
                         self.critic.parameters()):
            tp.data.mul_(1 - tau).add_(tau * p.data)

    def update(self, batch: Dict[str, torch.Tensor], gamma: float = 0.99,
               tau: float = 0.005, target_entropy: float = -3.0,
               compute_stats: bool = True):
        dev = self.device
        s = batch["s"].to(dev)

        def _m(key, default=None):
            m = batch.get(key, default)
            return m.to(dev).bool() if m is not None else None

        loss_critic = self._critic_phase(
            s, batch["s_next"].to(dev), batch["a_dc"].to(dev),
            batch["a_g"].to(dev), batch["r_eff"].to(dev).unsqueeze(-1),
            batch["done"].to(dev).unsqueeze(-1), gamma,
            _m("mask_dc_n", batch.get("mask_dc")),
            _m("mask_g_n", batch.get("mask_g")))
        actor_loss, temp_loss = self._actor_phase(
            s, _m("mask_dc"), _m("mask_g"), target_entropy)
        self._polyak(tau)

        if not compute_stats:
            return {}
        return {"loss_critic": loss_critic.item(), "loss_actor": actor_loss.item(),
                "loss_temp": temp_loss.item(), "alpha": self.alpha.item()}
