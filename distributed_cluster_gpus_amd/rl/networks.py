"""CHSAC-AF networks: state encoder, hybrid categorical actor, twin quantile critic.

Architecture parity with the reference (simcore/rl/encoders.py:5-18,
simcore/rl/hybrid_sac.py:10-80): 3x256 ReLU MLP encoder; two categorical
heads (DC choice, GPU-count choice); twin quantile critics (32 quantiles) on
[latent, one-hot(dc), one-hot(g)].  All GEMMs are 256-wide — on MI355X they
run through rocBLAS MFMA paths; the fused end-to-end inference kernel for
batched replica serving lives in ops/ (round-2 work if profiling warrants).
"""
from typing import Dict, Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from .masking import sample_categorical


class StateEncoder(nn.Module):
    def __init__(self, in_dim: int, hid: int = 256, out_dim: int = 256):
        super().__init__()
        self.out_dim = out_dim
        self.net = nn.Sequential(
            nn.Linear(in_dim, hid), nn.ReLU(),
            nn.Linear(hid, hid), nn.ReLU(),
            nn.Linear(hid, out_dim), nn.ReLU(),
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.net(x)


class HybridActor(nn.Module):
    """Two independent categorical heads over DC index and GPU count."""

    def __init__(self, latent_dim: int, n_dc: int, n_g: int, hid: int = 256):
        super().__init__()
        self.n_dc, self.n_g = n_dc, n_g
        self.head_dc = nn.Sequential(nn.Linear(latent_dim, hid), nn.ReLU(),
                                     nn.Linear(hid, n_dc))
        self.head_g = nn.Sequential(nn.Linear(latent_dim, hid), nn.ReLU(),
                                    nn.Linear(hid, n_g))

    def forward(self, h: torch.Tensor):
        return self.head_dc(h), self.head_g(h)

    def sample(self, h: torch.Tensor, mask_dc: Optional[torch.Tensor],
               mask_g: Optional[torch.Tensor]) -> Tuple[Dict[str, torch.Tensor], torch.Tensor]:
        logits_dc, logits_g = self.forward(h)
        a_dc, logp_dc = sample_categorical(logits_dc, mask_dc)
        a_g, logp_g = sample_categorical(logits_g, mask_g)
        return {"dc": a_dc, "g": a_g}, logp_dc + logp_g

    def greedy(self, h: torch.Tensor) -> Dict[str, torch.Tensor]:
        logits_dc, logits_g = self.forward(h)
        return {"dc": torch.argmax(logits_dc, dim=-1),
                "g": torch.argmax(logits_g, dim=-1)}


class TwinQuantileCritic(nn.Module):
    """Twin distributional critics returning n_quantiles values each."""

    def __init__(self, latent_dim: int, n_dc: int, n_g: int,
                 n_quantiles: int = 32, hid: int = 256):
        super().__init__()
        self.n_dc, self.n_g, self.nq = n_dc, n_g, n_quantiles
        in_dim = latent_dim + n_dc + n_g

        def mlp():
            return nn.Sequential(nn.Linear(in_dim, hid), nn.ReLU(),
                                 nn.Linear(hid, hid), nn.ReLU(),
                                 nn.Linear(hid, n_quantiles))
        self.q1, self.q2 = mlp(), mlp()

    def forward(self, h: torch.Tensor, a_dc: torch.Tensor, a_g: torch.Tensor):
        x = torch.cat([h,
                       F.one_hot(a_dc, num_classes=self.n_dc).float(),
                       F.one_hot(a_g, num_classes=self.n_g).float()], dim=-1)
        return self.q1(x), self.q2(x)
