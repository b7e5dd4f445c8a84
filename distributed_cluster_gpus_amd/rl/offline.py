"""Offline CHSAC-AF training from an .npz dataset.

The reference defines the offline dataset schema (simcore/rl/replay.py:74-95,
offline_schema_example.py) but never trains from it; this module closes the
loop: load an .npz (keys s, s_next, a_dc, a_g, r, done[, mask_dc, mask_g,
costs/<name>]), batch it, and run SAC updates.
"""
from typing import Dict, Optional

import torch

from .agent import CHSACAgent, CHSACAgentConfig
from .replay import load_offline_npz


def train_offline(path: str, agent: Optional[CHSACAgent] = None,
                  epochs: int = 1, batch_size: int = 256,
                  device: str = "cpu", seed: int = 0,
                  constraints: Optional[Dict[str, float]] = None):
    """Train an agent on an offline dataset; returns (agent, stats_list)."""
    data = load_offline_npz(path, device=device)
    n = int(data["s"].shape[0])
    obs_dim = int(data["s"].shape[1])
    if agent is None:
        n_dc = int(data["mask_dc"].shape[1]) if "mask_dc" in data else \
            int(data["a_dc"].max().item()) + 1
        n_g = int(data["mask_g"].shape[1]) if "mask_g" in data else \
            int(data["a_g"].max().item()) + 1
        cons = constraints if constraints is not None else \
            {k: 0.0 for k in data.get("costs", {})}
        agent = CHSACAgent(CHSACAgentConfig(
            obs_dim=obs_dim, n_dc=n_dc, n_g_choices=n_g,
            constraints=cons, device=device))
    gen = torch.Generator().manual_seed(seed)
    stats_list = []
    for _ in range(epochs):
        perm = torch.randperm(n, generator=gen)
        for lo in range(0, n - batch_size + 1, batch_size):
            idx = perm[lo:lo + batch_size].to(data["s"].device)
            batch = {
                "s": data["s"][idx].float(),
                "s_next": data["s_next"][idx].float(),
                "a_dc": data["a_dc"][idx].long(),
                "a_g": data["a_g"][idx].long(),
                "r": data["r"][idx].float(),
                "done": data["done"][idx].float(),
                "costs": {k: v[idx] for k, v in data.get("costs", {}).items()},
            }
            if "mask_dc" in data:
                batch["mask_dc"] = data["mask_dc"][idx].bool()
            if "mask_g" in data:
                batch["mask_g"] = data["mask_g"][idx].bool()
            stats_list.append(agent.train_step(batch))
    return agent, stats_list
