"""Preallocated tensor ring replay buffer + offline .npz dataset schema.

Capability parity with the reference ReplayBuffer / save_offline_npz /
load_offline_npz (simcore/rl/replay.py:26-95), re-designed MI355X-first: the
reference keeps a Python list of Transition objects and re-stacks numpy on
every sample; here all storage is preallocated torch tensors (on the training
device — 288 GB HBM3E per MI355X makes device-resident replay the default for
GPU runs), and sampling is a single fancy-index gather, no host round trip.

Offline dataset schema (unchanged, for checkpoint-format parity): .npz with
keys s, s_next, a_dc, a_g, r, done[, mask_dc, mask_g, pref, costs/<name>].
"""
from typing import Dict, List, Optional

import numpy as np
import torch


class ReplayRing:
    def __init__(self, capacity: int, obs_dim: int, n_costs: int,
                 cost_names: List[str], n_dc: int, n_g: int,
                 device: str = "cpu", seed: Optional[int] = None):
        self.capacity = int(capacity)
        self.device = torch.device(device)
        self.cost_names = list(cost_names)
        self.size = 0
        self.ptr = 0
        d = self.device
        self.s = torch.zeros((capacity, obs_dim), dtype=torch.float32, device=d)
        self.s_next = torch.zeros((capacity, obs_dim), dtype=torch.float32, device=d)
        self.a_dc = torch.zeros((capacity,), dtype=torch.long, device=d)
        self.a_g = torch.zeros((capacity,), dtype=torch.long, device=d)
        self.r = torch.zeros((capacity,), dtype=torch.float32, device=d)
        self.done = torch.zeros((capacity,), dtype=torch.float32, device=d)
        self.costs = torch.zeros((capacity, len(cost_names)), dtype=torch.float32, device=d)
        self.mask_dc = torch.ones((capacity, n_dc), dtype=torch.bool, device=d)
        self.mask_g = torch.ones((capacity, n_g), dtype=torch.bool, device=d)
        self.gen = torch.Generator(device="cpu")
        if seed is not None:
            self.gen.manual_seed(seed)

    def add(self, s, s_next, a_dc: int, a_g: int, r: float,
            costs: Dict[str, float], done: bool,
            mask_dc=None, mask_g=None):
        i = self.ptr
        self.s[i] = torch.as_tensor(s, dtype=torch.float32)
        self.s_next[i] = torch.as_tensor(s_next, dtype=torch.float32)
        self.a_dc[i] = int(a_dc)
        self.a_g[i] = int(a_g)
        self.r[i] = float(r)
        self.done[i] = float(done)
        for k, name in enumerate(self.cost_names):
            self.costs[i, k] = float(costs.get(name, 0.0))
        if mask_dc is not None:
            self.mask_dc[i] = torch.as_tensor(np.asarray(mask_dc), dtype=torch.bool)
        if mask_g is not None:
            self.mask_g[i] = torch.as_tensor(np.asarray(mask_g), dtype=torch.bool)
        self.ptr = (self.ptr + 1) % self.capacity
        self.size = min(self.size + 1, self.capacity)

    def add_batch(self, s, s_next, a_dc, a_g, r, costs, done, mask_dc=None, mask_g=None):
        """Vectorized insert of B transitions (the batched-engine path pushes
        whole device tensors of replica transitions at once)."""
        B = int(s.shape[0])
        idx = (self.ptr + torch.arange(B)) % self.capacity
        idx = idx.to(self.device)
        self.s[idx] = torch.as_tensor(s, dtype=torch.float32, device=self.device)
        self.s_next[idx] = torch.as_tensor(s_next, dtype=torch.float32, device=self.device)
        self.a_dc[idx] = torch.as_tensor(a_dc, dtype=torch.long, device=self.device)
        self.a_g[idx] = torch.as_tensor(a_g, dtype=torch.long, device=self.device)
        self.r[idx] = torch.as_tensor(r, dtype=torch.float32, device=self.device)
        self.done[idx] = torch.as_tensor(done, dtype=torch.float32, device=self.device)
        self.costs[idx] = torch.as_tensor(costs, dtype=torch.float32, device=self.device)
        if mask_dc is not None:
            self.mask_dc[idx] = torch.as_tensor(mask_dc, dtype=torch.bool, device=self.device)
        if mask_g is not None:
            self.mask_g[idx] = torch.as_tensor(mask_g, dtype=torch.bool, device=self.device)
        self.ptr = int((self.ptr + B) % self.capacity)
        self.size = min(self.size + B, self.capacity)

    def sample(self, batch_size: int) -> Dict[str, torch.Tensor]:
        idx = torch.randint(0, self.size, (batch_size,), generator=self.gen).to(self.device)
        return {
            "s": self.s[idx], "s_next": self.s_next[idx],
            "a_dc": self.a_dc[idx], "a_g": self.a_g[idx],
            "r": self.r[idx], "done": self.done[idx],
            "costs": {name: self.costs[idx, k] for k, name in enumerate(self.cost_names)},
            "mask_dc": self.mask_dc[idx], "mask_g": self.mask_g[idx],
        }

    # ---- offline .npz schema (reference replay.py:74-95) ----
    def to_npz_dict(self) -> Dict[str, np.ndarray]:
        n = self.size
        out = {
            "s": self.s[:n].cpu().numpy(), "s_next": self.s_next[:n].cpu().numpy(),
            "a_dc": self.a_dc[:n].cpu().numpy().astype(np.int64),
            "a_g": self.a_g[:n].cpu().numpy().astype(np.int64),
            "r": self.r[:n].cpu().numpy(), "done": self.done[:n].cpu().numpy(),
            "mask_dc": self.mask_dc[:n].cpu().numpy(),
            "mask_g": self.mask_g[:n].cpu().numpy(),
        }
        for k, name in enumerate(self.cost_names):
            out[f"costs/{name}"] = self.costs[:n, k].cpu().numpy()
        return out

    def save_npz(self, path: str):
        np.savez_compressed(path, **self.to_npz_dict())


def save_offline_npz(path: str, data: Dict[str, np.ndarray]):
    np.savez_compressed(path, **data)


def load_offline_npz(path: str, device: str = "cpu") -> Dict:
    """Load an offline dataset; 'costs/<name>' keys are regrouped into a dict,
    float-cast and moved to device (parity: reference load_offline_npz)."""
    z = np.load(path, allow_pickle=True)
    out: Dict = {}
    for k in z.files:
        v = z[k]
        if k.startswith("costs/"):
            out.setdefault("costs", {})
            out["costs"][k.split("/", 1)[1]] = torch.tensor(v)
        else:
            out[k] = torch.tensor(v)
    td: Dict = {}
    for k, v in out.items():
        if k == "costs":
            td[k] = {ck: cv.float().to(device) for ck, cv in v.items()}
        else:
            td[k] = v.to(device)
    return td


def offline_dataset_from_rows(rows) -> Dict[str, np.ndarray]:
    """Build the offline .npz dict from iterable env records
    (s, a_dict, r, costs_dict, s_next, done, mask_dc, mask_g)
    (capability parity: reference offline_schema_example.py:6-46)."""
    states, next_states, a_dc, a_g, rewards, dones = [], [], [], [], [], []
    mask_dc_l, mask_g_l = [], []
    cost_buf: Dict[str, list] = {}
    cost_names = None
    for (s, a, r, costs, s2, d, mdc, mg) in rows:
        states.append(s)
        next_states.append(s2)
        a_dc.append(a["dc"])
        a_g.append(a["g"])
        rewards.append(r)
        dones.append(d)
        mask_dc_l.append(mdc)
        mask_g_l.append(mg)
        if cost_names is None:
            cost_names = list(costs.keys())
            for k in cost_names:
                cost_buf[k] = []
        for k in cost_names:
            cost_buf[k].append(costs[k])
    data = {
        "s": np.asarray(states, dtype=np.float32),
        "s_next": np.asarray(next_states, dtype=np.float32),
        "a_dc": np.asarray(a_dc, dtype=np.int64),
        "a_g": np.asarray(a_g, dtype=np.int64),
        "r": np.asarray(rewards, dtype=np.float32),
        "done": np.asarray(dones, dtype=np.float32),
        "mask_dc": np.asarray(mask_dc_l, dtype=np.bool_),
        "mask_g": np.asarray(mask_g_l, dtype=np.bool_),
    }
    for k, v in cost_buf.items():
        data[f"costs/{k}"] = np.asarray(v, dtype=np.float32)
    return data
