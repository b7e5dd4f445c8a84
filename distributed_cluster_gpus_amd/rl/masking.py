"""Masked-categorical utilities for the hybrid-discrete policy.

Semantics parity with the reference masked_softmax / sample_categorical
(simcore/rl/utils.py:38-54): invalid entries get dtype-min before softmax.
On ROCm the masked softmax + sampling stays in torch (tiny tensors, batch
<=256); the GEMMs around it are what hit MFMA via rocBLAS.
"""
from typing import Optional, Tuple

import torch
import torch.nn.functional as F


def masked_softmax(logits: torch.Tensor, mask: Optional[torch.Tensor]) -> torch.Tensor:
    if mask is not None:
        mask = mask.to(dtype=torch.bool, device=logits.device)
        neg_inf = torch.finfo(logits.dtype).min
        logits = torch.where(mask, logits, torch.full_like(logits, neg_inf))
    return F.softmax(logits, dim=-1)


def sample_categorical(logits: torch.Tensor,
                       mask: Optional[torch.Tensor]) -> Tuple[torch.Tensor, torch.Tensor]:
    # Gumbel-max sampling: exact categorical draw with no host-side
    # validation or synchronization (hipGraph-capturable), equivalent in
    # distribution to torch.distributions.Categorical(probs).sample()
    if mask is not None:
        mask = mask.to(dtype=torch.bool, device=logits.device)
        neg_inf = torch.finfo(logits.dtype).min
        logits = torch.where(mask, logits, torch.full_like(logits, neg_inf))
    logp_all = torch.log_softmax(logits, dim=-1)
    u = torch.rand_like(logits).clamp_min(1e-20)
    gumbel = -torch.log(-torch.log(u))
    a = torch.argmax(logp_all + gumbel, dim=-1)
    logp = logp_all.gather(-1, a.unsqueeze(-1)).squeeze(-1)
    return a, logp
