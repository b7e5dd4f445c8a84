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
    probs = masked_softmax(logits, mask)
    dist = torch.distributions.Categorical(probs=probs)
    a = dist.sample()
    return a, dist.log_prob(a)
