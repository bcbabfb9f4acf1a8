"""Training losses (reference tools/loss.py).

compute_loss: masked mean-L1 between estimated and ground-truth flow
(loss.py:34-38).  sequence_loss: exponentially weighted sum over the
per-iteration predictions, weight gamma^(T-1-i) (loss.py:8-11).
"""

from __future__ import annotations

from typing import List, Sequence

import torch
from torch import Tensor


def compute_loss(est_flow: Tensor, batch) -> Tensor:
    mask = batch["ground_truth"][0][..., 0]
    true_flow = batch["ground_truth"][1]
    error = est_flow - true_flow
    # masked mean-L1 (identical to the reference's error[mask > 0].abs().mean(),
    # loss.py:34-38) written with static shapes so the training step is
    # hipGraph-capturable (boolean indexing has a data-dependent shape)
    m = (mask > 0).to(error.dtype)
    total = (error.abs() * m.unsqueeze(-1)).sum()
    count = m.sum() * error.shape[-1]
    return total / count.clamp(min=1)


def sequence_loss(est_flow: Sequence[Tensor], batch, gamma: float = 0.8) -> Tensor:
    n = len(est_flow)
    loss = 0.0
    for i in range(n):
        loss = loss + (gamma ** (n - i - 1)) * compute_loss(est_flow[i], batch)
    return loss
