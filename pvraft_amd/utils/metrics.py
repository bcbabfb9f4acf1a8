"""Scene-flow evaluation metrics (reference tools/metric.py).

EPE3D, Acc3DS (<0.05 abs or <5% rel), Acc3DR (<0.1 / <10%), Outliers
(>0.3 abs or >10% rel); rel = ||err|| / (||gt|| + 1e-4) -- thresholds and
epsilon exactly as metric.py:66-78.  The deprecated np.float of the
reference (metric.py:73-78) is not reproduced.

compute_epe_train runs on-device (torch); compute_epe is the evaluation
version returning python floats (numpy parity path).
"""

from __future__ import annotations

from typing import Tuple

import numpy as np
import torch
from torch import Tensor


def compute_epe_train(est_flow: Tensor, batch) -> Tensor:
    mask = batch["ground_truth"][0][..., 0]
    true_flow = batch["ground_truth"][1]
    error = est_flow - true_flow
    error = error[mask > 0]
    epe_per_point = torch.sqrt(torch.sum(torch.pow(error, 2.0), -1))
    return epe_per_point.mean()


def compute_epe(est_flow: Tensor, batch) -> Tuple[float, float, float, float]:
    mask = batch["ground_truth"][0].detach().cpu().numpy()[..., 0]
    sf_gt = batch["ground_truth"][1].detach().cpu().numpy()[mask > 0]
    sf_pred = est_flow.detach().cpu().numpy()[mask > 0]

    l2_norm = np.linalg.norm(sf_gt - sf_pred, axis=-1)
    epe3d = float(l2_norm.mean())

    sf_norm = np.linalg.norm(sf_gt, axis=-1)
    relative_err = l2_norm / (sf_norm + 1e-4)
    acc3d_strict = float(np.logical_or(l2_norm < 0.05, relative_err < 0.05).astype(np.float64).mean())
    acc3d_relax = float(np.logical_or(l2_norm < 0.1, relative_err < 0.1).astype(np.float64).mean())
    outlier = float(np.logical_or(l2_norm > 0.3, relative_err > 0.1).astype(np.float64).mean())
    return epe3d, acc3d_strict, acc3d_relax, outlier
