"""Loss and metric semantics (reference tools/loss.py, tools/metric.py)."""

import math

import numpy as np
import pytest
import torch

from pvraft_amd.utils import compute_epe, compute_epe_train, compute_loss, sequence_loss


def make_batch(B=2, N=8, mask_zero=None):
    flow = torch.randn(B, N, 3)
    mask = torch.ones(B, N, 1)
    if mask_zero is not None:
        mask[:, mask_zero] = 0
    return {"ground_truth": [mask, flow]}


def test_compute_loss_is_masked_mean_l1():
    batch = make_batch(mask_zero=0)
    est = batch["ground_truth"][1] + 1.0  # error of exactly 1 everywhere
    loss = compute_loss(est, batch)
    assert loss.item() == pytest.approx(1.0)
    # masked points must not contribute
    est2 = est.clone()
    est2[:, 0] += 1e6
    assert compute_loss(est2, batch).item() == pytest.approx(1.0)


def test_sequence_loss_gamma_weighting():
    batch = make_batch()
    gt = batch["ground_truth"][1]
    est = [gt + 1.0, gt + 2.0]  # losses 1 and 2
    loss = sequence_loss(est, batch, gamma=0.5)
    assert loss.item() == pytest.approx(0.5 * 1.0 + 1.0 * 2.0)


def test_epe_train():
    batch = make_batch()
    est = batch["ground_truth"][1] + torch.tensor([3.0, 4.0, 0.0])
    assert compute_epe_train(est, batch).item() == pytest.approx(5.0)


def test_compute_epe_thresholds():
    flow = torch.zeros(1, 4, 3)
    flow[0, :, 0] = 1.0  # |gt| = 1
    batch = {"ground_truth": [torch.ones(1, 4, 1), flow]}
    est = flow.clone()
    est[0, 0, 1] += 0.04   # strict-acc hit (0.04 < 0.05)
    est[0, 1, 1] += 0.09   # relax hit only
    est[0, 2, 1] += 0.5    # outlier (>0.3)
    epe, accs, accr, outl = compute_epe(est, batch)
    assert epe == pytest.approx((0.04 + 0.09 + 0.5) / 4)
    assert accs == pytest.approx(2 / 4)  # 0.04 and exact
    assert accr == pytest.approx(3 / 4)
    assert outl == pytest.approx(1 / 4)
