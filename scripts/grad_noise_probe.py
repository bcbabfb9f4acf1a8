"""Measure the natural run-to-run gradient noise of the model (fp32-atomic
GN statistics + bf16 + argmax/top-k selections make backward chaotic), to
set honest tolerances for the deferred-wgrad A/B test.

Runs the SAME plain-autograd backward twice on identical weights/batch and
prints the worst per-parameter deviation; then the deferred run for
comparison.
"""

import copy
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from pvraft_amd.data import synthetic_batch
from pvraft_amd.model import PVRaft, pointwise
from pvraft_amd.parallel import GradReducer
from pvraft_amd.utils import sequence_loss


def run(model, batch, deferred):
    model = copy.deepcopy(model)
    if deferred:
        reducer = GradReducer(model)
        reducer.zero_grad()
    with torch.autocast("cuda", dtype=torch.bfloat16):
        flows = model(batch["sequence"], num_iters=3)
        loss = sequence_loss(flows, batch, gamma=0.8)
    loss.backward()
    if deferred:
        reducer.finalize()
    return loss.item(), {n: p.grad.detach().clone() for n, p in model.named_parameters() if p.grad is not None}


def compare(a, b, label):
    worst = []
    for n in a:
        d = (a[n].float() - b[n].float()).abs().max().item()
        s = a[n].float().abs().max().item() + 1e-9
        cos = torch.nn.functional.cosine_similarity(
            a[n].float().flatten(), b[n].float().flatten(), dim=0
        ).item()
        worst.append((d / s, n, d, s, cos))
    worst.sort(reverse=True)
    print(f"--- {label}: worst rel-max deviations")
    for w in worst[:8]:
        print(f"  rel={w[0]:.3e} cos={w[4]:.5f} {w[1]} (absmax {w[2]:.2e} / scale {w[3]:.2e})")
    print(f"  min cosine: {min(w[4] for w in worst):.5f}")


def main():
    torch.manual_seed(5)
    dev = torch.device("cuda:0")
    model = PVRaft(truncate_k=64).to(dev)
    batch = synthetic_batch(2, 512, device=dev, seed=9)

    l1, g1 = run(model, batch, deferred=False)
    l2, g2 = run(model, batch, deferred=False)
    l3, g3 = run(model, batch, deferred=True)
    print(f"losses: plainA={l1:.6f} plainB={l2:.6f} deferred={l3:.6f}")
    compare(g1, g2, "plain vs plain (noise floor)")
    compare(g1, g3, "plain vs deferred")


if __name__ == "__main__":
    sys.exit(main())
