#!/usr/bin/env python3
"""Training-trajectory A/B: HIP kernels vs pure-PyTorch reference backend.

Runs the same seeded stage-1 training (synthetic data, real optimizer) for
--steps steps on both backends and prints the loss curves.  Kernel-level
unit tests bound per-op error; this bounds the accumulated effect on the
actual optimisation trajectory (GN atomics + bf16 make it non-bitwise; the
curves must track closely and both must descend).

    python scripts/convergence_check.py --steps 30 --points 2048
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from pvraft_amd.data import synthetic_batch
from pvraft_amd.model import PVRaft
from pvraft_amd.utils import compute_epe_train, sequence_loss


def run(backend_ref: bool, steps: int, points: int, batch: int, iters: int,
        truncate_k: int, amp: bool, seed: int = 7):
    if backend_ref:
        os.environ["PVRAFT_REF_OPS"] = "1"
    else:
        os.environ.pop("PVRAFT_REF_OPS", None)
    torch.manual_seed(seed)
    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    model = PVRaft(truncate_k=truncate_k).to(device)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    losses, epes = [], []
    for step in range(steps):
        batch_t = synthetic_batch(batch, points, device=device, seed=step % 8)
        opt.zero_grad(set_to_none=True)
        with torch.autocast("cuda", dtype=torch.bfloat16, enabled=amp and device.type == "cuda"):
            flows = model(batch_t["sequence"], num_iters=iters)
            loss = sequence_loss(flows, batch_t, gamma=0.8)
        loss.backward()
        opt.step()
        losses.append(loss.item())
        epes.append(compute_epe_train(flows[-1].float(), batch_t).item())
    return losses, epes


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--points", type=int, default=2048)
    p.add_argument("--batch", type=int, default=2)
    p.add_argument("--iters", type=int, default=4)
    p.add_argument("--truncate_k", type=int, default=256)
    p.add_argument("--no-amp", dest="amp", action="store_false")
    args = p.parse_args()

    l_hip, e_hip = run(False, args.steps, args.points, args.batch, args.iters, args.truncate_k, args.amp)
    l_ref, e_ref = run(True, args.steps, args.points, args.batch, args.iters, args.truncate_k, args.amp)

    print(f"{'step':>4} {'loss_hip':>10} {'loss_ref':>10} {'epe_hip':>9} {'epe_ref':>9}")
    for i in range(args.steps):
        print(f"{i:>4} {l_hip[i]:>10.4f} {l_ref[i]:>10.4f} {e_hip[i]:>9.4f} {e_ref[i]:>9.4f}")
    h0, h1 = sum(l_hip[:5]) / 5, sum(l_hip[-5:]) / 5
    r0, r1 = sum(l_ref[:5]) / 5, sum(l_ref[-5:]) / 5
    print(f"hip: {h0:.4f} -> {h1:.4f} | ref: {r0:.4f} -> {r1:.4f}")
    assert h1 < h0 * 0.9, "HIP path did not descend"
    assert abs(h1 - r1) < 0.25 * max(r1, 0.05), "HIP and reference trajectories diverged"
    print("CONVERGENCE OK")


if __name__ == "__main__":
    main()
