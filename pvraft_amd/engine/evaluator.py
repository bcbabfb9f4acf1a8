"""Standalone evaluation (reference test.py:70-156).

Builds the FT3D test split or KITTI, runs the (optionally refined) model
with 32 GRU iterations (test.py:120 hard-codes 32 regardless of --iters)
and reports mean EPE3D / Acc3DS / Acc3DR / Outlier.

On GPU the forward is served by the hipGraph Predictor (fixed shapes); it
exposes every per-iteration flow, so the logged loss is the same
gamma-weighted sequence loss the reference's test.py computes
(test.py:122).  Disable with --no_hipgraph for the eager path.
"""

from __future__ import annotations

import os

import torch
from torch.utils.data import DataLoader

from pvraft_amd.data import FT3D, Batch, Kitti, SyntheticSceneFlow
from pvraft_amd.model import build_model
from pvraft_amd.utils import (
    compute_epe,
    compute_loss,
    load_checkpoint,
    sequence_loss,
    setup_logger,
)

TEST_ITERS = 32  # reference test.py:120


def build_eval_dataset(args):
    if args.dataset == "FT3D":
        ddir = os.path.join(args.root, "data", "FlyingThings3D_subset_processed_35m")
        return FT3D(ddir, args.max_points, "test")
    if args.dataset == "KITTI":
        ddir = os.path.join(args.root, "data", "kitti_processed")
        return Kitti(ddir, args.max_points)
    if args.dataset == "SYNTH":
        return SyntheticSceneFlow(args.max_points, length=getattr(args, "synth_len", 32), seed=7)
    raise ValueError(f"Unknown dataset {args.dataset!r}")


@torch.no_grad()
def evaluate(args):
    log = setup_logger(args.root, args.exp_path or "test", f"TestAlone_{args.dataset}")
    device = torch.device("cuda:0") if torch.cuda.is_available() else torch.device("cpu")

    dataset = build_eval_dataset(args)
    loader = DataLoader(
        dataset,
        batch_size=1,
        shuffle=False,
        num_workers=getattr(args, "num_workers", 8),
        collate_fn=Batch,
        pin_memory=device.type == "cuda",
    )

    model = build_model(args).to(device)
    if args.weights:
        path = args.weights
        if not os.path.isfile(path):
            path = os.path.join(args.root, "experiments", args.weights, "checkpoints", "best_checkpoint.params")
        load_checkpoint(path, model, strict=True)
        log.info(f"Loaded weights from {path}")
    model.eval()

    dump = bool(getattr(args, "dump_results", False))
    # hipGraph-captured inference (fixed bs=1 x max_points shapes)
    predictor = None
    if device.type == "cuda" and getattr(args, "hipgraph", True):
        from .predictor import Predictor

        predictor = Predictor(model, points=args.max_points, batch=1, iters=TEST_ITERS,
                              amp=bool(getattr(args, "amp", False)))
    sums = [0.0] * 5
    n = 0
    for batch in loader:
        batch = batch.to(device, non_blocking=True)
        if predictor is not None and batch["sequence"][0].shape[1] == args.max_points:
            final = predictor(batch["sequence"][0], batch["sequence"][1])
            flows = predictor.last_flows
            # reference test.py:122 logs the gamma-weighted sequence loss
            # for the stage-1 model (single refined flow -> plain loss)
            if len(flows) > 1:
                loss = sequence_loss(flows, batch, gamma=args.gamma if hasattr(args, "gamma") else 0.8)
            else:
                loss = compute_loss(final, batch)
        else:
            est_flow = model(batch["sequence"], num_iters=TEST_ITERS)
            if isinstance(est_flow, (list, tuple)):
                loss = sequence_loss(est_flow, batch, gamma=args.gamma if hasattr(args, "gamma") else 0.8)
                final = est_flow[-1]
            else:
                loss = compute_loss(est_flow, batch)
                final = est_flow
        epe3d, accs, accr, outl = compute_epe(final.float(), batch)
        for j, v in enumerate((loss.item(), epe3d, accs, accr, outl)):
            sums[j] += v
        if dump:
            import numpy as np

            d = os.path.join(args.root, "result", args.dataset, str(n))
            os.makedirs(d, exist_ok=True)
            np.save(os.path.join(d, "pc1.npy"), batch["sequence"][0].cpu().numpy())
            np.save(os.path.join(d, "pc2.npy"), batch["sequence"][1].cpu().numpy())
            np.save(os.path.join(d, "flow.npy"), final.cpu().numpy())
        n += 1
    means = [s / max(n, 1) for s in sums]
    log.info(
        f"[test {args.dataset}] loss={means[0]:.4f} EPE3D={means[1]:.4f} "
        f"Acc3DS={means[2]:.4f} Acc3DR={means[3]:.4f} Outlier={means[4]:.4f}"
    )
    return {
        "loss": means[0],
        "epe": means[1],
        "acc3d_strict": means[2],
        "acc3d_relax": means[3],
        "outlier": means[4],
    }
