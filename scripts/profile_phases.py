#!/usr/bin/env python3
"""Coarse phase breakdown of the stage-1 train step on one GPU.

Times (CUDA events): encoder forwards, correlation build, the 8-iteration
GRU loop forward, loss+backward, optimizer.  Run on the GPU box:
    python scripts/profile_phases.py --points 8192 --batch 2 --iters 8
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time

import torch

from pvraft_amd.data import synthetic_batch
from pvraft_amd.model import PVRaft
from pvraft_amd.model.graph import Graph
from pvraft_amd.utils import sequence_loss


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--points", type=int, default=8192)
    p.add_argument("--batch", type=int, default=2)
    p.add_argument("--iters", type=int, default=8)
    p.add_argument("--truncate_k", type=int, default=512)
    p.add_argument("--reps", type=int, default=5)
    p.add_argument("--no-amp", dest="amp", action="store_false")
    args = p.parse_args()

    device = torch.device("cuda:0")
    model = PVRaft(truncate_k=args.truncate_k).to(device)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    batch = synthetic_batch(args.batch, args.points, device=device)
    xyz1, xyz2 = batch["sequence"]

    def timeit(label, fn, reps=args.reps):
        fn()  # warm
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(reps):
            out = fn()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / reps * 1e3
        print(f"{label:35s} {dt:8.2f} ms")
        return out

    amp = torch.autocast("cuda", dtype=torch.bfloat16, enabled=args.amp)

    with torch.no_grad(), amp:
        timeit("graph_build (1 cloud)", lambda: Graph.build(xyz1, 32))
        g1 = Graph.build(xyz1, 32)
        timeit("encoder fwd (1 cloud, has graph)", lambda: model.feature_extractor(xyz1))
        fmap1, graph1 = model.feature_extractor(xyz1)
        fmap2, _ = model.feature_extractor(xyz2)
        timeit("corr build (matmul+topk+gather)", lambda: model.corr_block.build(fmap1, fmap2, xyz2))
        field = model.corr_block.build(fmap1, fmap2, xyz2)
        coords = xyz1
        timeit("corr lookup voxel (1 iter)", lambda: model.corr_block._voxel_feature(field, coords))
        timeit("corr lookup knn   (1 iter)", lambda: model.corr_block._knn_feature(field, coords))
        fct1, gctx = model.context_extractor(xyz1)
        net, inp = torch.split(fct1, [64, 64], dim=1)
        net = torch.tanh(net)
        inp = torch.relu(inp)
        corr = model.corr_block(field, coords)
        timeit("update block (1 iter)", lambda: model.update_block(net, inp, corr, coords - xyz1, gctx))

    # component-wise forward+backward (isolates backward regressions)
    def comp_fwd_bwd(label, make_out):
        def run():
            with amp:
                out = make_out()
            g = torch.autograd.grad(out.float().sum(), [p for p in model.parameters() if p.requires_grad], allow_unused=True)
            return g
        timeit(label, run, reps=3)

    comp_fwd_bwd("encoder fwd+bwd", lambda: model.feature_extractor(xyz1)[0])
    def corr_build_out():
        fm1, _ = model.feature_extractor(xyz1)
        fm2, _ = model.feature_extractor(xyz2)
        fld = model.corr_block.build(fm1, fm2, xyz2)
        return fld.corr
    comp_fwd_bwd("enc x2 + corr build fwd+bwd", corr_build_out)
    field_d = model.corr_block.build(fmap1.detach().requires_grad_(True), fmap2.detach(), xyz2)
    comp_fwd_bwd("voxel lookup fwd+bwd(conv params)", lambda: model.corr_block._voxel_feature(field_d, xyz1))
    comp_fwd_bwd("knn lookup fwd+bwd(conv params)", lambda: model.corr_block._knn_feature(field_d, xyz1))

    def full_forward():
        with amp:
            return model(batch["sequence"], num_iters=args.iters)

    timeit("full forward (no grad)", lambda: torch.no_grad()(full_forward)())
    timeit("full forward (grad)", full_forward)

    def fwd_bwd():
        opt.zero_grad(set_to_none=True)
        with amp:
            flows = model(batch["sequence"], num_iters=args.iters)
            loss = sequence_loss(flows, batch, gamma=0.8)
        loss.backward()
        return loss

    timeit("forward+backward", fwd_bwd)

    def full_step():
        loss = fwd_bwd()
        opt.step()
        return loss

    timeit("full step (fwd+bwd+adam)", full_step)


if __name__ == "__main__":
    main()
