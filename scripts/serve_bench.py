"""Inference/serving throughput: hipGraph Predictor pairs/s.

    python scripts/serve_bench.py [--points 8192] [--iters 32] [--batches 1,4]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from pvraft_amd.engine import Predictor
from pvraft_amd.model import PVRaft

ap = argparse.ArgumentParser()
ap.add_argument("--points", type=int, default=8192)
ap.add_argument("--iters", type=int, default=32)
ap.add_argument("--truncate_k", type=int, default=512)
ap.add_argument("--batches", type=str, default="1,4")
ap.add_argument("--reps", type=int, default=20)
ap.add_argument("--no-amp", dest="amp", action="store_false")
a = ap.parse_args()

assert torch.cuda.is_available()
torch.manual_seed(0)
model = PVRaft(truncate_k=a.truncate_k).to("cuda:0").eval()
for bs in [int(x) for x in a.batches.split(",")]:
    pred = Predictor(model, points=a.points, batch=bs, iters=a.iters, amp=a.amp)
    xyz1 = torch.randn(bs, a.points, 3, device="cuda:0")
    xyz2 = xyz1 + 0.05 * torch.randn_like(xyz1)
    for _ in range(3):
        pred(xyz1, xyz2)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(a.reps):
        pred(xyz1, xyz2)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / a.reps
    print(
        f"serve bs={bs} points={a.points} iters={a.iters} amp={a.amp}: "
        f"{dt * 1e3:.2f} ms/call = {bs / dt:.1f} pairs/s"
    )
