"""Bisect the 2-epoch tiny-config NaN: run the end-to-end Trainer on GPU
with toggles for amp / hipgraph / workers and print per-epoch train loss +
val metrics.

    python scripts/debug_e2e.py [--amp 0|1] [--graph 0|1] [--workers N]
"""

import argparse
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import pvraft_amd.engine.trainer as trainer_mod
from pvraft_amd.engine import Trainer

ap = argparse.ArgumentParser()
ap.add_argument("--amp", type=int, default=1)
ap.add_argument("--graph", type=int, default=1)
ap.add_argument("--workers", type=int, default=2)
ap.add_argument("--epochs", type=int, default=2)
a = ap.parse_args()

tmp = tempfile.mkdtemp()
args = argparse.Namespace(
    root=tmp, exp_path="dbg", dataset="SYNTH", max_points=512,
    corr_levels=3, base_scales=0.25, truncate_k=64, iters=2, gamma=0.8,
    batch_size=2, gpus="", num_epochs=a.epochs, weights=None,
    checkpoint_interval=5, refine=False, num_workers=a.workers,
    amp=bool(a.amp), synth_len=6, hipgraph=bool(a.graph),
)
trainer_mod.VAL_ITERS = 2
t = Trainer(args)
ok = True
for epoch in range(1, a.epochs + 1):
    t.training(epoch)
    r = t.val_test(epoch, mode="val")
    print(f"epoch {epoch} val {r}", flush=True)
    if not (r["epe"] >= 0):
        ok = False
print("RESULT", "OK" if ok else "NAN", f"amp={a.amp} graph={a.graph} workers={a.workers}")
sys.exit(0 if ok else 1)
