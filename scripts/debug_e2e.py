"""Bisect the 2-epoch tiny-config hipGraph NaN: run the end-to-end Trainer
on GPU with toggles for amp / workers and for each inter-epoch component
(val pass, checkpoint saves, epe computation, prefetcher), printing every
train-step loss.

    python scripts/debug_e2e.py [--amp 0|1] [--graph 0|1] [--workers N]
        [--no_val] [--no_save] [--no_epe] [--no_prefetch] [--epochs E]
"""

import argparse
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import pvraft_amd.engine.trainer as trainer_mod
from pvraft_amd.engine import Trainer

ap = argparse.ArgumentParser()
ap.add_argument("--amp", type=int, default=1)
ap.add_argument("--graph", type=int, default=1)
ap.add_argument("--workers", type=int, default=0)
ap.add_argument("--epochs", type=int, default=2)
ap.add_argument("--no_val", action="store_true")
ap.add_argument("--no_save", action="store_true")
ap.add_argument("--no_epe", action="store_true")
ap.add_argument("--no_prefetch", action="store_true")
a = ap.parse_args()

if a.no_save:
    trainer_mod.save_checkpoint = lambda *x, **k: None
    trainer_mod.save_train_state = lambda *x, **k: None
if a.no_epe:
    trainer_mod.compute_epe_train = lambda f, b: torch.zeros((), device=f.device)
if a.no_prefetch:
    class _Plain:
        def __init__(self, loader, device):
            self.loader, self.device = loader, device

        def __len__(self):
            return len(self.loader)

        def __iter__(self):
            for b in self.loader:
                yield b.to(self.device)

    trainer_mod.CudaPrefetcher = _Plain

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

orig_step = t.train_step
losses = []


def scan(batch, tag):
    bad = []
    for name, p in t.model.named_parameters():
        if not torch.isfinite(p).all():
            bad.append(f"param:{name}")
        if p.grad is not None and not torch.isfinite(p.grad).all():
            bad.append(f"grad:{name}")
    for i, b in enumerate(t.reducer.buckets):
        if not torch.isfinite(b.flat).all():
            bad.append(f"bucket{i}")
    for key in batch.data:
        for i, x in enumerate(batch.data[key]):
            if not torch.isfinite(x).all():
                bad.append(f"batch:{key}[{i}]")
    if getattr(t, "_static_batch", None) is not None:
        for key in t._static_batch.data:
            for i, x in enumerate(t._static_batch.data[key]):
                if not torch.isfinite(x).all():
                    bad.append(f"static:{key}[{i}]")
    gs = getattr(t, "_graph_step", None)
    if gs is not None and gs.static_final_flow is not None:
        if not torch.isfinite(gs.static_final_flow).all():
            bad.append("static_final_flow")
    if bad:
        print(f"  NONFINITE at {tag}: {bad[:8]}", flush=True)
    return bad


def step_spy(batch):
    pre = scan(batch, f"pre-step{len(losses)}")
    loss, flow = orig_step(batch)
    losses.append(float(loss.item()))
    if losses[-1] != losses[-1] or abs(losses[-1]) > 1e6:
        scan(batch, f"post-step{len(losses) - 1} loss={losses[-1]:.3e}")
    return loss, flow


t.train_step = step_spy

ok = True
for epoch in range(1, a.epochs + 1):
    losses.clear()
    t.training(epoch)
    print(f"epoch {epoch} losses {[round(x, 4) for x in losses]}", flush=True)
    if not a.no_val:
        r = t.val_test(epoch, mode="val")
        print(f"epoch {epoch} val epe {r['epe']:.4f}", flush=True)
        if not (r["epe"] >= 0):
            ok = False
    if any(x != x for x in losses):
        ok = False
toggles = f"amp={a.amp} graph={a.graph} workers={a.workers} no_val={a.no_val} no_save={a.no_save} no_epe={a.no_epe} no_prefetch={a.no_prefetch}"
print("RESULT", "OK" if ok else "NAN", toggles)
sys.exit(0 if ok else 1)
