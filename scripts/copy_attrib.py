#!/usr/bin/env python3
"""Parent-chain attribution of aten::copy_/clone in the train step."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from collections import Counter
import torch
from pvraft_amd.data import synthetic_batch
from pvraft_amd.model import PVRaft
from pvraft_amd.utils import sequence_loss

dev = "cuda:0" if torch.cuda.is_available() else "cpu"
model = PVRaft(truncate_k=512 if dev != "cpu" else 32).to(dev)
opt = torch.optim.Adam(model.parameters(), lr=1e-3)
batch = synthetic_batch(2, 8192 if dev != "cpu" else 256, device=dev)

def step():
    opt.zero_grad(set_to_none=True)
    with torch.autocast("cuda", dtype=torch.bfloat16, enabled=dev != "cpu"):
        flows = model(batch["sequence"], num_iters=8)
        loss = sequence_loss(flows, batch, gamma=0.8)
    loss.backward()
    opt.step()

step()
torch.cuda.synchronize() if dev != "cpu" else None
with torch.profiler.profile(record_shapes=True) as prof:
    step()
    torch.cuda.synchronize() if dev != "cpu" else None
parents = Counter()
times = Counter()
for e in prof.events():
    if e.name in ("aten::copy_",):
        p = e.cpu_parent
        chain = []
        while p is not None and len(chain) < 4:
            chain.append(p.name)
            p = p.cpu_parent
        shapes = str(tuple(tuple(s) for s in (e.input_shapes or [])))[:44]
        key = (shapes, " <- ".join(chain[:4])[:90])
        parents[key] += 1
        times[key] += e.self_device_time_total
for k, v in sorted(times.items(), key=lambda kv: -kv[1])[:14]:
    print(f"{times[k]/1000.0:8.2f}ms x{parents[k]:<5} {k[0]:44s} {k[1]}")
