#!/usr/bin/env python3
"""Op-level attribution of the train step (torch.profiler, 2 steps)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from pvraft_amd.data import synthetic_batch
from pvraft_amd.model import PVRaft
from pvraft_amd.utils import sequence_loss


def main():
    device = torch.device("cuda:0")
    model = PVRaft(truncate_k=512).to(device)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    batch = synthetic_batch(2, 8192, device=device)

    def step():
        opt.zero_grad(set_to_none=True)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            flows = model(batch["sequence"], num_iters=8)
            loss = sequence_loss(flows, batch, gamma=0.8)
        loss.backward()
        opt.step()

    for _ in range(3):
        step()
    torch.cuda.synchronize()
    from torch.profiler import ProfilerActivity, profile

    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA], record_shapes=True, with_stack=os.environ.get("PVRAFT_PROF_STACK", "0") == "1") as prof:
        for _ in range(2):
            step()
        torch.cuda.synchronize()
    print(prof.key_averages().table(sort_by="cuda_time_total", row_limit=25, max_name_column_width=45))
    print("=== by shape (copy-ish ops) ===")
    ka = prof.key_averages(group_by_input_shape=True)
    rows = [e for e in ka if e.key in ("aten::copy_", "aten::contiguous", "aten::clone", "aten::cat")]
    rows.sort(key=lambda e: -e.self_device_time_total)
    for e in rows[:25]:
        print(f"{e.key:18s} {str(e.input_shapes)[:90]:90s} {e.count:>5} {e.self_device_time_total/1000.0:8.2f}ms")
    if os.environ.get("PVRAFT_PROF_STACK", "0") == "1":
        print("=== stacks for hot copies ===")
        ka2 = prof.key_averages(group_by_stack_n=6)
        rows2 = [e for e in ka2 if e.key == "aten::copy_"]
        rows2.sort(key=lambda e: -e.self_device_time_total)
        for e in rows2[:8]:
            print(f"--- {e.count} calls, {e.self_device_time_total/1000.0:.2f}ms")
            for line in (e.stack or [])[:6]:
                print("   ", line)


if __name__ == "__main__":
    main()
