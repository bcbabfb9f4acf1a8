"""Attribute per-kernel GPU time to ATen ops for the flagship training step.

The rocprof kernel-stats show ~300 bf16 copy launches and ~240 small fp32
adds per step; under hipGraph replay they cannot be attributed.  This runs
the EAGER step (same math) under torch.profiler and prints, for the copy /
add kernel families, which ATen ops launch them, so the fusion work targets
the real producers.  Usage (GPU box):
    python scripts/step_attrib.py
"""
import sys

import torch
from torch.profiler import ProfilerActivity, profile

sys.path.insert(0, ".")

from pvraft_amd.data import synthetic_batch  # noqa: E402
from pvraft_amd.model import PVRaft  # noqa: E402
from pvraft_amd.parallel import GradReducer  # noqa: E402
from pvraft_amd.utils import sequence_loss  # noqa: E402


def main():
    device = torch.device("cuda")
    torch.manual_seed(1234)
    model = PVRaft(truncate_k=512).to(device)
    reducer = GradReducer(model)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    batch = synthetic_batch(2, 8192, device=device, seed=100)
    model.train()

    def step():
        reducer.zero_grad()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            est = model(batch["sequence"], num_iters=8)
            loss = sequence_loss(est, batch, gamma=0.8)
        loss.backward()
        reducer.finalize()
        opt.step()

    for _ in range(3):
        step()
    torch.cuda.synchronize()

    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof:
        for _ in range(2):
            step()
        torch.cuda.synchronize()

    evs = prof.key_averages()
    print(evs.table(sort_by="self_device_time_total", row_limit=45))

    # map device kernels back to the launching aten op
    print("\n=== copy/add kernel attribution (op -> kernels) ===")
    interesting = ("copy", "add", "cat", "to_copy", "fill", "mul", "sum")
    for ev in sorted(evs, key=lambda e: -e.self_device_time_total):
        name = ev.key
        if any(t in name.lower() for t in interesting) and ev.self_device_time_total > 0:
            print(
                f"{name[:70]:70s} calls={ev.count:5d} "
                f"self_gpu={ev.self_device_time_total:10.0f}us"
            )

    # second pass WITH stacks: where do the add_/copy_ calls come from?
    with profile(
        activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
        with_stack=True,
    ) as prof2:
        step()
        torch.cuda.synchronize()
    print("\n=== add_/copy_ by stack ===")
    shown = 0
    for ev in sorted(
        prof2.key_averages(group_by_stack_n=6),
        key=lambda e: -e.self_device_time_total,
    ):
        if ev.key in ("aten::add_", "aten::copy_", "aten::add") and shown < 14:
            shown += 1
            stk = " <- ".join(
                s.split("/")[-1] for s in (ev.stack or [])[:6]
            )
            print(
                f"{ev.key:14s} calls={ev.count:4d} "
                f"gpu={ev.self_device_time_total:8.0f}us  {stk[:200]}"
            )


if __name__ == "__main__":
    main()
