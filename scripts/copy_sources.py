"""Pinpoint the Python call sites that launch aten::copy_ / _to_copy /
aten::add during the flagship forward (TorchDispatchMode + stack walk).

Backward-engine ops carry no Python frames -- those report as
'<backward/C++>' -- but every forward-side cast/contiguous/cat shows its
model file:line, which is what the fusion work needs.  Run on the GPU box:
    python scripts/copy_sources.py
"""
import sys
import traceback
from collections import Counter

import torch
from torch.utils._python_dispatch import TorchDispatchMode

sys.path.insert(0, ".")

from pvraft_amd.data import synthetic_batch  # noqa: E402
from pvraft_amd.model import PVRaft  # noqa: E402
from pvraft_amd.parallel import GradReducer  # noqa: E402
from pvraft_amd.utils import sequence_loss  # noqa: E402

WATCH = ("copy_", "_to_copy", "add.Tensor", "add_.Tensor", "cat")


class CopyTracer(TorchDispatchMode):
    def __init__(self):
        super().__init__()
        self.sites = Counter()

    def __torch_dispatch__(self, func, types, args=(), kwargs=None):
        name = str(func)
        if any(w in name for w in WATCH):
            site = "<backward/C++>"
            for fr in reversed(traceback.extract_stack()):
                fn = fr.filename
                if "pvraft_amd" in fn or fn.endswith(("bench.py", "copy_sources.py")):
                    if "copy_sources" in fn:
                        continue
                    site = f"{fn.split('pvraft_amd/')[-1]}:{fr.lineno}"
                    break
            numel = 0
            if args and isinstance(args[0], torch.Tensor):
                numel = args[0].numel()
            self.sites[(name.split(".")[0], site, numel)] += 1
        return func(*args, **(kwargs or {}))


def main():
    device = torch.device("cuda")
    torch.manual_seed(1234)
    model = PVRaft(truncate_k=512).to(device)
    reducer = GradReducer(model)
    batch = synthetic_batch(2, 8192, device=device, seed=100)
    model.train()

    def step():
        reducer.zero_grad()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            est = model(batch["sequence"], num_iters=8)
            loss = sequence_loss(est, batch, gamma=0.8)
        loss.backward()
        reducer.finalize()

    step()  # warm caches / registries so steady-state counts are honest

    tracer = CopyTracer()
    with tracer:
        step()

    print(f"{'op':10s} {'count':>5s} {'numel':>9s}  site")
    for (op, site, numel), c in tracer.sites.most_common(40):
        print(f"{op:10s} {c:5d} {numel:9d}  {site}")


if __name__ == "__main__":
    main()
