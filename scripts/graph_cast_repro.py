"""Confirm the captured-refresh semantics: does a weight->mirror batched
copy recorded in a hipGraph re-read the UPDATED weights at replay?

    python scripts/graph_cast_repro.py

Tests ATen _foreach_copy_ vs the kernel-arg multi_cast (csrc/cast_pack.hip)
under capture: mutate the fp32 sources after capture, replay, check the
bf16 mirrors.  A refresh that freezes at capture values silently stops
graphed training from learning (weights served from stale mirrors).
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import pvraft_amd._C as C


def trial(name, refresh):
    torch.manual_seed(0)
    srcs = [torch.randn(64, 64, device="cuda") for _ in range(95)]
    dsts = [torch.empty(64, 64, device="cuda", dtype=torch.bfloat16)
            for _ in range(95)]
    # warmup on a side stream
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        refresh(srcs, dsts)
    torch.cuda.current_stream().wait_stream(s)
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        refresh(srcs, dsts)
    for t in srcs:
        t.fill_(3.25)
    g.replay()
    torch.cuda.synchronize()
    ok = all(torch.all(d == 3.25) for d in dsts)
    print(f"{name}: {'re-reads sources at replay (OK)' if ok else 'STALE at replay (BROKEN)'}")
    return ok


def main():
    trial("foreach_copy_", lambda s, d: torch._foreach_copy_(d, s))
    trial("multi_cast_bf16", lambda s, d: C.multi_cast_bf16(s, d))


if __name__ == "__main__":
    main()
