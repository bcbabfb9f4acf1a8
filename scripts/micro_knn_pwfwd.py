"""Microbenchmarks for knn_select and pw_fwd on their flagship shapes
(both measured well above their op-count estimates; run under rocprofv3
--pmc to see VALU utilisation / LDS conflicts)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import pvraft_amd._C as C


def timed(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    dev = torch.device("cuda:0")
    torch.manual_seed(0)

    # knn_select flagship: B=2, N=8192, k=32
    xyz = torch.randn(2, 8192, 3, device=dev)
    us = timed(lambda: C.knn_graph(xyz, 32))
    print(f"knn_graph (2,8192,k32): {us:.1f} us")

    # pw_fwd single-part conv shape (64,64) x (2,64,16384)
    w = torch.randn(64, 64, device=dev, dtype=torch.bfloat16)
    x = torch.randn(2, 64, 16384, device=dev, dtype=torch.bfloat16)
    b = torch.randn(64, device=dev)
    us = timed(lambda: C.pw_fwd([w], [x], b, None, 1))
    print(f"pw_fwd (64x64 S16384 relu): {us:.1f} us")
    us = timed(lambda: torch.bmm(w.unsqueeze(0).expand(2, -1, -1), x))
    print(f"bmm same shape:            {us:.1f} us")

    # pw_fwd 3-part GRU-m shape
    ws = [torch.randn(192, c, device=dev, dtype=torch.bfloat16) for c in (64, 61, 3)]
    xs = [torch.randn(2, c, 8192, device=dev, dtype=torch.bfloat16) for c in (64, 61, 3)]
    us = timed(lambda: C.pw_fwd(ws, xs, None, None, 0))
    print(f"pw_fwd 3-part (192 S8192): {us:.1f} us")

    # corr_topk flagship
    f1 = torch.randn(2, 8192, 128, device=dev, dtype=torch.bfloat16)
    f2 = torch.randn(2, 8192, 128, device=dev, dtype=torch.bfloat16)
    us = timed(lambda: C.corr_topk(f1, f2, 512), iters=10)
    print(f"corr_topk (2,8192,8192,128,K512): {us:.1f} us")


if __name__ == "__main__":
    main()
