#!/usr/bin/env python3
"""Microbenchmarks for the custom kernels on their real model shapes."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from pvraft_amd import _C
import pvraft_amd.ops as ops


def timeit(label, fn, reps=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    import time

    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / reps * 1e6
    print(f"{label:58s} {dt:9.1f} us")
    return dt


def bench_wgrad():
    print("--- pw_wgrad vs einsum (bf16) ---")
    shapes = [(2, 96, 67, 262144), (2, 64, 4, 262144), (2, 128, 96, 8192),
              (2, 64, 64, 8192), (2, 128, 81, 8192), (2, 64, 3, 8192)]
    for B, Co, Ci, S in shapes:
        dy = torch.randn(B, Co, S, device="cuda", dtype=torch.bfloat16)
        x = torch.randn(B, Ci, S, device="cuda", dtype=torch.bfloat16)
        timeit(f"einsum   {Co}x{Ci} S={S}", lambda: torch.einsum("bos,bis->oi", dy, x))
        for sc in (0, 4, 16, 64, 256):
            timeit(f"pw_wgrad {Co}x{Ci} S={S} sc={sc}", lambda sc=sc: _C.pw_wgrad(dy, x, sc))


def bench_gn():
    print("--- group_norm fwd/bwd (bf16, act=lrelu) ---")
    shapes = [(2, 96, 32 * 8192), (2, 128, 8192), (2, 64, 32 * 8192), (2, 48, 32 * 8192)]
    for B, C, S in shapes:
        x = torch.randn(B, C, S, device="cuda", dtype=torch.bfloat16, requires_grad=True)
        w = torch.randn(C, device="cuda")
        b = torch.randn(C, device="cuda")
        timeit(f"gn fwd   C={C} S={S}", lambda: ops.group_norm_act(x.detach(), 8, w, b, 1e-5, "lrelu"))
        y = ops.group_norm_act(x, 8, w, b, 1e-5, "lrelu")
        g = torch.randn_like(y)
        timeit(f"gn fwd+bwd C={C} S={S}", lambda: torch.autograd.grad(
            ops.group_norm_act(x, 8, w, b, 1e-5, "lrelu"), x, g))
        tf = torch.nn.functional
        xf = x.detach().float()
        timeit(f"ATen gn fwd C={C} S={S} (f32)", lambda: tf.group_norm(xf, 8, w, b, 1e-5))


def bench_misc():
    print("--- knn_graph / voxel / knn_corr ---")
    xyz = torch.randn(2, 8192, 3, device="cuda")
    timeit("knn_graph N=8192 k=32", lambda: _C.knn_graph(xyz, 32))
    corr = torch.randn(2, 8192, 512, device="cuda")
    coords = torch.randn(2, 8192, 3, device="cuda")
    cxyz = coords.unsqueeze(2) + torch.randn(2, 8192, 512, 3, device="cuda") * 0.5
    timeit("voxel_corr fwd", lambda: _C.voxel_corr_fwd(corr, cxyz, coords, 0.25, 3, 3))
    timeit("knn_corr fwd", lambda: _C.knn_corr_fwd(corr, cxyz, coords, 32))


if __name__ == "__main__":
    bench_wgrad()
    bench_gn()
    bench_misc()
