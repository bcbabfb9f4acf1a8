"""Replay-correctness check for the fused kNN-branch op (knn_gnmp):
capture fwd+bwd once, then refill the static inputs with NEW data and
compare every replay gradient against an eager run on the same data.

    python scripts/graph_kg_repro.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from pvraft_amd import ops


def main():
    torch.manual_seed(0)
    dev = "cuda:0"
    B, K, N, C, G = 2, 32, 8192, 64, 8
    raw = torch.randn(B, 4, K, N, device=dev, requires_grad=True)
    w = torch.randn(C, 4, 1, 1, device=dev).mul(0.2).requires_grad_(True)
    cb = torch.randn(C, device=dev).mul(0.1).requires_grad_(True)
    ga = torch.rand(C, device=dev).add(0.5).requires_grad_(True)
    be = torch.randn(C, device=dev).mul(0.1).requires_grad_(True)
    sl = torch.tensor([0.25], device=dev).requires_grad_(True)
    leaves = [raw, w, cb, ga, be, sl]
    go = torch.randn(B, C, N, device=dev)

    def fwd_bwd():
        for t in leaves:
            t.grad = None if t.grad is None else t.grad.detach().zero_()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            y = ops.knn_gnmp(raw, w, cb, G, ga, be, 1e-5, sl)
        y.backward(go.to(y.dtype))
        return y

    # make grads exist so the capture records accumulation adds
    for t in leaves:
        t.grad = torch.zeros_like(t)

    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(3):
            fwd_bwd()
    torch.cuda.current_stream().wait_stream(s)
    torch.cuda.synchronize()

    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        y_static = fwd_bwd()

    for trial in range(3):
        with torch.no_grad():
            raw.copy_(torch.randn(B, 4, K, N, device=dev))
            go.copy_(torch.randn(B, C, N, device=dev))
        g.replay()
        torch.cuda.synchronize()
        replay_grads = [t.grad.clone() for t in leaves]
        y_replay = y_static.clone()

        # eager reference on the same data
        y_eager = fwd_bwd()
        torch.cuda.synchronize()
        ok = torch.allclose(y_replay.float(), y_eager.float(), atol=1e-2, rtol=1e-2)
        print(f"trial {trial}: y match={ok} "
              f"maxdiff={(y_replay.float()-y_eager.float()).abs().max().item():.4f}")
        for name, rg, t in zip(["raw", "w", "cb", "ga", "be", "sl"],
                               replay_grads, leaves):
            d = (rg.float() - t.grad.float()).abs().max().item()
            m = t.grad.float().abs().max().item()
            print(f"    d{name}: maxdiff={d:.5f} (scale {m:.4f})"
                  f"{'  <-- MISMATCH' if d > 0.03 * (m + 1e-6) else ''}")


if __name__ == "__main__":
    main()
