"""Per-parameter gradient check of the CAPTURED training step vs an eager
step on identical weights and data.  Pinpoints which parameters get wrong
gradients under full-step capture (graphed training was observed to stop
learning with the fused kNN branch enabled while the op in isolation
replays clean).

    python scripts/graph_step_gradcheck.py [--replays 3]
"""
import argparse
import copy
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from pvraft_amd.data import synthetic_batch
from pvraft_amd.engine.graphed import build_graphed_step
from pvraft_amd.model import PVRaft
from pvraft_amd.parallel import GradReducer
from pvraft_amd.utils import sequence_loss


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--replays", type=int, default=3)
    ap.add_argument("--points", type=int, default=8192)
    args = ap.parse_args()
    dev = torch.device("cuda:0")
    torch.manual_seed(1234)
    model = PVRaft(truncate_k=512).to(dev)
    sd0 = copy.deepcopy(model.state_dict())
    batch = synthetic_batch(2, args.points, device=dev, seed=100)
    reducer = GradReducer(model)
    reducer.hooks_enabled = False
    graphed = build_graphed_step(model, batch, num_iters=8, gamma=0.8,
                                 reducer=reducer, amp=True)

    for trial in range(args.replays):
        # fresh data into the static batch
        nb = synthetic_batch(2, args.points, device=dev, seed=500 + trial)
        for key in batch.data:
            for dst, src in zip(batch.data[key], nb.data[key]):
                dst.copy_(src)
        graphed.replay()
        torch.cuda.synchronize()
        ggrads = {n: p.grad.clone() for n, p in model.named_parameters()
                  if p.grad is not None}
        # determinism probe: replay AGAIN on the same data -- a race gives
        # different garbage, a deterministic bug gives identical values
        graphed.replay()
        torch.cuda.synchronize()
        kk = "corr_block.knn_conv.0.weight"
        g2 = dict(model.named_parameters())[kk].grad
        same = torch.equal(ggrads[kk], g2)
        print(f"  replay-determinism({kk}): {'identical' if same else 'DIFFERS'}"
              f" max1={ggrads[kk].abs().max().item():.3e}"
              f" max2={g2.abs().max().item():.3e}")

        # eager step, same weights, same data (weights were never updated)
        model.load_state_dict(sd0)
        reducer.zero_grad()
        from pvraft_amd.model import pointwise
        with torch.autocast("cuda", dtype=torch.bfloat16):
            flows = model(batch["sequence"], num_iters=8)
            loss = sequence_loss(flows, batch, gamma=0.8)
        loss.backward()
        reducer.finalize()
        torch.cuda.synchronize()

        bad = []
        for n, p in model.named_parameters():
            if p.grad is None or n not in ggrads:
                continue
            a, b = ggrads[n].float(), p.grad.float()
            scale = b.abs().max().item() + 1e-6
            d = (a - b).abs().max().item()
            cos = torch.nn.functional.cosine_similarity(
                a.flatten(), b.flatten(), dim=0).item() if b.numel() > 1 else 1.0
            if d > 0.10 * scale and cos < 0.97:
                bad.append((n, d, scale, cos))
        print(f"trial {trial}: loss(graph)={graphed.static_loss.item():.4f} "
              f"loss(eager)={loss.item():.4f} bad={len(bad)}")
        for n, d, sc, cos in bad[:25]:
            print(f"    {n}: maxdiff={d:.4f} scale={sc:.4f} cos={cos:.3f}")


if __name__ == "__main__":
    main()
