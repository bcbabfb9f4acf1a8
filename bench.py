#!/usr/bin/env python3
"""Flagship training benchmark: PV-RAFT stage-1 train step throughput.

Measures the BASELINE.json metric -- train samples(pairs)/sec at 8192
points, 8 GRU iterations -- on synthetic FT3D-shaped pairs with
random-init weights (no network => no real dataset), bf16 autocast
compute, full training step in the timed region (forward 8 iters +
sequence loss + backward + gradient all-reduce + Adam step).

Single node, one rank per GPU over RCCL:
  python bench.py --gpus 1 --steps 16 --warmup 4
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 bench.py --gpus 8 --steps 16 --warmup 4

Weak scaling: per-GPU batch is fixed (default 2 pairs/GPU); the reported
value is the whole-job aggregate pairs/sec (max step time over ranks).
Baseline: ~1.85 pairs/s on 2x RTX 2080 Ti (BASELINE.md derived from
reference README.md:62-64).
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

BASELINE_PAIRS_PER_S = 1.85


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=16)
    p.add_argument("--warmup", type=int, default=4)
    p.add_argument("--batch", type=int, default=2, help="pairs per GPU (weak scaling)")
    p.add_argument("--points", type=int, default=8192)
    p.add_argument("--iters", type=int, default=8)
    p.add_argument("--truncate_k", type=int, default=512)
    p.add_argument("--no-amp", dest="amp", action="store_false")
    p.add_argument("--no-graph", dest="graph", action="store_false",
                   help="disable hipGraph capture of the train step")
    p.add_argument("--device", type=str, default=None, help="force device (cpu for plumbing tests)")
    p.add_argument("--master_port", type=int, default=0,
                   help="rendezvous port for the self-launch (default: free port)")
    p.add_argument("--refine", action="store_true",
                   help="stage-2 benchmark: frozen backbone + refine head, "
                        "32 GRU iterations (BASELINE config 4)")
    args = p.parse_args()
    if args.refine and args.iters == 8:
        args.iters = 32  # reference run.sh stage-2 convention
    return args


def main():
    args = parse_args()

    # self-launch one rank per GPU if asked for N>1 outside torchrun
    if args.gpus > 1 and "RANK" not in os.environ:
        from pvraft_amd.cli import free_port

        port = args.master_port or free_port()
        cmd = [
            sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
            f"--nproc-per-node={args.gpus}", "--master-addr=127.0.0.1",
            f"--master-port={port}", os.path.abspath(__file__),
        ] + sys.argv[1:]
        os.execvpe(cmd[0], cmd, dict(os.environ))

    from pvraft_amd.data import synthetic_batch
    from pvraft_amd.model import PVRaft, PVRaftRefine
    from pvraft_amd.parallel import GradReducer, broadcast_module, init_distributed
    from pvraft_amd.utils import compute_loss, sequence_loss

    info = init_distributed()
    device = torch.device(args.device) if args.device else info.device
    cuda = device.type == "cuda"
    amp = args.amp and cuda

    torch.manual_seed(1234 + info.rank)
    if args.refine:
        model = PVRaftRefine(truncate_k=args.truncate_k).to(device)
        model.freeze_backbone()
        loss_fn = lambda flows, b: compute_loss(flows, b)  # noqa: E731
    else:
        model = PVRaft(truncate_k=args.truncate_k).to(device)
        loss_fn = None
    broadcast_module(model)
    reducer = GradReducer(model)
    optimizer = torch.optim.Adam(
        [p for p in model.parameters() if p.requires_grad], lr=1e-3)
    batch = synthetic_batch(args.batch, args.points, device=device, seed=100 + info.rank)

    model.train()
    use_graph = args.graph and cuda
    if use_graph:
        try:
            from pvraft_amd.engine.graphed import build_graphed_step

            reducer.hooks_enabled = False
            graphed = build_graphed_step(
                model, batch, num_iters=args.iters, gamma=0.8, reducer=reducer,
                amp=amp, loss_fn=loss_fn,
            )
        except Exception as e:  # pragma: no cover - capture-env specific
            print(f"[bench] hipGraph capture failed ({e!r}); falling back to eager", file=sys.stderr)
            use_graph = False
            reducer.hooks_enabled = True
    if use_graph:

        def step():
            loss = graphed.replay()
            reducer.reduce_all()
            optimizer.step()
            return loss

    else:

        def step():
            reducer.zero_grad()
            with torch.autocast("cuda", dtype=torch.bfloat16, enabled=amp):
                est_flow = model(batch["sequence"], num_iters=args.iters)
                loss = (loss_fn(est_flow, batch) if loss_fn is not None
                        else sequence_loss(est_flow, batch, gamma=0.8))
            loss.backward()
            reducer.finalize()
            optimizer.step()
            return loss

    for _ in range(args.warmup):
        step()

    if info.distributed:
        torch.distributed.barrier()
    if cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if cuda:
        torch.cuda.synchronize()
    if info.distributed:
        torch.distributed.barrier()
    elapsed = time.perf_counter() - t0

    # max over ranks
    el = torch.tensor([elapsed], dtype=torch.float64, device=device if cuda else "cpu")
    if info.distributed:
        torch.distributed.all_reduce(el, op=torch.distributed.ReduceOp.MAX)
    elapsed = el.item()

    n_gpus = info.world_size if cuda else args.gpus
    global_batch = args.batch * info.world_size
    pairs_per_s = global_batch * args.steps / elapsed
    # stage-2 baseline: 17,640 pairs x 10 epochs / 38 h on 2x2080Ti
    # (reference README.md:66-71) = 1.29 pairs/s
    baseline = 1.29 if args.refine else BASELINE_PAIRS_PER_S
    if info.is_main:
        print(
            json.dumps(
                {
                    "metric": "refine_train_pairs_per_sec" if args.refine else "train_pairs_per_sec",
                    "value": pairs_per_s,
                    "unit": "pairs/s",
                    "n_gpus": n_gpus,
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": elapsed / args.steps * 1e3,
                    "higher_is_better": True,
                    "scaling": "weak",
                    "vs_baseline": pairs_per_s / baseline,
                    "dtype": "bf16" if amp else "fp32",
                    "data": "synthetic",
                    "config": {
                        "model": "PV-RAFT (RSF_refine stage 2)" if args.refine else "PV-RAFT (RSF stage 1)",
                        "global_batch": global_batch,
                        "points": args.points,
                        "gru_iters": args.iters,
                        "truncate_k": args.truncate_k,
                        "parallelism": f"dp{info.world_size}",
                        "hipgraph": use_graph,
                    },
                }
            )
        )
    if info.distributed:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
