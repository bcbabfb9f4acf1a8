"""Shared CLI argument surface (reference train.py:8-71 / test.py:20-67).

All reference flags are preserved; MI355X-native additions: --amp (bf16
autocast), --num_workers, --dataset SYNTH (synthetic pairs, no data on
disk needed).  --gpus keeps the reference semantics of selecting devices
but launches one process per GPU (torch.distributed over RCCL) instead of
DataParallel: when more than one GPU is requested and the process is not
already under torchrun, the CLI re-execs itself through
``torch.distributed.run`` on 127.0.0.1.
"""

from __future__ import annotations

import argparse
import os
import socket
import sys


def free_port() -> int:
    """Pick a currently-free TCP port on 127.0.0.1 for the rendezvous.

    Hardcoded ports (29531/29533 in round 1) collide when two multi-GPU
    jobs share a host; binding port 0 lets the kernel choose.
    """
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def add_common_args(parser: argparse.ArgumentParser, training: bool) -> None:
    parser.add_argument("--root", help="workspace path", default="", type=str)
    parser.add_argument("--exp_path", help="specified experiment log path", default=None, type=str)
    parser.add_argument(
        "--dataset", help="choose dataset from 'FT3D', 'KITTI' and 'SYNTH'", default="FT3D", type=str
    )
    parser.add_argument(
        "--max_points", help="maximum number of points sampled from a point cloud", default=8192, type=int
    )
    parser.add_argument("--corr_levels", help="number of correlation pyramid levels", default=3, type=int)
    parser.add_argument("--base_scales", help="voxelize base scale", default=0.25, type=float)
    parser.add_argument("--truncate_k", help="value of truncate_k in corr block", default=512, type=int)
    parser.add_argument("--iters", help="number of iterations in GRU module", default=8, type=int)
    parser.add_argument("--gpus", help="gpus used for training, e.g. '0,1'", default="0", type=str)
    parser.add_argument("--weights", help="checkpoint weights to be loaded", default=None, type=str)
    parser.add_argument("--refine", help="refine mode", action="store_true")
    parser.add_argument("--num_workers", help="dataloader workers per rank", default=8, type=int)
    parser.add_argument("--amp", help="bf16 autocast compute", action="store_true")
    parser.add_argument("--no_hipgraph", dest="hipgraph", help="disable hipGraph train-step capture", action="store_false")
    parser.add_argument("--synth_len", help="synthetic dataset length", default=256, type=int)
    parser.add_argument(
        "--master_port",
        help="rendezvous port for multi-GPU launch (default: pick a free port)",
        default=0,
        type=int,
    )
    if training:
        parser.add_argument("--gamma", help="exponential weights", default=0.8, type=float)
        parser.add_argument("--batch_size", help="global mini-batch size", default=1, type=int)
        parser.add_argument("--num_epochs", help="number of epochs for training", default=20, type=int)
        parser.add_argument("--checkpoint_interval", help="save checkpoint every N epoch", default=5, type=int)


def maybe_relaunch_distributed(args, script: str) -> bool:
    """Spawn one rank per requested GPU via torch.distributed.run.

    Returns True when this process performed the launch (caller should
    exit); False when we are already a worker (or single GPU / CPU).
    """
    gpus = [g for g in str(args.gpus).split(",") if g != ""]
    if "RANK" in os.environ or len(gpus) <= 1:
        # select the single requested device for the non-distributed path
        if len(gpus) == 1 and "RANK" not in os.environ:
            os.environ.setdefault("HIP_VISIBLE_DEVICES", gpus[0])
        return False
    os.environ["HIP_VISIBLE_DEVICES"] = ",".join(gpus)
    port = int(getattr(args, "master_port", 0)) or free_port()
    cmd = [
        sys.executable,
        "-m",
        "torch.distributed.run",
        "--nnodes=1",
        f"--nproc-per-node={len(gpus)}",
        "--master-addr=127.0.0.1",
        f"--master-port={port}",
        script,
    ] + [a for a in sys.argv[1:]]
    os.execvpe(cmd[0], cmd, os.environ)
    return True  # unreachable
