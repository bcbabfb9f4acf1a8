#!/usr/bin/env python3
"""Evaluation CLI (same flag surface as reference test.py:20-67).

python test.py --dataset=FT3D --weights=experiments/pvraft/checkpoints/best_checkpoint.params ...
Runs 32 GRU iterations per pair (reference test.py:120) and prints mean
EPE3D / Acc3DS / Acc3DR / Outlier.
"""

import argparse
import os

from pvraft_amd.cli import add_common_args
from pvraft_amd.engine.evaluator import evaluate


def parse_args():
    parser = argparse.ArgumentParser(description="Testing Argument")
    add_common_args(parser, training=False)
    parser.add_argument("--gamma", help="exponential weights", default=0.8, type=float)
    parser.add_argument("--dump_results", help="write result/<dataset>/<i>/{pc1,pc2,flow}.npy for visual.py", action="store_true")
    return parser.parse_args()


if __name__ == "__main__":
    args = parse_args()
    gpus = [g for g in str(args.gpus).split(",") if g != ""]
    if gpus and "RANK" not in os.environ:
        os.environ.setdefault("HIP_VISIBLE_DEVICES", gpus[0])
    results = evaluate(args)
    print(results)
