#!/usr/bin/env python3
"""Training CLI (same flag surface as reference train.py:8-71).

Stage 1:  python train.py --exp_path=pvraft --batch_size=2 --gpus=0,1 \
              --num_epochs=20 --max_points=8192 --iters=8 --root=./
Refine:   python train.py --refine --weights=pvraft --iters=32 ...

Multi-GPU runs launch one process per GPU over RCCL/xGMI (see
pvraft_amd/cli.py); epoch loop mirrors reference train.py:81-84.
"""

import argparse

from pvraft_amd.cli import add_common_args, maybe_relaunch_distributed
from pvraft_amd.engine import RefineTrainer, Trainer
from pvraft_amd.parallel import cleanup


def parse_args():
    parser = argparse.ArgumentParser(description="Training Argument")
    add_common_args(parser, training=True)
    return parser.parse_args()


def main(args):
    print(args)
    trainer = RefineTrainer(args) if args.refine else Trainer(args)
    for epoch in range(trainer.begin_epoch, args.num_epochs + 1):
        trainer.training(epoch)
        trainer.val_test(epoch, mode="val")
    trainer.val_test(mode="test")
    cleanup()


if __name__ == "__main__":
    args = parse_args()
    if args.exp_path is None:
        args.exp_path = "default"
    maybe_relaunch_distributed(args, __file__)
    main(args)
