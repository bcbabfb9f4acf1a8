"""Checkpointing with the reference on-disk format (tools/utils.py:6-29).

Format: ``{'epoch': int, 'state_dict': <model state dict>}`` via torch.save.
File names: ``last_checkpoint.params`` every epoch, ``{epoch:03d}.params``
every ``checkpoint_interval`` epochs, ``best_checkpoint.params`` on best
validation EPE.  Under distributed training only rank 0 writes (the
reference's DataParallel ``.module`` unwrap at utils.py:19-23 becomes a
rank guard here).
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.nn as nn


def _unwrap(model: nn.Module) -> nn.Module:
    return model.module if hasattr(model, "module") else model


def checkpoint_dir(root: str, exp_path: str) -> str:
    return os.path.join(root, "experiments", exp_path, "checkpoints")


def save_checkpoint(
    model: nn.Module,
    args,
    epoch: int,
    mode: str = "train",
    rank: int = 0,
) -> Optional[str]:
    """Returns the written path (rank 0) or None (other ranks)."""
    if rank != 0:
        return None
    if mode == "train":
        if epoch % args.checkpoint_interval != 0:
            name = "last_checkpoint.params"
        else:
            name = f"{epoch:03d}.params"
    else:
        name = "best_checkpoint.params"
    ckpt_dir = checkpoint_dir(args.root, args.exp_path)
    os.makedirs(ckpt_dir, exist_ok=True)
    path = os.path.join(ckpt_dir, name)
    torch.save({"epoch": epoch, "state_dict": _unwrap(model).state_dict()}, path)
    return path


def load_checkpoint(path: str, model: nn.Module, strict: bool = True) -> int:
    """Load a reference-format checkpoint; returns the stored epoch.

    Accepts either a direct file path or an experiment name whose
    best_checkpoint.params should be used (reference engine.py:100-108
    semantics).
    """
    ckpt = torch.load(path, map_location="cpu", weights_only=True)
    state = ckpt.get("state_dict", ckpt)
    _unwrap(model).load_state_dict(state, strict=strict)
    return int(ckpt.get("epoch", 0))
