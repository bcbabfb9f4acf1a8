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


def save_train_state(
    args,
    epoch: int,
    optimizer,
    lr_scheduler=None,
    best_val_epe: float = float("inf"),
    rank: int = 0,
) -> Optional[str]:
    """Extension over the reference format: optimizer/scheduler state for
    exact resume (the reference drops Adam moments on resume -- SURVEY.md
    item 5.4).  Written beside the model checkpoints as train_state.pt;
    the model files themselves keep the reference layout untouched.
    """
    if rank != 0:
        return None
    ckpt_dir = checkpoint_dir(args.root, args.exp_path)
    os.makedirs(ckpt_dir, exist_ok=True)
    path = os.path.join(ckpt_dir, "train_state.pt")
    torch.save(
        {
            "epoch": epoch,
            "optimizer": optimizer.state_dict(),
            "lr_scheduler": lr_scheduler.state_dict() if lr_scheduler is not None else None,
            "best_val_epe": best_val_epe,
        },
        path,
    )
    return path


def load_train_state(args, optimizer, lr_scheduler=None) -> Optional[dict]:
    path = os.path.join(checkpoint_dir(args.root, args.exp_path), "train_state.pt")
    if not os.path.isfile(path):
        return None
    state = torch.load(path, map_location="cpu", weights_only=True)
    optimizer.load_state_dict(state["optimizer"])
    if lr_scheduler is not None and state.get("lr_scheduler") is not None:
        lr_scheduler.load_state_dict(state["lr_scheduler"])
    return state


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
