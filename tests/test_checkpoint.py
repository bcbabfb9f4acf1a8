"""Checkpoint format and naming (reference tools/utils.py:6-29)."""

import argparse
import os

import torch

from pvraft_amd.model import PVRaft
from pvraft_amd.utils import load_checkpoint, save_checkpoint


def make_args(tmp_path, interval=5):
    return argparse.Namespace(root=str(tmp_path), exp_path="exp", checkpoint_interval=interval)


def test_format_and_roundtrip(tmp_path):
    args = make_args(tmp_path)
    model = PVRaft(truncate_k=8)
    path = save_checkpoint(model, args, epoch=3, mode="train")
    assert path.endswith("last_checkpoint.params")
    ckpt = torch.load(path, map_location="cpu", weights_only=True)
    assert set(ckpt.keys()) == {"epoch", "state_dict"}
    assert ckpt["epoch"] == 3

    model2 = PVRaft(truncate_k=8)
    epoch = load_checkpoint(path, model2)
    assert epoch == 3
    for (n1, p1), (n2, p2) in zip(model.state_dict().items(), model2.state_dict().items()):
        assert n1 == n2 and torch.equal(p1, p2)


def test_interval_and_best_names(tmp_path):
    args = make_args(tmp_path)
    model = PVRaft(truncate_k=8)
    assert save_checkpoint(model, args, epoch=5, mode="train").endswith("005.params")
    assert save_checkpoint(model, args, epoch=6, mode="train").endswith("last_checkpoint.params")
    assert save_checkpoint(model, args, epoch=6, mode="best").endswith("best_checkpoint.params")


def test_nonzero_rank_writes_nothing(tmp_path):
    args = make_args(tmp_path)
    model = PVRaft(truncate_k=8)
    assert save_checkpoint(model, args, epoch=1, mode="train", rank=1) is None
    assert not os.path.exists(os.path.join(str(tmp_path), "experiments", "exp", "checkpoints"))
