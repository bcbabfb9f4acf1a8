"""Checkpoint format and naming (reference tools/utils.py:6-29)."""

import argparse
import os

import pytest
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


def test_train_state_roundtrip(tmp_path):
    import torch.nn as nn

    args = make_args(tmp_path)
    model = nn.Linear(4, 4)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    sched = torch.optim.lr_scheduler.CosineAnnealingLR(opt, T_max=100)
    model(torch.randn(2, 4)).sum().backward()
    opt.step()
    sched.step()

    from pvraft_amd.utils import load_train_state, save_train_state

    path = save_train_state(args, epoch=2, optimizer=opt, lr_scheduler=sched, best_val_epe=0.5)
    assert path.endswith("train_state.pt")

    opt2 = torch.optim.Adam(model.parameters(), lr=1e-3)
    sched2 = torch.optim.lr_scheduler.CosineAnnealingLR(opt2, T_max=100)
    state = load_train_state(args, opt2, sched2)
    assert state["epoch"] == 2 and state["best_val_epe"] == 0.5
    assert sched2.last_epoch == sched.last_epoch
    s1 = opt.state_dict()["state"]
    s2 = opt2.state_dict()["state"]
    for k in s1:
        assert torch.equal(s1[k]["exp_avg"], s2[k]["exp_avg"])


@pytest.mark.parametrize("refine", [False, True])
def test_state_dict_parity_with_reference_model(refine):
    """Key-for-key, shape-for-shape state-dict parity with the ACTUAL
    reference model classes (checkpoints interchangeable both ways).
    Runs only where the reference repo is mounted (read-only import)."""
    import argparse
    import importlib
    import sys

    if not os.path.isdir("/root/reference/model"):
        pytest.skip("reference repo not available")
    sys.path.insert(0, "/root/reference")
    try:
        try:
            mod = importlib.import_module(
                "model.RAFTSceneFlowRefine" if refine else "model.RAFTSceneFlow"
            )
        except ImportError as e:  # missing reference-era deps (torch_scatter)
            pytest.skip(f"reference model unimportable: {e}")
        cls = getattr(mod, "RSF_refine" if refine else "RSF")
        args = argparse.Namespace(corr_levels=3, base_scales=0.25, truncate_k=64)
        ref_sd = {k: tuple(v.shape) for k, v in cls(args).state_dict().items()}
    finally:
        sys.path.remove("/root/reference")

    from pvraft_amd.model import PVRaft, PVRaftRefine

    ours = (PVRaftRefine if refine else PVRaft)(
        corr_levels=3, base_scales=0.25, truncate_k=64
    )
    our_sd = {k: tuple(v.shape) for k, v in ours.state_dict().items()}
    assert set(ref_sd) == set(our_sd), (
        sorted(set(ref_sd) - set(our_sd)), sorted(set(our_sd) - set(ref_sd)))
    mismatch = {k: (ref_sd[k], our_sd[k]) for k in ref_sd if ref_sd[k] != our_sd[k]}
    assert not mismatch, mismatch
