"""End-to-end engine tests on CPU: one tiny epoch of stage-1 training on
synthetic data, validation, checkpointing, and the refine trainer."""

import argparse
import os

import pytest
import torch

from pvraft_amd.engine import RefineTrainer, Trainer


def make_args(tmp_path, **over):
    base = dict(
        root=str(tmp_path),
        exp_path="exp",
        dataset="SYNTH",
        max_points=48,
        corr_levels=3,
        base_scales=0.25,
        truncate_k=16,
        iters=2,
        gamma=0.8,
        batch_size=2,
        gpus="",
        num_epochs=2,
        weights=None,
        checkpoint_interval=5,
        refine=False,
        num_workers=0,
        amp=False,
        synth_len=4,
    )
    base.update(over)
    return argparse.Namespace(**base)


@pytest.fixture()
def patched_val_iters(monkeypatch):
    # validation runs 32 GRU iters by default (reference engine.py:198);
    # keep CPU tests fast
    monkeypatch.setattr("pvraft_amd.engine.trainer.VAL_ITERS", 2)


def test_stage1_epoch_and_val(tmp_path, patched_val_iters):
    args = make_args(tmp_path)
    trainer = Trainer(args)
    trainer.training(1)
    results = trainer.val_test(1, mode="val")
    assert results["epe"] >= 0 and results["loss"] >= 0
    ckpt = os.path.join(str(tmp_path), "experiments", "exp", "checkpoints", "last_checkpoint.params")
    assert os.path.exists(ckpt)
    # best checkpoint written on first val
    best = os.path.join(str(tmp_path), "experiments", "exp", "checkpoints", "best_checkpoint.params")
    assert os.path.exists(best)
    # scalar log written
    assert os.path.exists(os.path.join(str(tmp_path), "experiments", "exp", "scalars.jsonl"))


def test_resume_from_checkpoint(tmp_path, patched_val_iters):
    args = make_args(tmp_path)
    t1 = Trainer(args)
    t1.training(1)
    args2 = make_args(tmp_path, weights=os.path.join(
        str(tmp_path), "experiments", "exp", "checkpoints", "last_checkpoint.params"))
    t2 = Trainer(args2)
    assert t2.begin_epoch == 2
    sd1 = t1.model.state_dict()
    sd2 = t2.model.state_dict()
    for k in sd1:
        assert torch.equal(sd1[k], sd2[k])


def test_refine_trainer_step(tmp_path, patched_val_iters):
    args = make_args(tmp_path, refine=True, iters=2, exp_path="exp_refine")
    trainer = RefineTrainer(args)
    # backbone frozen: optimizer only sees refine params
    n_opt = sum(p.numel() for g in trainer.optimizer.param_groups for p in g["params"])
    n_refine = sum(p.numel() for n, p in trainer.model.named_parameters() if n.startswith("refine_block"))
    assert n_opt == n_refine
    trainer.training(1)
    results = trainer.val_test(1, mode="val")
    assert results["epe"] >= 0


def test_lr_schedule_near_constant(tmp_path, patched_val_iters):
    """Reference engine.py:58,168: cosine T_max = epochs*len(ds), stepped
    once per epoch -> effectively constant lr ~1e-3 (documented quirk)."""
    args = make_args(tmp_path, synth_len=2000, num_epochs=20)
    trainer = Trainer(args)
    lr0 = trainer.optimizer.param_groups[0]["lr"]
    trainer.lr_scheduler.step()
    lr1 = trainer.optimizer.param_groups[0]["lr"]
    assert lr0 == pytest.approx(1e-3)
    assert abs(lr1 - lr0) / lr0 < 0.01
