"""CLI smoke tests (subprocess, CPU, synthetic data): train.py one tiny
epoch, test.py eval with --dump_results, visual.py render."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run(args, cwd, timeout=420):
    out = subprocess.run([sys.executable] + args, capture_output=True, text=True, timeout=timeout, cwd=cwd)
    assert out.returncode == 0, f"{' '.join(args)}\n--- stdout\n{out.stdout[-1500:]}\n--- stderr\n{out.stderr[-3000:]}"
    return out


def common_flags(tmp_path):
    return [
        "--root", str(tmp_path), "--dataset", "SYNTH", "--max_points", "48",
        "--truncate_k", "16", "--iters", "2", "--synth_len", "4",
        "--num_workers", "0", "--gpus", "",
    ]


def test_train_and_test_cli(tmp_path):
    run(
        [os.path.join(REPO, "train.py"), "--exp_path", "cli_exp", "--batch_size", "2",
         "--num_epochs", "1"] + common_flags(tmp_path),
        cwd=REPO, timeout=600,
    )
    ckpt = os.path.join(str(tmp_path), "experiments", "cli_exp", "checkpoints", "best_checkpoint.params")
    assert os.path.exists(ckpt)

    run(
        [os.path.join(REPO, "test.py"), "--exp_path", "cli_exp", "--weights", ckpt,
         "--dump_results"] + common_flags(tmp_path),
        cwd=REPO, timeout=600,
    )
    dumped = os.path.join(str(tmp_path), "result", "SYNTH", "0")
    assert os.path.exists(os.path.join(dumped, "flow.npy"))

    run(
        [os.path.join(REPO, "visual.py"), "--root", str(tmp_path), "--dataset", "SYNTH", "--index", "0"],
        cwd=REPO, timeout=300,
    )
    assert os.path.exists(os.path.join(dumped, "view.png"))
