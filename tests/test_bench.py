"""bench.py contract test (CPU plumbing config)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_emits_contract_json():
    out = subprocess.run(
        [
            sys.executable,
            os.path.join(REPO, "bench.py"),
            "--device", "cpu", "--points", "96", "--steps", "2", "--warmup", "1",
            "--batch", "1", "--truncate_k", "16", "--no-amp",
        ],
        capture_output=True,
        text=True,
        timeout=300,
        cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert lines, out.stdout
    rec = json.loads(lines[-1])
    for key in (
        "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
        "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
    ):
        assert key in rec, key
    assert rec["metric"] == "train_pairs_per_sec"
    assert rec["value"] > 0
    assert rec["steps"] == 2 and rec["warmup"] == 1
    assert rec["scaling"] == "weak"
    assert rec["data"] == "synthetic"
    assert rec["config"]["gru_iters"] == 8


def test_bench_two_ranks_cpu_gloo():
    """The distributed bench path (torchrun, gloo on CPU) emits aggregate
    whole-job throughput with n_gpus from the world size env."""
    env = dict(os.environ)
    env.pop("HIP_VISIBLE_DEVICES", None)
    env["CUDA_VISIBLE_DEVICES"] = ""  # force CPU so gloo is selected
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
            "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
            "--master-port", "29711", os.path.join(REPO, "bench.py"),
            "--device", "cpu", "--points", "64", "--steps", "2", "--warmup", "1",
            "--batch", "1", "--truncate_k", "16", "--no-amp", "--no-graph",
        ],
        capture_output=True, text=True, timeout=420, cwd=REPO, env=env,
    )
    assert out.returncode == 0, out.stderr[-3000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    rec = json.loads(lines[-1])
    assert rec["config"]["parallelism"] == "dp2"
    assert rec["config"]["global_batch"] == 2
