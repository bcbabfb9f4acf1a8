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
