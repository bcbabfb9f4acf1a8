"""Build driver for the in-tree HIP extension.

    python -m pvraft_amd.ops.build

Runs ``setup.py build_ext --inplace`` with PYTORCH_ROCM_ARCH=gfx950 so
hipcc cross-compiles for MI355X (works without a GPU present).
"""

from __future__ import annotations

import os
import subprocess
import sys

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def build(verbose: bool = True) -> None:
    env = dict(os.environ)
    env.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    env.setdefault("MAX_JOBS", str(min(os.cpu_count() or 4, 16)))
    cmd = [sys.executable, "setup.py", "build_ext", "--inplace"]
    proc = subprocess.run(
        cmd, cwd=REPO_ROOT, env=env, capture_output=not verbose, text=True
    )
    if proc.returncode != 0:
        out = "" if verbose else (proc.stdout or "") + (proc.stderr or "")
        raise RuntimeError(f"HIP extension build failed (rc={proc.returncode})\n{out[-4000:]}")


if __name__ == "__main__":
    build()
    print("pvraft_amd._C built in-tree")
