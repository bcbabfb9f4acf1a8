#!/usr/bin/env python3
"""Visualise a predicted scene flow (capability of reference visual.py,
which renders pc1/pc2/pc1+flow with mayavi; here matplotlib 3D -> PNG,
since mayavi is not in this environment).

Reads <root>/result/<dataset>/<index>/{pc1,pc2,flow}.npy as written by
``test.py --dump_results`` and writes view.png alongside.

    python visual.py --index 0 --dataset FT3D --root ./
"""

import argparse
import os

import numpy as np


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--root", default="", type=str)
    parser.add_argument("--dataset", default="FT3D", type=str)
    parser.add_argument("--index", default=0, type=int)
    parser.add_argument("--point_size", default=0.6, type=float)
    parser.add_argument("--out", default=None, type=str)
    args = parser.parse_args()

    d = os.path.join(args.root, "result", args.dataset, str(args.index))
    pc1 = np.load(os.path.join(d, "pc1.npy")).reshape(-1, 3)
    pc2 = np.load(os.path.join(d, "pc2.npy")).reshape(-1, 3)
    flow = np.load(os.path.join(d, "flow.npy")).reshape(-1, 3)

    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    fig = plt.figure(figsize=(12, 9))
    ax = fig.add_subplot(111, projection="3d")
    ax.scatter(*pc1.T, s=args.point_size, c="b", label="pc1")
    ax.scatter(*pc2.T, s=args.point_size, c="r", label="pc2")
    warped = pc1 + flow
    ax.scatter(*warped.T, s=args.point_size, c="g", label="pc1 + flow")
    ax.legend()
    out = args.out or os.path.join(d, "view.png")
    fig.savefig(out, dpi=150, bbox_inches="tight")
    print(f"wrote {out}")


if __name__ == "__main__":
    main()
