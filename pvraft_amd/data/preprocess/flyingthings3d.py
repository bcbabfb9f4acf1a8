"""Offline FlyingThings3D-subset preprocessing CLI.

Produces the pc1.npy/pc2.npy per-sample directories consumed by
pvraft_amd.data.FT3D (HPLFlowNet layout).  Behaviour parity with reference
data_preprocess/process_flyingthings3d_subset.py: back-project the left
disparity map (pinhole f=-1050, cx=479.5, cy=269.5, baseline 1.0) to pc1,
back-project (disparity + disparity_change) at flow-displaced pixels to
pc2, drop pixels occluded in either map, optionally keep only points
nearer than 35 m (z > -35 in this camera convention).

    python -m pvraft_amd.data.preprocess.flyingthings3d \
        --raw_data_path <raw> --save_path <out> --only_save_near_pts
"""

from __future__ import annotations

import argparse
import os
import os.path as osp
import sys
from concurrent.futures import ProcessPoolExecutor

import numpy as np

from .io import read_flo, read_pfm, read_png

F_LEN = -1050.0
CX = 479.5
CY = 269.5
BASELINE = 1.0
NEAR_Z = -35.0


def backproject(disparity: np.ndarray, flow: np.ndarray = None) -> np.ndarray:
    """Disparity (H, W) -> camera-space cloud (H, W, 3).

    With ``flow`` given, the pixel grid is displaced by the optical flow
    before back-projection (the next-frame cloud of a point-aligned pair).
    """
    h, w = disparity.shape
    px = np.tile(np.arange(w, dtype=np.float32)[None, :], (h, 1))
    py = np.tile(np.arange(h, dtype=np.float32)[:, None], (1, w))
    if flow is not None:
        px = px + flow[..., 0]
        py = py + flow[..., 1]
    depth = -F_LEN * BASELINE / disparity
    x = -(px - CX) / disparity
    y = (py - CY) / disparity
    return np.stack([x, y, depth], axis=-1).astype(np.float32)


def process_one_sample(root_path: str, save_path: str, split: str, fname: str,
                       save_near: bool = False) -> int:
    disp1 = read_pfm(osp.join(root_path, split, "disparity", "left", fname + ".pfm"))
    disp1_occ = read_png(osp.join(root_path, split, "disparity_occlusions", "left", fname + ".png"))
    disp_change = read_pfm(
        osp.join(root_path, split, "disparity_change", "left", "into_future", fname + ".pfm")
    )
    flow = read_flo(osp.join(root_path, split, "flow", "left", "into_future", fname + ".flo"))
    flow_occ = read_png(
        osp.join(root_path, split, "flow_occlusions", "left", "into_future", fname + ".png")
    )

    pc1 = backproject(disp1)
    pc2 = backproject(disp1 + disp_change, flow=flow)

    valid = np.logical_and(disp1_occ == 0, flow_occ == 0)
    pc1, pc2 = pc1[valid], pc2[valid]
    if save_near:
        near = np.logical_and(pc1[..., -1] > NEAR_Z, pc2[..., -1] > NEAR_Z)
        pc1, pc2 = pc1[near], pc2[near]

    out_dir = osp.join(save_path, split, fname)
    os.makedirs(out_dir, exist_ok=True)
    np.save(osp.join(out_dir, "pc1.npy"), pc1)
    np.save(osp.join(out_dir, "pc2.npy"), pc2)
    return pc1.shape[0]


def main(argv=None):
    parser = argparse.ArgumentParser()
    parser.add_argument("--raw_data_path", type=str, required=True)
    parser.add_argument("--save_path", type=str, required=True)
    parser.add_argument("--only_save_near_pts", dest="save_near", action="store_true")
    parser.add_argument("--workers", type=int, default=4)
    args = parser.parse_args(argv)

    jobs = []
    for split in ("train", "val"):
        listing_dir = osp.join(args.raw_data_path, split, "disparity_change", "left", "into_future")
        for item in sorted(os.listdir(listing_dir)):
            jobs.append((split, item.split(".")[0]))

    with ProcessPoolExecutor(max_workers=args.workers) as pool:
        futs = [
            pool.submit(process_one_sample, args.raw_data_path, args.save_path, s, f, args.save_near)
            for s, f in jobs
        ]
        for (s, f), fut in zip(jobs, futs):
            try:
                fut.result()
            except Exception as e:
                print(f"error processing {s}/{f}: {e}", file=sys.stderr)
    print(f"Processed {len(jobs)} samples")


if __name__ == "__main__":
    main()
