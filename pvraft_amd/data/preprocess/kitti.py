"""Offline KITTI scene-flow 2015 preprocessing CLI.

Produces the per-frame pc1.npy/pc2.npy directories consumed by
pvraft_amd.data.Kitti.  Behaviour parity with reference
data_preprocess/process_kitti.py: disp_occ_0/disp_occ_1 uint16 PNGs ->
depth via the P_rect_02 calibration (baseline 0.54 m), pc2's pixel grid
displaced by the occluded optical flow (KITTI encoding (v - 2^15)/64),
points valid where both disparities and the flow are valid.  The
reference's per-pixel python loop (process_kitti.py:56-72) is vectorised.

    python -m pvraft_amd.data.preprocess.kitti \
        --raw_data_path <kitti_sf> --calib_path <calib_cam_to_cam> --save_path <out>
"""

from __future__ import annotations

import argparse
import os
import os.path as osp
import sys
from concurrent.futures import ProcessPoolExecutor

import numpy as np

from .io import read_png

BASELINE = 0.54
N_FRAMES = 200


def load_disp(path: str):
    arr = read_png(path)
    valid = arr > 0
    disp = arr.astype(np.float32) / 256.0
    disp[~valid] = -1.0
    return disp, valid


def load_flow(path: str):
    arr = read_png(path)  # (H, W, 3) uint16: u, v, valid
    valid = arr[..., 2] == 1
    flow = (arr[..., :2].astype(np.float32) - 2 ** 15) / 64.0
    return flow, valid


def disp_to_depth(disp: np.ndarray, valid: np.ndarray, focal_px: float) -> np.ndarray:
    depth = focal_px * BASELINE / (disp + 1e-5)
    depth[~valid] = -1.0
    return depth


def parse_p_rect_02(calib_path: str) -> np.ndarray:
    with open(calib_path) as fd:
        rows = [line for line in fd if line.startswith("P_rect_02")]
    if len(rows) != 1:
        raise ValueError(f"{calib_path}: expected exactly one P_rect_02 line")
    vals = np.array([float(v) for v in rows[0].split()[1:]], dtype=np.float32)
    P = vals.reshape(3, 4)
    if P[0, 0] != P[1, 1] or P[0, 1] != 0 or P[1, 0] != 0:
        raise ValueError(f"{calib_path}: unexpected P_rect_02 structure")
    return P


def pixel_to_xyz(depth: np.ndarray, P: np.ndarray, px=None, py=None) -> np.ndarray:
    """Back-project a depth map through a rectified projection matrix.

    x/y signs are flipped to match the HPLFlowNet camera convention
    (reference kitti_utils.py:27).
    """
    f = P[0, 0]
    h, w = depth.shape
    if px is None:
        px = np.tile(np.arange(w, dtype=np.float32)[None, :], (h, 1))
    if py is None:
        py = np.tile(np.arange(h, dtype=np.float32)[:, None], (1, w))
    x = (px * (depth + P[2, 3]) - (P[0, 2] * depth + P[0, 3])) / f
    y = (py * (depth + P[2, 3]) - (P[1, 2] * depth + P[1, 3])) / f
    pc = np.stack([-x, -y, depth], axis=-1)
    return pc.astype(np.float32)


def process_one_frame(data_root: str, calib_root: str, save_path: str, idx: int) -> int:
    sidx = f"{idx:06d}"
    P = parse_p_rect_02(osp.join(calib_root, sidx + ".txt"))
    focal = float(P[0, 0])

    disp1, valid1 = load_disp(osp.join(data_root, "training", "disp_occ_0", sidx + "_10.png"))
    disp2, valid2 = load_disp(osp.join(data_root, "training", "disp_occ_1", sidx + "_10.png"))
    depth1 = disp_to_depth(disp1, valid1, focal)
    depth2 = disp_to_depth(disp2, valid2, focal)

    flow, valid_flow = load_flow(osp.join(data_root, "training", "flow_occ", sidx + "_10.png"))
    valid_disp = np.logical_and(valid1, valid2)
    valid = np.logical_and(valid_disp, valid_flow)

    h, w = depth1.shape
    base_px = np.tile(np.arange(w, dtype=np.float32)[None, :], (h, 1))
    base_py = np.tile(np.arange(h, dtype=np.float32)[:, None], (1, w))
    px2 = np.where(valid, base_px + flow[..., 0], 0.0).astype(np.float32)
    py2 = np.where(valid, base_py + flow[..., 1], 0.0).astype(np.float32)

    pc1 = pixel_to_xyz(depth1, P)
    pc2 = pixel_to_xyz(depth2, P, px=px2, py=py2)

    out_dir = osp.join(save_path, sidx)
    os.makedirs(out_dir, exist_ok=True)
    np.save(osp.join(out_dir, "pc1.npy"), pc1[valid])
    np.save(osp.join(out_dir, "pc2.npy"), pc2[valid])
    return int(valid.sum())


def main(argv=None):
    parser = argparse.ArgumentParser()
    parser.add_argument("--raw_data_path", type=str, required=True)
    parser.add_argument("--calib_path", type=str, required=True)
    parser.add_argument("--save_path", type=str, required=True)
    parser.add_argument("--workers", type=int, default=4)
    args = parser.parse_args(argv)

    with ProcessPoolExecutor(max_workers=args.workers) as pool:
        futs = {
            i: pool.submit(process_one_frame, args.raw_data_path, args.calib_path, args.save_path, i)
            for i in range(N_FRAMES)
        }
        for i, fut in futs.items():
            try:
                fut.result()
            except Exception as e:
                print(f"error processing frame {i}: {e}", file=sys.stderr)
    print(f"Processed {N_FRAMES} frames")


if __name__ == "__main__":
    main()
