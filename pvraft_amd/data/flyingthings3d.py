"""FlyingThings3D (HPLFlowNet preprocessing) dataset.

Reference datasets/flyingthings3d_hplflownet.py: directories train/0* hold
19,640 samples split into train/val by 2,000 evenly spaced validation
indices (np.linspace(0, 19639, 2000)); test = val/0* directories (3,824
samples).  Each sample directory has pc1.npy / pc2.npy in camera space; x
and z signs are flipped on load; the clouds are point-aligned so
flow = pc2 - pc1 and the mask is all ones.

``strict_sizes=False`` relaxes the reference's hard dataset-size asserts
(flyingthings3d_hplflownet.py:58,71) for subset copies.
"""

from __future__ import annotations

import glob
import os

import numpy as np

from .base import SceneFlowDataset

FT3D_TRAIN_SIZE = 19640
FT3D_VAL_COUNT = 2000
FT3D_TEST_SIZE = 3824


class FT3D(SceneFlowDataset):
    def __init__(self, root_dir: str, nb_points: int, mode: str, strict_sizes: bool = True):
        super().__init__(nb_points)
        if mode not in ("train", "val", "test"):
            raise ValueError(f"Mode {mode} unknown")
        self.mode = mode
        self.root_dir = root_dir
        self.strict_sizes = strict_sizes
        self.filenames = self._file_list()

    def __len__(self):
        return len(self.filenames)

    def _file_list(self):
        pattern = "train/0*" if self.mode in ("train", "val") else "val/0*"
        filenames = np.sort(glob.glob(os.path.join(self.root_dir, pattern)))
        if self.mode in ("train", "val"):
            n = len(filenames)
            if self.strict_sizes and n != FT3D_TRAIN_SIZE:
                raise RuntimeError(f"Expected {FT3D_TRAIN_SIZE} FT3D training dirs, found {n}")
            # 2000 evenly spaced val indices at the reference size; scale
            # proportionally (>=1) for reduced copies (strict_sizes=False)
            n_val = FT3D_VAL_COUNT if n >= FT3D_TRAIN_SIZE else max(1, n * FT3D_VAL_COUNT // FT3D_TRAIN_SIZE)
            ind_val = set(np.linspace(0, max(n - 1, 0), min(n_val, n)).astype(int))
            if self.mode == "train":
                keep = sorted(set(range(n)) - ind_val)
            else:
                keep = sorted(ind_val)
            filenames = filenames[keep]
        elif self.strict_sizes and len(filenames) != FT3D_TEST_SIZE:
            raise RuntimeError(f"Expected {FT3D_TEST_SIZE} FT3D test dirs, found {len(filenames)}")
        return list(filenames)

    def load_sequence(self, idx: int):
        sequence = []
        for fname in ("pc1.npy", "pc2.npy"):
            pc = np.load(os.path.join(self.filenames[idx], fname))
            pc[..., 0] *= -1
            pc[..., -1] *= -1
            sequence.append(pc)
        ground_truth = [np.ones_like(sequence[0][:, 0:1]), sequence[1] - sequence[0]]
        return sequence, ground_truth
