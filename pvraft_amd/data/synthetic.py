"""Synthetic scene-flow pairs for benchmarking and tests.

Random point clouds with a smooth synthetic flow field, shaped exactly like
the FT3D samples (pc1/pc2 point-aligned, mask all ones, flow = pc2 - pc1).
Used when no real dataset is on disk (this environment has no network) --
bench.py and the plumbing tests run on these.
"""

from __future__ import annotations

import numpy as np
import torch

from .base import SceneFlowDataset
from .batch import Batch


class SyntheticSceneFlow(SceneFlowDataset):
    def __init__(self, nb_points: int, length: int = 256, seed: int = 0, extent: float = 10.0):
        super().__init__(nb_points)
        self.length = length
        self.seed = seed
        self.extent = extent

    def __len__(self):
        return self.length

    def load_sequence(self, idx: int):
        rng = np.random.default_rng(self.seed * 1_000_003 + idx)
        n = self.nb_points
        pc1 = (rng.random((n, 3), dtype=np.float32) - 0.5) * self.extent
        # smooth flow: global translation + small rotation + noise
        t = rng.normal(0.0, 0.5, size=(1, 3)).astype(np.float32)
        ang = rng.normal(0.0, 0.02)
        c, s = np.cos(ang, dtype=np.float32), np.sin(ang, dtype=np.float32)
        rot = np.array([[c, -s, 0], [s, c, 0], [0, 0, 1]], dtype=np.float32)
        pc2 = pc1 @ rot.T + t + rng.normal(0.0, 0.01, size=(n, 3)).astype(np.float32)
        ground_truth = [np.ones_like(pc1[:, 0:1]), pc2 - pc1]
        return [pc1, pc2], ground_truth


def synthetic_batch(
    batch_size: int, nb_points: int, device="cpu", seed: int = 0
) -> Batch:
    ds = SyntheticSceneFlow(nb_points, length=batch_size, seed=seed)
    return Batch([ds[i] for i in range(batch_size)]).to(device)
