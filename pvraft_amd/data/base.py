"""Scene-flow dataset base class (reference datasets/generic.py:69-215).

Each item: {"sequence": [pc1 (1,n,3), pc2 (1,m,3)],
            "ground_truth": [mask (1,n,1), flow (1,n,3)]}.
Point clouds are randomly subsampled to nb_points.  Samples with fewer
points than nb_points are skipped forward to the next index; unlike the
reference (generic.py:101-110, which can run off the end of the dataset)
the scan wraps around modulo len(self).
"""

from __future__ import annotations

import numpy as np
import torch
from torch.utils.data import Dataset


class SceneFlowDataset(Dataset):
    def __init__(self, nb_points: int):
        super().__init__()
        self.nb_points = nb_points

    def __getitem__(self, idx: int):
        for attempt in range(len(self)):
            data = self._load_item((idx + attempt) % len(self))
            n = data["sequence"][0].shape[1]
            m = data["sequence"][1].shape[1]
            if n == self.nb_points and m == self.nb_points:
                return data
        raise RuntimeError(
            f"No sample in {type(self).__name__} has >= {self.nb_points} points"
        )

    def _load_item(self, idx: int):
        sequence, ground_truth = self.subsample_points(*self.load_sequence(idx))
        sequence, ground_truth = self.to_torch(sequence, ground_truth)
        return {"sequence": sequence, "ground_truth": ground_truth}

    @staticmethod
    def to_torch(sequence, ground_truth):
        sequence = [torch.from_numpy(np.ascontiguousarray(s)).float().unsqueeze(0) for s in sequence]
        ground_truth = [torch.from_numpy(np.ascontiguousarray(g)).float().unsqueeze(0) for g in ground_truth]
        return sequence, ground_truth

    def subsample_points(self, sequence, ground_truth):
        ind1 = np.random.permutation(sequence[0].shape[0])[: self.nb_points]
        sequence[0] = sequence[0][ind1]
        ground_truth = [g[ind1] for g in ground_truth]
        ind2 = np.random.permutation(sequence[1].shape[0])[: self.nb_points]
        sequence[1] = sequence[1][ind2]
        return sequence, ground_truth

    def load_sequence(self, idx: int):
        """Return ([pc1 (N,3), pc2 (M,3)], [mask (N,1), flow (N,3)]) numpy arrays."""
        raise NotImplementedError
