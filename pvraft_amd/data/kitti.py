"""KITTI scene-flow dataset (HPLFlowNet preprocessing).

Reference datasets/kitti_hplflownet.py: walks the processed directory
(expects 200 leaf scene dirs with pc1.npy/pc2.npy), filters scenes through
the HPLFlowNet KITTI_mapping.txt (142 usable scenes), removes ground points
(y < -1.4 in both clouds) and points at z >= 35 m, and uses flow =
pc2 - pc1 with an all-ones mask.

The mapping file (third-party data originating from HPLFlowNet's KITTI
raw-drive index, same data the reference ships as
datasets/KITTI_mapping.txt) is bundled at ``pvraft_amd/data/KITTI_mapping.txt``
and used by default, so KITTI metrics always cover the 142-scene protocol.
A mapping file placed in the dataset root or passed via ``mapping_file``
overrides the bundled copy; ``mapping_file=""`` explicitly opts out and
evaluates all 200 scenes (non-comparable numbers -- opt-in only).
"""

from __future__ import annotations

import os

import numpy as np

from .base import SceneFlowDataset

KITTI_SCENES = 200
GROUND_Y = -1.4
MAX_DEPTH = 35.0
BUNDLED_MAPPING = os.path.join(os.path.dirname(__file__), "KITTI_mapping.txt")


class Kitti(SceneFlowDataset):
    def __init__(
        self,
        root_dir: str,
        nb_points: int,
        mapping_file: str = None,
        strict_sizes: bool = True,
    ):
        super().__init__(nb_points)
        self.root_dir = root_dir
        self.strict_sizes = strict_sizes
        self.paths = self._make_dataset(mapping_file)

    def __len__(self):
        return len(self.paths)

    def _make_dataset(self, mapping_file):
        root = os.path.realpath(os.path.expanduser(self.root_dir))
        useful_paths = [d for d, subdirs, _ in sorted(os.walk(root)) if len(subdirs) == 0]
        if self.strict_sizes and len(useful_paths) != KITTI_SCENES:
            raise RuntimeError(f"Expected {KITTI_SCENES} KITTI scene dirs, found {len(useful_paths)}")

        if mapping_file is None:
            cand = os.path.join(root, "KITTI_mapping.txt")
            mapping_file = cand if os.path.exists(cand) else BUNDLED_MAPPING
        if mapping_file == "":  # explicit opt-out: all 200 scenes
            return useful_paths

        with open(mapping_file) as fd:
            lines = [line.strip() for line in fd.readlines()]
        return [p for p in useful_paths if lines[int(os.path.split(p)[-1])] != ""]

    def load_sequence(self, idx: int):
        sequence = [
            np.load(os.path.join(self.paths[idx], "pc1.npy")),
            np.load(os.path.join(self.paths[idx], "pc2.npy")),
        ]
        not_ground = np.logical_not(
            np.logical_and(sequence[0][:, 1] < GROUND_Y, sequence[1][:, 1] < GROUND_Y)
        )
        sequence = [pc[not_ground] for pc in sequence]
        is_close = np.logical_and(sequence[0][:, 2] < MAX_DEPTH, sequence[1][:, 2] < MAX_DEPTH)
        sequence = [pc[is_close] for pc in sequence]
        ground_truth = [np.ones_like(sequence[0][:, 0:1]), sequence[1] - sequence[0]]
        return sequence, ground_truth
