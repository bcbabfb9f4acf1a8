"""Nearest-neighbour graph over a point cloud.

MI355X-native redesign of reference model/flot/graph.py: instead of a full
B x N x N distance matrix + argsort (graph.py:53-60) and flattened global
edge indices (graph.py:77-79), the graph is just the (B, N, k) neighbour
index tensor; edge features (relative coordinates) are produced on the fly
by the fused gather kernel (ops.gather_edge_concat), never stored.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
from torch import Tensor

from pvraft_amd import ops


@dataclass
class Graph:
    idx: Tensor  # (B, N, k) int64 neighbour indices (self included)
    xyz: Tensor  # (B, N, 3) the cloud the graph was built on

    @property
    def k(self) -> int:
        return self.idx.shape[-1]

    @staticmethod
    def build(xyz: Tensor, k: int) -> "Graph":
        return Graph(idx=ops.knn_graph(xyz, k), xyz=xyz)
