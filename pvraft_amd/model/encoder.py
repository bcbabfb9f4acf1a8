"""Point-cloud feature encoder (reference model/extractor.py: FlotEncoder).

Three stacked SetConvs 3 -> 32 -> 64 -> 128 sharing one kNN graph (k=32)
built on the input cloud.  Returns features as (B, 128, N) plus the graph
(the graph is reused by the flow head / refine head on pc1).
Attribute names feat_conv1..3 match reference state dicts.
"""

from __future__ import annotations

import torch.nn as nn
from torch import Tensor

from .graph import Graph
from .setconv import SetConv


class PointEncoder(nn.Module):
    def __init__(self, num_neighbors: int = 32, width: int = 32):
        super().__init__()
        self.num_neighbors = num_neighbors
        self.feat_conv1 = SetConv(3, width)
        self.feat_conv2 = SetConv(width, 2 * width)
        self.feat_conv3 = SetConv(2 * width, 4 * width)

    def forward(self, pc: Tensor, graph: Graph = None):
        """pc (B, N, 3) -> ((B, 4*width, N), Graph).

        ``graph`` may be passed in when the caller already built the kNN
        graph of this cloud (the model shares pc1's graph between the
        feature and context encoders -- identical numerics, one build
        fewer than the reference, which rebuilds it per encoder).
        """
        if graph is None:
            graph = Graph.build(pc, self.num_neighbors)
        from pvraft_amd import ops

        x = self.feat_conv1(pc, graph)
        x = self.feat_conv2(x, graph)
        x = self.feat_conv3(x, graph)
        return ops.transpose_last2(x), graph
