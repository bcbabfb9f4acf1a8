"""PointNet++/DGCNN-style edge convolution.

Capability parity with reference model/flot/gconv.py (SetConv): gather the k
neighbours' features, subtract the centre, concatenate the 3-d edge offset,
then Conv2d 1x1 -> GroupNorm(8) -> LeakyReLU(0.1) -> max over k ->
(Conv1d -> GN -> LReLU) x 2.  Channel rule gconv.py:21-24: mid = out//2 when
in is odd else (in+out)//2; all convs bias-free (gconv.py:26-33).

Parameter names (fc1/gn1/fc2/gn2/fc3/gn3) match the reference so its
checkpoints load directly.  The gather+concat runs through the fused HIP op
on GPU (ops.gather_edge_concat) instead of the flattened fancy-indexing of
gconv.py:64-66.
"""

from __future__ import annotations

import torch
import torch.nn as nn
from torch import Tensor

from pvraft_amd import ops
from .graph import Graph
from .norm import FusedGroupNorm


class SetConv(nn.Module):
    def __init__(self, in_ch: int, out_ch: int):
        super().__init__()
        mid = out_ch // 2 if in_ch % 2 != 0 else (out_ch + in_ch) // 2
        self.fc1 = nn.Conv2d(in_ch + 3, mid, 1, bias=False)
        self.gn1 = FusedGroupNorm(8, mid, act="lrelu", slope=0.1)
        self.fc2 = nn.Conv1d(mid, out_ch, 1, bias=False)
        self.gn2 = FusedGroupNorm(8, out_ch, act="lrelu", slope=0.1)
        self.fc3 = nn.Conv1d(out_ch, out_ch, 1, bias=False)
        self.gn3 = FusedGroupNorm(8, out_ch, act="lrelu", slope=0.1)

    def forward(self, feats: Tensor, graph: Graph) -> Tensor:
        """feats (B, N, C) -> (B, N, out_ch).  LeakyReLU(0.1) is fused into
        each GroupNorm (reference order gconv.py:71-83: conv -> GN -> lrelu)."""
        x = ops.gather_edge_concat(feats, graph.idx, graph.xyz, csr=graph.csr())  # B, C+3, K, N
        x = self.gn1(self.fc1(x))
        x = x.max(dim=2)[0]  # max-pool over neighbours -> B, mid, N
        x = self.gn2(self.fc2(x))
        x = self.gn3(self.fc3(x))
        return x.transpose(1, 2)
