"""Flow refinement head (reference model/refine.py: FlotRefine).

Three SetConvs 3 -> 32 -> 64 -> 128 applied to the flow field over pc1's
graph, a Linear 128 -> 3, residual added to the input flow.
Attribute names match reference state dicts.
"""

from __future__ import annotations

import torch.nn as nn
from torch import Tensor

from .graph import Graph
from .setconv import SetConv


class RefineHead(nn.Module):
    def __init__(self, width: int = 32):
        super().__init__()
        self.ref_conv1 = SetConv(3, width)
        self.ref_conv2 = SetConv(width, 2 * width)
        self.ref_conv3 = SetConv(2 * width, 4 * width)
        self.fc = nn.Linear(4 * width, 3)

    def forward(self, flow: Tensor, graph: Graph) -> Tensor:
        """flow (B, N, 3) over pc1's graph -> refined flow (B, N, 3)."""
        x = self.ref_conv1(flow, graph)
        x = self.ref_conv2(x, graph)
        x = self.ref_conv3(x, graph)
        return flow + self.fc(x)
