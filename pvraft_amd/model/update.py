"""GRU update block (reference model/update.py).

MotionEncoder: corr (64) and flow (3) -> 64-d motion feature with the raw
flow appended (update.py:15-20).  ConvGRU: pointwise gates over B x C x N
(update.py:31-40).  FlowHead: Conv1d branch || SetConv branch on the context
graph, concat -> Conv1d 128->64 -> ReLU -> Conv1d 64->3 (update.py:57-72).
Attribute names match reference state dicts.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F
from torch import Tensor

from .graph import Graph
from .pointwise import PwConv1d
from .setconv import SetConv


class MotionEncoder(nn.Module):
    def __init__(self):
        super().__init__()
        self.conv_corr = PwConv1d(64, 64, 1)
        self.conv_flow = PwConv1d(3, 64, 1)
        self.conv = PwConv1d(64 + 64, 64 - 3, 1)

    def forward(self, flow: Tensor, corr: Tensor) -> Tensor:
        """flow (B, N, 3), corr (B, 64, N) -> (B, 64, N)."""
        flow_t = flow.transpose(1, 2).contiguous()
        cor = F.relu(self.conv_corr(corr))
        flo = F.relu(self.conv_flow(flow_t))
        out = F.relu(self.conv(torch.cat([cor, flo], dim=1)))
        return torch.cat([out, flow_t], dim=1)


class ConvGRU(nn.Module):
    def __init__(self, input_dim: int = 128, hidden_dim: int = 64):
        super().__init__()
        self.convz = PwConv1d(input_dim + hidden_dim, hidden_dim, 1)
        self.convr = PwConv1d(input_dim + hidden_dim, hidden_dim, 1)
        self.convq = PwConv1d(input_dim + hidden_dim, hidden_dim, 1)

    def forward(self, h: Tensor, x: Tensor) -> Tensor:
        hx = torch.cat([h, x], dim=1)
        z = torch.sigmoid(self.convz(hx))
        r = torch.sigmoid(self.convr(hx))
        q = torch.tanh(self.convq(torch.cat([r * h, x], dim=1)))
        return (1 - z) * h + z * q


class ConvRNN(nn.Module):
    """Vanilla tanh RNN alternative to ConvGRU (reference update.py:43-54;
    defined for parity -- the reference never instantiates it either)."""

    def __init__(self, input_dim: int = 128, hidden_dim: int = 64):
        super().__init__()
        self.convx = PwConv1d(input_dim, hidden_dim, 1)
        self.convh = PwConv1d(hidden_dim, hidden_dim, 1)

    def forward(self, h: Tensor, x: Tensor) -> Tensor:
        return torch.tanh(self.convx(x) + self.convh(h))


class FlowHead(nn.Module):
    def __init__(self, input_dim: int = 64):
        super().__init__()
        self.conv1 = PwConv1d(input_dim, 64, 1)
        self.setconv = SetConv(64, 64)
        self.out_conv = nn.Sequential(
            PwConv1d(128, 64, 1),
            nn.ReLU(),
            PwConv1d(64, 3, 1),
        )

    def forward(self, x: Tensor, graph: Graph) -> Tensor:
        out = self.conv1(x)
        out_set = self.setconv(x.transpose(1, 2).contiguous(), graph).transpose(1, 2).contiguous()
        return self.out_conv(torch.cat([out_set, out], dim=1))


class UpdateBlock(nn.Module):
    def __init__(self, input_dim: int = 128, hidden_dim: int = 64):
        super().__init__()
        self.motion_encoder = MotionEncoder()
        self.gru = ConvGRU(input_dim=input_dim, hidden_dim=hidden_dim)
        self.flow_head = FlowHead(input_dim=hidden_dim)

    def forward(self, net: Tensor, inp: Tensor, corr: Tensor, flow: Tensor, graph: Graph):
        motion = self.motion_encoder(flow, corr)
        net = self.gru(net, torch.cat([inp, motion], dim=1))
        delta_flow = self.flow_head(net, graph).transpose(1, 2).contiguous()
        return net, delta_flow
