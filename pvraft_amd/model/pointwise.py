"""1x1 convolutions as plain GEMMs.

Every convolution in PV-RAFT has kernel size 1 (reference gconv.py:26-33,
corr.py:15-29, update.py:11-29,60-66), i.e. it IS a GEMM over the flattened
spatial dim.  MIOpen's conv path wraps it in NHWC transposes and igemm
kernels (visible in rocprof as batched_transpose_* + igemm_wrw_*); routing
through torch.matmul hits hipBLASLt directly and lets autograd produce
plain GEMM backward.  Subclasses keep nn.Conv1d/Conv2d parameter layout so
state dicts stay interchangeable with the reference.
"""

from __future__ import annotations

import torch
import torch.nn as nn
from torch import Tensor


class PwConv1d(nn.Conv1d):
    """nn.Conv1d(k=1) with a matmul forward."""

    def forward(self, x: Tensor) -> Tensor:
        # x (B, Cin, N) -> (B, Cout, N)
        w = self.weight.squeeze(-1)  # (Cout, Cin)
        y = torch.matmul(w, x)
        if self.bias is not None:
            y = y + self.bias.view(1, -1, 1)
        return y


class PwConv2d(nn.Conv2d):
    """nn.Conv2d(k=1) with a matmul forward over flattened spatial dims."""

    def forward(self, x: Tensor) -> Tensor:
        B, C, H, W = x.shape
        w = self.weight.view(self.out_channels, C)
        y = torch.matmul(w, x.reshape(B, C, H * W))
        if self.bias is not None:
            y = y + self.bias.view(1, -1, 1)
        return y.view(B, self.out_channels, H, W)
