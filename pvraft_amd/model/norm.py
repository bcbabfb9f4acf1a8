"""GroupNorm module backed by the CDNA4 kernel (pvraft_amd/csrc/group_norm.hip).

Subclasses nn.GroupNorm so parameter names/shapes (weight, bias) and
state-dict layout stay identical to the reference modules; only forward is
replaced.  ``act="lrelu"`` fuses the LeakyReLU that always follows GN in
SetConv (reference gconv.py:71-83) into the normalize pass.
"""

from __future__ import annotations

import torch.nn as nn
from torch import Tensor

from pvraft_amd import ops


class PReLUAny(nn.PReLU):
    """nn.PReLU that follows the input dtype.

    ATen's GroupNorm always upcasts to fp32 under autocast, so the
    reference never feeds PReLU bf16; FusedGroupNorm keeps bf16, and
    F.prelu refuses mixed dtypes -- cast the (fp32) slope to the input.
    """

    def forward(self, x: Tensor) -> Tensor:
        import torch.nn.functional as F

        return F.prelu(x, self.weight.to(x.dtype))


class FusedGroupNorm(nn.GroupNorm):
    def __init__(self, num_groups: int, num_channels: int, act: str = "none",
                 slope: float = 0.1, **kw):
        super().__init__(num_groups, num_channels, **kw)
        self.act = act
        self.slope = slope

    def forward(self, x: Tensor) -> Tensor:
        return ops.group_norm_act(
            x, self.num_groups, self.weight, self.bias, self.eps, self.act, self.slope
        )
