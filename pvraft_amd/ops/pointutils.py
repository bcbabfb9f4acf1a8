"""General point-cloud utilities.

Capability parity with reference model/pointconv.py (square_distance,
knn_point -- dead code there, kept as public utilities here since they are
the building blocks users of the reference import from that module).
"""

from __future__ import annotations

import torch
from torch import Tensor


def square_distance(src: Tensor, dst: Tensor) -> Tensor:
    """Pairwise squared euclidean distance.

    src: (B, N, C), dst: (B, M, C) -> (B, N, M).
    """
    return (
        (src * src).sum(-1, keepdim=True)
        + (dst * dst).sum(-1).unsqueeze(1)
        - 2.0 * torch.bmm(src, dst.transpose(1, 2))
    )


def knn_point(nsample: int, xyz: Tensor, new_xyz: Tensor) -> Tensor:
    """Indices of the nsample nearest points in xyz for each query in new_xyz.

    xyz: (B, N, C), new_xyz: (B, S, C) -> (B, S, nsample) int64.
    """
    d = square_distance(new_xyz, xyz)
    return d.topk(nsample, dim=-1, largest=False).indices
