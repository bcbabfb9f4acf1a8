"""Nearest-neighbour graph over a point cloud.

MI355X-native redesign of reference model/flot/graph.py: instead of a full
B x N x N distance matrix + argsort (graph.py:53-60) and flattened global
edge indices (graph.py:77-79), the graph is just the (B, N, k) neighbour
index tensor; edge features (relative coordinates) are produced on the fly
by the fused gather kernel (ops.gather_edge_concat), never stored.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional, Tuple

import torch
from torch import Tensor

from pvraft_amd import ops


@dataclass
class Graph:
    idx: Tensor  # (B, N, k) int64 neighbour indices (self included)
    xyz: Tensor  # (B, N, 3) the cloud the graph was built on
    _csr: Optional[Tuple[Tensor, Tensor]] = field(default=None, repr=False)
    _idx32: Optional[Tensor] = field(default=None, repr=False)

    @property
    def idx32(self) -> Tensor:
        """int32 copy of idx, cached (shared by every SetConv on the graph)."""
        if self._idx32 is None:
            self._idx32 = self.idx.to(torch.int32).contiguous()
        return self._idx32

    @property
    def k(self) -> int:
        return self.idx.shape[-1]

    @staticmethod
    def build(xyz: Tensor, k: int) -> "Graph":
        return Graph(idx=ops.knn_graph(xyz, k), xyz=xyz)

    def csr(self) -> Optional[Tuple[Tensor, Tensor, Tensor]]:
        """Inverse adjacency in CSR form, for the deterministic SetConv
        backward: (order (B, N*k) int32 = edge ids sorted by target node,
        offsets (B, N+1) int32, order_n (B, N*k) int32 = source point of
        each ordered edge, order_j (B, N*k) uint8 = its neighbour slot, so
        walkers need no id decomposition).
        Built lazily once per graph (GPU only) and shared by every
        SetConv/FlowHead call on this graph.
        """
        if not self.idx.is_cuda:
            return None
        if self._csr is None:
            B, N, k = self.idx.shape
            # edge ids in (j, n) order: id = j*N + n, matching the
            # (B, C, K*N) -> (B, K*N, C) transpose of the gradient tensor
            flat = self.idx.permute(0, 2, 1).reshape(B, k * N)
            order = flat.argsort(dim=1)
            targets = flat.gather(1, order)
            bounds = torch.arange(N + 1, device=flat.device).expand(B, N + 1).contiguous()
            offsets = torch.searchsorted(targets, bounds, side="left")
            self._csr = (
                order.to(torch.int32).contiguous(),
                offsets.to(torch.int32).contiguous(),
                (order % N).to(torch.int32).contiguous(),
                (order // N).to(torch.uint8).contiguous(),
            )
        return self._csr
