"""Batch collation (reference datasets/generic.py:6-66).

Concatenates per-sample tensors along the batch dimension for the keys
"sequence" ([pc1 (B,N,3), pc2 (B,M,3)]) and "ground_truth"
([mask (B,N,1), flow (B,N,3)]).  Supports .to(device) and .pin_memory so it
can be used directly as a DataLoader collate_fn.
"""

from __future__ import annotations

from typing import Dict, List

import torch


class Batch:
    KEYS = ("sequence", "ground_truth")

    def __init__(self, samples: List[Dict]):
        self.data = {}
        for key in self.KEYS:
            self.data[key] = [
                torch.cat([s[key][i] for s in samples], dim=0) for i in range(2)
            ]

    def __getitem__(self, key: str):
        return self.data[key]

    def to(self, *args, **kwargs) -> "Batch":
        for key in self.data:
            self.data[key] = [d.to(*args, **kwargs) for d in self.data[key]]
        return self

    def pin_memory(self) -> "Batch":
        for key in self.data:
            self.data[key] = [d.pin_memory() for d in self.data[key]]
        return self

    @property
    def batch_size(self) -> int:
        return self.data["sequence"][0].shape[0]
