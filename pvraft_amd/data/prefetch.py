"""Async host-to-device prefetching.

Wraps a DataLoader: while the model computes step t, batch t+1 is copied to
the GPU on a dedicated HIP stream (from pinned host memory), so the H2D
copy never sits on the critical path.  The reference has no overlap at all
(engine.py:133 copies synchronously inside the step).
"""

from __future__ import annotations

from typing import Iterable, Iterator, Optional

import torch

from .batch import Batch


class CudaPrefetcher:
    def __init__(self, loader: Iterable, device: torch.device):
        self.loader = loader
        self.device = device
        self.stream: Optional[torch.cuda.Stream] = (
            torch.cuda.Stream(device) if device.type == "cuda" else None
        )

    def __len__(self):
        return len(self.loader)

    def __iter__(self) -> Iterator[Batch]:
        if self.stream is None:
            for batch in self.loader:
                yield batch.to(self.device)
            return

        it = iter(self.loader)
        next_batch = self._preload(it)
        while next_batch is not None:
            torch.cuda.current_stream(self.device).wait_stream(self.stream)
            batch = next_batch
            # tensors were copied on self.stream; record them on the compute
            # stream so the caching allocator doesn't reuse too early
            for key in batch.data:
                for t in batch.data[key]:
                    t.record_stream(torch.cuda.current_stream(self.device))
            next_batch = self._preload(it)
            yield batch

    def _preload(self, it) -> Optional[Batch]:
        try:
            batch = next(it)
        except StopIteration:
            return None
        with torch.cuda.stream(self.stream):
            return batch.to(self.device, non_blocking=True)
