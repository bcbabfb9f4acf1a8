"""Inference wrapper: hipGraph-captured scene-flow prediction.

Evaluation/serving runs the backbone 32 GRU iterations per pair
(reference test.py:120) with fixed shapes (bs x max_points), so the whole
no-grad forward is captured once and replayed per sample -- the same
launch-collapse as the training step, applied to serving.

    pred = Predictor(model, points=8192, batch=1, iters=32)
    flow = pred(xyz1, xyz2)   # (B, N, 3)
"""

from __future__ import annotations

from typing import Optional

import torch


class Predictor:
    def __init__(self, model, points: int, batch: int = 1, iters: int = 32,
                 amp: bool = False, use_graph: bool = True):
        self.model = model.eval()
        self.iters = iters
        self.amp = amp
        self.device = next(model.parameters()).device
        # ROOT-CAUSED TOOLCHAIN BUG (round 2, profiles/README.md): hipGraph
        # replay of the inference forward at batch >= 5 faults in the ROCm
        # graph-pool page mapping -- the PURE-ATEN forward (PVRAFT_REF_OPS=1,
        # zero custom kernels) reproduces it identically ("write access to a
        # read-only page" / aperture violation on first replay), eager is
        # clean at every batch size under strict per-tensor allocation, and
        # batch <= 4 graphs are exercised throughout training/eval.
        # Repros: scripts/stress_graph_bs.py, scripts/graph_region_bisect.py.
        # Permanent guard: serve batch >= 5 eagerly on this toolchain.
        self.use_graph = use_graph and self.device.type == "cuda" and batch <= 4
        self._graph: Optional[torch.cuda.CUDAGraph] = None
        self._in1 = torch.zeros(batch, points, 3, device=self.device)
        self._in2 = torch.zeros(batch, points, 3, device=self.device)
        self._flows = None
        self.last_flows = None

    def _forward(self):
        with torch.no_grad(), torch.autocast("cuda", dtype=torch.bfloat16, enabled=self.amp):
            flows = self.model([self._in1, self._in2], num_iters=self.iters)
        if not isinstance(flows, (list, tuple)):
            flows = [flows]
        return [f.float() for f in flows]

    def _capture(self) -> None:
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self._forward()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        self._graph = torch.cuda.CUDAGraph()
        # capture on the warmup stream (see engine/graphed.py: nodes
        # pinned to another stream record cross-stream work that races
        # at replay)
        with torch.cuda.graph(self._graph, stream=s):
            self._flows = self._forward()

    @torch.no_grad()
    def __call__(self, xyz1: torch.Tensor, xyz2: torch.Tensor) -> torch.Tensor:
        """Returns the FINAL flow (B, N, 3); ``last_flows`` holds the whole
        per-iteration list (static graph outputs -- consume before the next
        call) for sequence-loss evaluation."""
        if not self.use_graph:
            self._in1, self._in2 = xyz1.to(self.device), xyz2.to(self.device)
            self.last_flows = self._forward()
            return self.last_flows[-1]
        if self._graph is None:
            self._in1.copy_(xyz1)
            self._in2.copy_(xyz2)
            self._capture()
        self._in1.copy_(xyz1, non_blocking=True)
        self._in2.copy_(xyz2, non_blocking=True)
        self._graph.replay()
        self.last_flows = self._flows
        return self._flows[-1].clone()
