"""hipGraph-captured training step.

The PV-RAFT step is fully static in shape (fixed point count, batch, GRU
iterations), so the whole forward + loss + backward region can be captured
once into a hipGraph (torch.cuda.CUDAGraph == hipGraph on ROCm) and
replayed each step -- collapsing the several thousand small kernel launches
of the 8-iteration GRU loop into one graph launch.

The distributed all-reduce and the optimizer step stay OUTSIDE the graph:
gradients land in the GradReducer's flat bucket buffers (p.grad are views),
so after replay() a single eager all-reduce + Adam step completes the step.
GradReducer's per-bucket hooks are bypassed in graph mode (reduce_after
handles it); buckets are zeroed inside the capture so replay is
self-contained.
"""

from __future__ import annotations

from typing import Callable, List, Optional

import torch

from pvraft_amd.parallel import GradReducer


class GraphedTrainStep:
    """Capture fn(batch)->loss (incl. backward) once; replay per step.

    fn must: zero grads, run forward, compute loss, call loss.backward().
    Static tensors referenced by fn (the batch buffers) must be filled
    in-place before each replay.
    """

    def __init__(self, fn: Callable[[], tuple], warmup: int = 3):
        self.fn = fn
        self.graph: Optional[torch.cuda.CUDAGraph] = None
        self.static_loss: Optional[torch.Tensor] = None
        self.static_final_flow: Optional[torch.Tensor] = None
        self._warmup = warmup

    def capture(self) -> None:
        # warmup on a side stream (allocator state, autotune, RCCL lazy init)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(self._warmup):
                self.fn()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.static_loss, self.static_final_flow = self.fn()

    def replay(self) -> torch.Tensor:
        self.graph.replay()
        return self.static_loss


def build_graphed_step(model, batch, num_iters: int, gamma: float,
                       reducer: GradReducer, amp: bool, warmup: int = 3,
                       loss_fn=None) -> GraphedTrainStep:
    """Standard stage-1 step: zero -> fwd(iters) -> sequence_loss -> bwd.

    ``loss_fn(flows, batch) -> loss`` overrides the default sequence loss
    (the refine stage uses compute_loss on a single flow).
    """
    from pvraft_amd.utils import sequence_loss

    def fn():
        from pvraft_amd.model import pointwise

        for b in reducer.buckets:
            b.flat.zero_()
        pointwise.wgrad_defer_begin()
        with torch.autocast("cuda", dtype=torch.bfloat16, enabled=amp):
            flows = model(batch["sequence"], num_iters=num_iters)
            if loss_fn is not None:
                loss = loss_fn(flows, batch)
            else:
                loss = sequence_loss(flows, batch, gamma=gamma)
        loss.backward()
        # deferred wgrad batch + stacked-weight scatter, recorded in the
        # graph (at replay the captured kernels rerun; the Python queue
        # stays empty)
        pointwise.wgrad_flush()
        final = flows[-1] if isinstance(flows, (list, tuple)) else flows
        return loss, final

    step = GraphedTrainStep(fn, warmup=warmup)
    step.capture()
    return step
