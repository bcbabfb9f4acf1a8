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

    ``pre`` (optional) runs EAGERLY before every replay (and once before
    capture): work that must stay outside the graph -- here the Morton
    point relabeling, whose in-capture argsort temp allocations shift the
    graph pool layout enough to re-trigger the ROCm pool page-mapping
    fault (the bs>=5 toolchain bug; see profiles/README.md).
    """

    def __init__(self, fn: Callable[[], tuple], warmup: int = 3,
                 pre: Optional[Callable[[], None]] = None):
        self.fn = fn
        self.pre = pre
        self.graph: Optional[torch.cuda.CUDAGraph] = None
        self.static_loss: Optional[torch.Tensor] = None
        self.static_final_flow: Optional[torch.Tensor] = None
        self._warmup = warmup

    def capture(self) -> None:
        if self.pre is not None:
            self.pre()
        # warmup on a side stream (allocator state, autotune, RCCL lazy init)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(self._warmup):
                self.fn()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()

        # capture on the SAME stream the warmup ran on: per-leaf
        # AccumulateGrad nodes are pinned to the stream of the first
        # backward that touched them (the warmup stream) and a capture on
        # a different stream records their writes as cross-stream work
        # that RACES at replay
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph, stream=s):
            self.static_loss, self.static_final_flow = self.fn()

    def replay(self) -> torch.Tensor:
        if self.pre is not None:
            self.pre()
        self.graph.replay()
        return self.static_loss


def build_graphed_step(model, batch, num_iters: int, gamma: float,
                       reducer: GradReducer, amp: bool, warmup: int = 3,
                       loss_fn=None) -> GraphedTrainStep:
    """Standard stage-1 step: zero -> fwd(iters) -> sequence_loss -> bwd.

    ``loss_fn(flows, batch) -> loss`` overrides the default sequence loss
    (the refine stage uses compute_loss on a single flow).
    """
    import os

    from pvraft_amd import ops
    from pvraft_amd.utils import sequence_loss

    # Morton relabeling runs OUTSIDE the capture (see GraphedTrainStep):
    # pre() permutes the batch into secondary static buffers each step and
    # the captured fn consumes those with the model's internal relabeling
    # off.  The loss is computed in permuted order (identical value: it is
    # a masked per-point mean and gt/mask are permuted identically); the
    # returned final flow is gathered back to the caller's point order
    # through a static inverse-permutation buffer.
    seq = batch["sequence"]
    use_morton = (
        seq[0].is_cuda
        and ops.hip_available()
        and num_iters >= 12  # pre-phase sorts amortise past ~12 iterations
        and os.environ.get("PVRAFT_REF_OPS", "0") != "1"
        and os.environ.get("PVRAFT_NO_MORTON", "0") != "1"
    )
    if use_morton:
        gt = batch["ground_truth"]
        data_p = {
            "sequence": [torch.empty_like(t) for t in seq],
            "ground_truth": [torch.empty_like(t) for t in gt],
        }
        inv_s = torch.empty(seq[0].shape[:2], device=seq[0].device,
                            dtype=torch.int64)

        class _View:
            data = data_p

            def __getitem__(self, k):
                return data_p[k]

        batch_use = _View()

        def pre():
            with torch.no_grad():
                perm1, inv1 = ops.morton_order(seq[0])
                perm2, _ = ops.morton_order(seq[1], need_inv=False)
                inv_s.copy_(inv1)
                g1 = perm1.unsqueeze(-1)
                torch.gather(seq[0], 1, g1.expand(-1, -1, seq[0].shape[-1]),
                             out=data_p["sequence"][0])
                torch.gather(seq[1], 1,
                             perm2.unsqueeze(-1).expand(-1, -1, seq[1].shape[-1]),
                             out=data_p["sequence"][1])
                for dst, src in zip(data_p["ground_truth"], gt):
                    torch.gather(src, 1, g1.expand(-1, -1, src.shape[-1]),
                                 out=dst)

    else:
        batch_use = batch
        pre = None
        inv_s = None

    def fn():
        from pvraft_amd.model import pointwise

        for b in reducer.buckets:
            b.flat.zero_()
        pointwise.wgrad_defer_begin()
        with torch.autocast("cuda", dtype=torch.bfloat16, enabled=amp):
            flows = model(batch_use["sequence"], num_iters=num_iters,
                          morton=not use_morton)
            if loss_fn is not None:
                loss = loss_fn(flows, batch_use)
            else:
                loss = sequence_loss(flows, batch_use, gamma=gamma)
        loss.backward()
        # deferred wgrad batch + stacked-weight scatter, recorded in the
        # graph (at replay the captured kernels rerun; the Python queue
        # stays empty)
        pointwise.wgrad_flush()
        final = flows[-1] if isinstance(flows, (list, tuple)) else flows
        if inv_s is not None:
            final = final.gather(
                1, inv_s.unsqueeze(-1).expand(-1, -1, final.shape[-1]))
        return loss, final

    step = GraphedTrainStep(fn, warmup=warmup, pre=pre)
    step.capture()
    return step
