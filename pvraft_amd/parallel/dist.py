"""Distributed runtime: one process per GPU over RCCL/xGMI.

Replaces the reference's single-process nn.DataParallel (engine.py:63-64)
with the MI355X-idiomatic layout: torch.distributed with the "nccl" backend
(= RCCL on ROCm) over the node's xGMI links, one rank per GPU, gradients
reduced by an explicit bucketed all-reduce overlapped with backward.

Design notes:
* xGMI is 7 point-to-point links per GPU; the whole PV-RAFT model is ~5.5 M
  params (~22 MB fp32 grads), i.e. ~one bucket -- a single flattened
  all-reduce per step is near-optimal, so the bucket cap defaults to 32 MB
  and overlap matters only for the first buckets of larger models.
* Gradients accumulate directly into per-bucket flat buffers (p.grad is a
  view), so reduction needs no gather/scatter copies.
* On CPU (tests) the same code runs over gloo.
"""

from __future__ import annotations

import datetime
import os
from dataclasses import dataclass
from typing import List, Optional

import torch
import torch.distributed as dist


@dataclass
class DistInfo:
    rank: int = 0
    world_size: int = 1
    local_rank: int = 0
    device: torch.device = torch.device("cpu")

    @property
    def is_main(self) -> bool:
        return self.rank == 0

    @property
    def distributed(self) -> bool:
        return self.world_size > 1


def init_distributed(backend: Optional[str] = None, timeout_s: int = 600) -> DistInfo:
    """Initialise torch.distributed from torchrun-style env vars.

    Single-process when RANK/WORLD_SIZE are absent.  Backend defaults to
    nccl (RCCL) when CUDA/HIP devices are visible, else gloo.
    """
    if dist.is_initialized():
        return _info_from_env()
    if "RANK" not in os.environ or "WORLD_SIZE" not in os.environ:
        device = torch.device("cuda:0") if torch.cuda.is_available() else torch.device("cpu")
        if device.type == "cuda":
            torch.cuda.set_device(device)
        return DistInfo(device=device)

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29531")
    # host driver on this pool only supports dmabuf IPC; legacy-mode IPC
    # makes RCCL cross-process tensor sharing fail with
    # hipIpcGetMemHandle: invalid argument
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(backend=backend, timeout=datetime.timedelta(seconds=timeout_s))
    return _info_from_env()


def _info_from_env() -> DistInfo:
    rank = dist.get_rank()
    world = dist.get_world_size()
    local_rank = int(os.environ.get("LOCAL_RANK", rank % max(torch.cuda.device_count(), 1)))
    if torch.cuda.is_available():
        device = torch.device(f"cuda:{local_rank}")
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")
    return DistInfo(rank=rank, world_size=world, local_rank=local_rank, device=device)


def cleanup() -> None:
    if dist.is_initialized():
        dist.destroy_process_group()


def broadcast_module(module: torch.nn.Module, src: int = 0) -> None:
    """Make all ranks start from rank src's parameters/buffers."""
    if not (dist.is_initialized() and dist.get_world_size() > 1):
        return
    for t in list(module.parameters()) + list(module.buffers()):
        dist.broadcast(t.data, src=src)


def all_reduce_mean_(t: torch.Tensor) -> torch.Tensor:
    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
        t /= dist.get_world_size()
    return t


def all_reduce_sum_(t: torch.Tensor) -> torch.Tensor:
    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t


class _Bucket:
    __slots__ = ("params", "flat", "pending", "work")

    def __init__(self, params: List[torch.nn.Parameter], device: torch.device):
        self.params = params
        numel = sum(p.numel() for p in params)
        self.flat = torch.zeros(numel, dtype=params[0].dtype, device=device)
        offset = 0
        for p in params:
            p.grad = self.flat[offset : offset + p.numel()].view_as(p)
            offset += p.numel()
        self.pending = 0
        self.work = None


class GradReducer:
    """Bucketed gradient all-reduce overlapped with backward.

    Parameters' .grad tensors are views into per-bucket flat buffers.
    A post-accumulate-grad hook counts arrivals per bucket; when a bucket is
    complete its all-reduce (mean) is launched asynchronously (RCCL runs it
    on its own HIP stream, overlapping the rest of backward).
    ``finalize()`` waits for all outstanding reductions -- call it between
    loss.backward() and optimizer.step().  ``zero_grad()`` zeroes the flat
    buffers -- use it instead of optimizer.zero_grad().
    """

    def __init__(
        self,
        module: torch.nn.Module,
        bucket_cap_mb: float = 32.0,
        process_group=None,
    ):
        self.pg = process_group
        self.enabled = dist.is_initialized() and dist.get_world_size() > 1
        params = [p for p in module.parameters() if p.requires_grad]
        # reverse registration order approximates backward arrival order
        params = params[::-1]
        cap = int(bucket_cap_mb * 1024 * 1024)
        self.buckets: List[_Bucket] = []
        cur: List[torch.nn.Parameter] = []
        size = 0
        for p in params:
            bytes_ = p.numel() * p.element_size()
            if cur and size + bytes_ > cap:
                self.buckets.append(_Bucket(cur, p.device))
                cur, size = [], 0
            cur.append(p)
            size += bytes_
        if cur:
            self.buckets.append(_Bucket(cur, cur[0].device))

        self._param_bucket = {}
        for b in self.buckets:
            for p in b.params:
                self._param_bucket[p] = b
        self._hooks = [
            p.register_post_accumulate_grad_hook(self._on_grad) for p in params
        ]
        self._reset_pending()

    def _reset_pending(self):
        for b in self.buckets:
            b.pending = len(b.params)
            b.work = None

    hooks_enabled: bool = True

    def _on_grad(self, p: torch.nn.Parameter) -> None:
        if not self.hooks_enabled:
            return
        b = self._param_bucket[p]
        b.pending -= 1
        if b.pending == 0 and self.enabled:
            b.work = dist.all_reduce(b.flat, op=dist.ReduceOp.SUM, group=self.pg, async_op=True)

    def _flush_deferred(self) -> None:
        # deferred wgrad jobs must land in the flat buffers before any
        # remaining bucket reduction (buckets containing deferred params
        # never fire their hooks early, so ordering is safe)
        from pvraft_amd.model import pointwise

        pointwise.wgrad_flush()

    def reduce_all(self) -> None:
        """Eager mean-all-reduce of every bucket (hipGraph mode: the
        backward ran inside a captured graph, hooks were disabled)."""
        self._flush_deferred()
        if not self.enabled:
            return
        world = dist.get_world_size()
        works = [
            dist.all_reduce(b.flat, op=dist.ReduceOp.SUM, group=self.pg, async_op=True)
            for b in self.buckets
        ]
        for w, b in zip(works, self.buckets):
            w.wait()
            b.flat /= world

    def finalize(self) -> None:
        self._flush_deferred()
        world = dist.get_world_size() if self.enabled else 1
        for b in self.buckets:
            if b.work is None and self.enabled:
                # some params saw no gradient this step (e.g. frozen or
                # unused submodules): reduce the bucket now so ranks agree
                b.work = dist.all_reduce(b.flat, op=dist.ReduceOp.SUM, group=self.pg, async_op=True)
            if b.work is not None:
                b.work.wait()
                b.flat /= world
        self._reset_pending()

    def zero_grad(self) -> None:
        for b in self.buckets:
            b.flat.zero_()
        self._reset_pending()
        if self.buckets and self.buckets[0].flat.is_cuda:
            # arm the deferred batched-wgrad path for this step (the flat
            # views are the accumulation targets; flush in finalize)
            from pvraft_amd.model import pointwise

            pointwise.wgrad_defer_begin()
