"""1x1 convolutions as plain GEMMs with split-K weight gradients.

Every convolution in PV-RAFT has kernel size 1 (reference gconv.py:26-33,
corr.py:15-29, update.py:11-29,60-66), i.e. it IS a GEMM over the flattened
spatial dim.  MIOpen's conv path wraps it in NHWC transposes and igemm
kernels; torch.matmul hits hipBLASLt directly.

The weight gradient of these layers is a (Cout x Cin) output with a
B*N ~ 16-65k reduction dim: hipBLASLt's heuristic picks a non-split-K
kernel (a handful of workgroups on 256 CUs, measured ~210 us for a ~5 us
problem), so _PwMatmul computes dW as a chunked batched GEMM (SPLITK
partial products summed) instead of relying on matmul autograd.

Subclasses keep nn.Conv1d/Conv2d parameter layout so state dicts stay
interchangeable with the reference.
"""

from __future__ import annotations

import os

import torch
import torch.nn as nn
from torch import Tensor

SPLITK = 16

# Persistent autocast weight/bias cast buffers.  Each fp32 parameter gets
# ONE bf16 mirror; PVRaft/PVRaftRefine call refresh_casts() at forward
# entry, which re-fills every registered mirror with a single
# _foreach_copy_ (a few multi-tensor launches) instead of ~95 individual
# ~4.5 us cast kernels per step (measured launch-bound inside the hipGraph
# replay).  Entries hold the source tensor so ids cannot be reused while
# an entry lives; a version check on cache hits still catches in-place
# updates between refreshes (direct sub-block calls outside
# PVRaft.forward).  Inside hipGraph capture the refresh runs at the top of
# the captured fn, so the recorded foreach-copy re-reads the fp32 weights
# on every replay.
_CASTS: dict = {}  # (id(src), dt) -> [src, buf, version]


def refresh_casts() -> None:
    if not _CASTS:
        return
    srcs = []
    dsts = []
    slow = []
    for key, ent in _CASTS.items():
        if isinstance(key[0], tuple) and not isinstance(key[0][1], tuple):
            # _cast_key-tagged derived stack (fresh storage each forward):
            # re-filled when its fresh tensor arrives, skip here
            continue
        # detach FRESH each refresh: a stored detached alias of a view
        # (e.g. weight.view(o, -1)) trips autograd's stale-view check
        # once the optimizer updates the base in place
        src = ent[0].detach()
        ent[2] = ent[0]._version
        if (
            src.is_cuda
            and src.is_contiguous()
            and src.dtype == torch.float32
            and ent[1].dtype == torch.bfloat16
        ):
            srcs.append(src)
            dsts.append(ent[1])
        else:
            slow.append((src, ent[1]))
    with torch.no_grad():
        if srcs:
            # capture-safe batched refresh: pointer tables ride kernel
            # arguments (csrc/cast_pack.hip), so a hipGraph replay
            # re-reads the live fp32 weights.  ATen _foreach_copy_ was
            # measured to FREEZE mirrors at capture values under replay
            # (its device pointer staging is not re-uploaded) -- graphed
            # training silently stopped learning.
            from pvraft_amd import _C

            _C.multi_cast_bf16(srcs, dsts)
        for src, dst in slow:
            dst.copy_(src)


def clear_step_cache() -> None:
    """Kept for direct callers/tests: drops all cast mirrors."""
    _CASTS.clear()


# ---------------------------------------------------------------------------
# Deferred batched weight gradients.
#
# Weight gradients of 1x1 convs are not consumed by the rest of backward
# (only dx is), so they can be DEFERRED: backward queues (dy, x, grad
# buffer) and one batched MFMA kernel at the end of backward computes all
# ~180 of them in a single launch, accumulating directly into each
# parameter's fp32 .grad (the GradReducer flat-buffer views).  This removes
# per-call kernel launches, per-call zero-fills and the autograd
# accumulate-add chain (~4 ms/step measured in round 1).
#
# Activation is explicit (GradReducer.zero_grad / the graphed step turn it
# on; finalize / the captured region flush) so arbitrary users of the ops
# keep plain autograd semantics.  Under hipGraph capture the queueing
# Python runs once; the batched kernel, the H2D descriptor copy (pinned,
# kept alive) and the post-callbacks are recorded and replay correctly.
# ---------------------------------------------------------------------------


class _DeferState:
    on = False
    jobs: list = []   # (dy, x, dw_flat_fp32, dbias_fp32_or_None)
    posts: list = []  # callbacks run after the batched kernel
    keep_capture: list = []  # descriptor buffers owned by live hipGraphs
    keep_eager: list = []    # recent eager descriptors (pinned-copy drain)


_DEFER = _DeferState()


def wgrad_defer_active() -> bool:
    """FORWARD-side check (wcache construction): grad mode reflects the
    caller there, so a no_grad inference forward skips the defer buffers.
    BACKWARD-side code must test ``_DEFER.on`` directly instead -- the
    engine runs Function.backward with grad mode DISABLED, and gating on
    is_grad_enabled() there silently reroutes weight grads through
    AccumulateGrad (see ops._gn_defer_targets)."""
    return _DEFER.on and torch.is_grad_enabled()


def wgrad_defer_begin() -> None:
    """Arm deferred-wgrad collection for the current step (requires every
    deferred parameter to already have a .grad buffer, e.g. GradReducer's
    flat views, zeroed at step start)."""
    _DEFER.on = True
    # drop leftovers from an aborted step (stale tensors must not flush
    # into this step's gradients); old eager descriptors (their async H2D
    # copies drained many steps ago) can go -- captured graphs' descriptor
    # buffers live in keep_capture and are never trimmed
    _DEFER.jobs.clear()
    _DEFER.posts.clear()
    if len(_DEFER.keep_eager) > 64:
        del _DEFER.keep_eager[:32]


def wgrad_defer_end() -> None:
    _DEFER.on = False


def wgrad_flush() -> None:
    """Run all queued weight-gradient jobs in one batched kernel, then the
    post-callbacks (stacked-weight slice scatter).  Call between
    loss.backward() and the gradient all-reduce / optimizer step."""
    jobs, _DEFER.jobs = _DEFER.jobs, []
    posts, _DEFER.posts = _DEFER.posts, []
    _DEFER.on = False
    if jobs:
        from pvraft_amd import _C

        empty = torch.empty(0)
        keep = _C.pw_wgrad_batched(
            [j[0] for j in jobs],
            [j[1] for j in jobs],
            [j[2] for j in jobs],
            [j[3] if j[3] is not None else empty for j in jobs],
        )
        if torch.cuda.is_current_stream_capturing():
            _DEFER.keep_capture.append(keep)  # graph reads these at replay
        else:
            _DEFER.keep_eager.append(keep)  # until the async copy drains
    for fn in posts:
        fn()


def _grad_buffer(t: Tensor) -> Tensor:
    """fp32 accumulation target for a deferred tensor: a leaf parameter's
    .grad (created if absent), or the tensor itself when it already IS a
    plain fp32 buffer (stacked-weight case)."""
    if t.requires_grad and t.is_leaf:
        if t.grad is None:
            t.grad = torch.zeros_like(t)
        return t.grad
    return t


def defer_buffer(shape, device, post) -> Tensor:
    """Zeroed fp32 buffer for a stacked (non-leaf) weight's gradient;
    ``post(buf)`` scatters it into the source parameters' .grad at flush."""
    buf = torch.zeros(shape, dtype=torch.float32, device=device)
    _DEFER.posts.append(lambda: post(buf))
    return buf


def _cast_cached(t: Tensor, dt) -> Tensor:
    if t.dtype == dt:
        return t
    import os

    if os.environ.get("PVRAFT_NO_CAST_CACHE", "0") == "1":
        return t.to(dt)
    # derived weights (per-forward cat/slice stacks) carry a stable
    # _cast_key so each step's fresh tensor REPLACES its entry instead of
    # growing the registry forever (eager-mode leak otherwise).  Plain
    # VIEWS (weight.squeeze(-1) / .view(o, -1), fresh objects every call)
    # key by their base parameter + view geometry, so the 8-iteration GRU
    # loop hits instead of re-casting ~40 weights per step.
    ck = getattr(t, "_cast_key", None)
    if ck is None:
        base = t._base if t._base is not None else t
        ck = (id(base), tuple(t.shape), tuple(t.stride()),
              t.storage_offset())
    key = (ck, dt)
    ent = _CASTS.get(key)
    if ent is None:
        _CASTS[key] = [t, t.detach().to(dt), t._version]
        return _CASTS[key][1]
    # in-place updates between refreshes (optimizer steps around direct
    # sub-block calls outside PVRaft.forward) bump _version: re-fill the
    # mirror in place.  Inside hipGraph capture _version is stable and
    # the captured refresh re-reads the fp32 source on every replay.
    # a fresh view object over the SAME storage is still a hit; only new
    # storage (rebuilt cat stacks) or an in-place base update re-fills
    if ent[0].data_ptr() != t.data_ptr() or ent[2] != t._version:
        ent[0] = t
        ent[1].copy_(t.detach())
        ent[2] = t._version
    return ent[1]


class _PwMatmul(torch.autograd.Function):
    """y (B, Co, S) = w (Co, Ci) @ x (B, Ci, S) with split-K backward.

    GEMMs run as bmm with a stride-0 batched weight: aten::matmul's 2D@3D
    decomposition clones activation-sized tensors whenever gradients are
    required (measured ~14 ms/step of hidden aten::copy_); bmm is a native
    batched op with no fold path.

    The autocast weight cast happens INSIDE the Function (``compute_dt``):
    the Function's inputs stay the fp32 leaves, so the fp32 weight/bias
    grads from the wgrad kernel are returned directly -- no dw->bf16 copy
    + ToCopyBackward cast-back pair per weight per iteration (~250 copy
    kernels/step measured when the cast was an autograd-visible op).
    """

    @staticmethod
    def forward(ctx, w: Tensor, x: Tensor, bias, compute_dt, targets) -> Tensor:
        wc = w if compute_dt is None else _cast_cached(w, compute_dt)
        bc = bias if (bias is None or compute_dt is None) else _cast_cached(bias, compute_dt)
        ctx.save_for_backward(wc, x)
        ctx.has_bias = bias is not None
        ctx.grad_dtypes = (w.dtype, bias.dtype if bias is not None else None)
        ctx.targets = targets
        y = torch.bmm(wc.unsqueeze(0).expand(x.shape[0], -1, -1), x)
        if bc is not None:
            y = y + bc.view(1, -1, 1)
        return y

    @staticmethod
    def backward(ctx, dy: Tensor):
        wc, x = ctx.saved_tensors
        w_dtype, b_dtype = ctx.grad_dtypes
        dy = dy.contiguous()
        dx = (
            torch.bmm(wc.t().unsqueeze(0).expand(dy.shape[0], -1, -1), dy)
            if ctx.needs_input_grad[1]
            else None
        )
        dw = None
        dbias = None
        need_w = ctx.needs_input_grad[0]
        need_bias = ctx.has_bias and ctx.needs_input_grad[2]
        # deferred path: queue the wgrad job (batched kernel at flush time,
        # accumulated into the grad buffers directly) and return no grads
        # for w/bias so autograd does no per-call accumulate
        if (
            need_w
            and _DEFER.on
            and ctx.targets is not None
            and x.is_cuda
            and x.dtype == torch.bfloat16
            and dy.dtype == torch.bfloat16
            and (not need_bias or ctx.targets[1] is not None)
        ):
            wt, bt = ctx.targets
            _DEFER.jobs.append((
                dy,
                x,
                _grad_buffer(wt).view(-1),
                _grad_buffer(bt).view(-1) if need_bias else None,
            ))
            return None, dx, None, None, None
        if need_w:
            if x.is_cuda and x.dtype == torch.bfloat16:
                try:
                    from pvraft_amd import _C

                    dw, db = _C.pw_wgrad(dy, x, 0, need_bias)
                    if need_bias:
                        dbias = db.to(b_dtype) if b_dtype != db.dtype else db
                        need_bias = False
                except (ImportError, AttributeError):
                    dw = torch.einsum("bos,bis->oi", dy, x)
            else:
                dw = torch.einsum("bos,bis->oi", dy, x)
            if dw.dtype != w_dtype:
                dw = dw.to(w_dtype)
        if need_bias:
            dbias = dy.sum(dim=(0, 2))
            if dbias.dtype != b_dtype:
                dbias = dbias.to(b_dtype)
        return dw, dx, dbias, None, None


class _PwFused(torch.autograd.Function):
    """One MFMA launch for y = act(sum_i W_i @ x_i + bias + addend)
    (csrc/pw_fwd.hip) -- the whole concat-free conv stack piece with its
    epilogue, instead of per-part hipBLASLt GEMMs + add/bias/ReLU
    elementwise launches.  Weight/bias grads ride the deferred batched
    wgrad path when armed."""

    @staticmethod
    def forward(ctx, bias, addend, conf, *wx):
        from pvraft_amd import _C

        act_id, targets, p, compute_dt = conf
        ws = [
            w if compute_dt is None else _cast_cached(w, compute_dt)
            for w in wx[:p]
        ]
        xs = list(wx[p:])
        y = _C.pw_fwd(ws, xs, bias, addend, act_id)
        ctx.save_for_backward(*ws, *xs, y)
        ctx.conf = (act_id, targets, p)
        ctx.has_bias = bias is not None
        ctx.has_addend = addend is not None
        ctx.w_dtypes = tuple(w.dtype for w in wx[:p])
        return y

    @staticmethod
    def backward(ctx, dy: Tensor):
        act_id, targets, p = ctx.conf
        saved = ctx.saved_tensors
        ws, xs, y = saved[:p], saved[p : 2 * p], saved[2 * p]
        dy = dy.contiguous()
        dpre = (
            torch.ops.aten.threshold_backward(dy, y, 0) if act_id == 1 else dy
        )
        d_addend = dpre if ctx.has_addend else None
        dxs = [
            torch.bmm(w.t().unsqueeze(0).expand(dpre.shape[0], -1, -1), dpre)
            if ctx.needs_input_grad[3 + p + i]
            else None
            for i, w in enumerate(ws)
        ]
        dws = [None] * p
        d_bias = None
        need_bias = ctx.has_bias and ctx.needs_input_grad[0]
        if _DEFER.on and targets is not None:
            for i in range(p):
                if ctx.needs_input_grad[3 + i]:
                    bt = None
                    if i == 0 and need_bias and targets[p] is not None:
                        bt = _grad_buffer(targets[p]).view(-1)
                        need_bias = False
                    _DEFER.jobs.append(
                        (dpre, xs[i], _grad_buffer(targets[i]).view(-1), bt)
                    )
        else:
            for i in range(p):
                if ctx.needs_input_grad[3 + i]:
                    try:
                        from pvraft_amd import _C

                        dw, db = _C.pw_wgrad(dpre, xs[i], 0, i == 0 and need_bias)
                        if i == 0 and need_bias:
                            d_bias = db
                            need_bias = False
                    except (ImportError, AttributeError):
                        dw = torch.einsum("bos,bis->oi", dpre, xs[i])
                    wd = ctx.w_dtypes[i]
                    dws[i] = dw.to(wd) if dw.dtype != wd else dw
        if need_bias:
            d_bias = dpre.sum(dim=(0, 2)).float()
        return (d_bias, d_addend, None, *dws, *dxs)


def pw_fused(parts, bias=None, bias_target=None, addend=None, act="none") -> Tensor:
    """act(sum_i w_i @ x_i + bias + addend) for 1x1-conv stacks.

    parts: list of (w (Co, Ci), x (B, Ci, S), wgrad_target_or_None).
    GPU + bf16 compute: one fused MFMA kernel.  Otherwise: composed
    pw_matmul calls (+ add/activation), same semantics.
    """
    import torch.nn.functional as F

    x0 = parts[0][1]
    dt = None
    if x0.is_cuda and torch.is_autocast_enabled():
        dt = torch.get_autocast_dtype("cuda")
    fused_ok = (
        dt == torch.bfloat16
        and len(parts) <= 4
        and parts[0][0].shape[0] <= 256
        and all(w.shape[1] <= 224 for w, _x, _t in parts)
        # degenerate skinny-k giant-S GEMMs (the kNN-branch 4->64 conv over
        # K*N) pad k to 32 and waste ~7/8 of the MFMA work; hipBLASLt wins
        # there (30 vs ~60 us measured) -- keep those on the bmm path
        and not (x0.shape[2] > 100_000 and min(w.shape[1] for w, _x, _t in parts) < 16)
        and os.environ.get("PVRAFT_NO_PWFWD", "0") != "1"
    )
    if fused_ok:
        try:
            from pvraft_amd import _C  # noqa: F401
        except ImportError:
            fused_ok = False
    if fused_ok:
        act_id = {"none": 0, "relu": 1}[act]
        p = len(parts)
        targets = None
        if all(t is not None for _w, _x, t in parts):
            targets = tuple(t for _w, _x, t in parts) + (bias_target,)
        wx = tuple(w for w, _x, _t in parts) + tuple(
            x.to(dt).contiguous() for _w, x, _t in parts
        )
        b = bias.float() if bias is not None and bias.dtype != torch.float32 else bias
        return _PwFused.apply(b, addend, (act_id, targets, p, dt), *wx)
    out = None
    for i, (w, x, tgt) in enumerate(parts):
        b = bias if i == 0 else None
        t = (tgt, bias_target) if tgt is not None else None
        term = pw_matmul(w, x, b, targets=t)
        out = term if out is None else out + term
    if addend is not None:
        out = out + addend
    if act == "relu":
        out = F.relu(out)
    return out


def pw_matmul(weight: Tensor, x: Tensor, bias: Tensor = None, targets=None) -> Tensor:
    """``targets`` = (w_target, bias_target): the fp32 accumulation targets
    for the DEFERRED wgrad path -- the underlying Parameters when ``weight``
    / ``bias`` are views of them, or plain fp32 buffers for stacked
    weights.  Ignored when deferral is inactive."""
    w = weight
    dt = None
    if x.is_cuda and torch.is_autocast_enabled():
        dt = torch.get_autocast_dtype("cuda")
        x = x.to(dt)
        if w.dtype == dt:
            dt = None  # already compute-dtype (e.g. pre-cast slice)
    elif x.dtype != w.dtype:
        x = x.to(w.dtype)
    if bias is not None and dt is None and bias.dtype != x.dtype:
        bias = bias.to(x.dtype)
    return _PwMatmul.apply(w, x.contiguous(), bias, dt, targets)


class PwConv1d(nn.Conv1d):
    """nn.Conv1d(k=1): fused MFMA GEMM+bias on the bf16 GPU path, bmm
    otherwise."""

    def forward(self, x: Tensor) -> Tensor:
        return pw_fused(
            [(self.weight.squeeze(-1), x, self.weight)],
            bias=self.bias, bias_target=self.bias,
        )


class PwConv2d(nn.Conv2d):
    """nn.Conv2d(k=1) over flattened spatial dims (same GEMM)."""

    def forward(self, x: Tensor) -> Tensor:
        B, C, H, W = x.shape
        y = pw_fused(
            [(self.weight.view(self.out_channels, C), x.reshape(B, C, H * W), self.weight)],
            bias=self.bias, bias_target=self.bias,
        )
        return y.view(B, self.out_channels, H, W)
