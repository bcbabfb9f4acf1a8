"""GRU update block (reference model/update.py).

MotionEncoder: corr (64) and flow (3) -> 64-d motion feature with the raw
flow appended (update.py:15-20).  ConvGRU: pointwise gates over B x C x N
(update.py:31-40).  FlowHead: Conv1d branch || SetConv branch on the context
graph, concat -> Conv1d 128->64 -> ReLU -> Conv1d 64->3 (update.py:57-72).
Attribute names match reference state dicts.

Concat-free execution: every concatenation in this block feeds a 1x1 conv,
and W @ cat(a, b) == W_a @ a + W_b @ b with W_a/W_b column slices of the
weight -- so the big per-iteration cat tensors (hx, [r*h; x], [cor; flo],
[out_set; out], each ~2-6 MB x 8 GRU iterations) are never materialised;
the gate preactivations are sums of partial GEMMs on the existing tensors.
Weight slicing is a view (no copy); values and state dicts are identical
to the reference formulation up to fp summation order.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F
from torch import Tensor

from .graph import Graph
from .pointwise import PwConv1d, pw_matmul
from .setconv import SetConv


def _slice_weight(weight: Tensor, sizes):
    """Contiguous column slices of a (Co, Ci, 1) conv weight, each paired
    with a deferred-wgrad target: a zeroed fp32 buffer whose flush-time
    post-callback adds it into the right column block of the parameter's
    .grad (see pointwise.defer_buffer).  Returns [(w_slice, target|None)]."""
    from . import pointwise

    w = weight.squeeze(-1)
    defer = pointwise.wgrad_defer_active()
    out, lo = [], 0
    for s_ in sizes:
        sl = w[:, lo : lo + s_].contiguous()
        # stable registry key: each step's fresh slice replaces (not
        # grows) its autocast-mirror entry (pointwise._cast_cached)
        sl._cast_key = (id(weight), "slice", lo, s_)
        tgt = None
        if defer:
            lo_c, sz = lo, s_
            tgt = pointwise.defer_buffer(
                sl.shape, weight.device,
                lambda buf, lo_c=lo_c, sz=sz: pointwise._grad_buffer(weight)
                .view(weight.shape[0], -1)[:, lo_c : lo_c + sz]
                .add_(buf),
            )
        out.append((sl, tgt))
        lo += s_
    return out


def _split_mm(weight: Tensor, bias, parts, wparts=None, act="none"):
    """act(sum_i W[:, lo_i:hi_i] @ parts[i] + bias): the concat-free 1x1
    conv with its activation.

    weight: (Co, Ci_total, 1) conv weight; parts: list of (B, Ci_i, N)
    tensors whose channel sizes sum to Ci_total.  ``wparts`` supplies
    pre-sliced contiguous weights (cached once per forward -- the GRU loop
    reuses the same slices for all iterations, which also lets autocast's
    weight-cast cache hit).  On GPU the whole sum + bias + activation is
    one fused MFMA kernel (pointwise.pw_fused).
    """
    from .pointwise import pw_fused

    if wparts is None:
        wparts = _slice_weight(weight, [p.shape[1] for p in parts])
    return pw_fused(
        [(w_i, p, tgt) for (w_i, tgt), p in zip(wparts, parts)],
        bias=bias, bias_target=bias, act=act,
    )


class MotionEncoder(nn.Module):
    def __init__(self):
        super().__init__()
        self.conv_corr = PwConv1d(64, 64, 1)
        self.conv_flow = PwConv1d(3, 64, 1)
        self.conv = PwConv1d(64 + 64, 64 - 3, 1)

    def forward(self, flow: Tensor, corr: Tensor, wcache=None):
        """flow (B, N, 3), corr (B, 64, N) -> (motion61 (B, 61, N), flow_t (B, 3, N)).

        The reference returns cat([motion61, flow_t]) (update.py:19-20); the
        parts are kept separate here and consumed slice-wise downstream.
        """
        from pvraft_amd import ops
        from .pointwise import pw_fused

        flow_t = ops.transpose_last2(flow)
        cor = pw_fused(
            [(self.conv_corr.weight.squeeze(-1), corr, self.conv_corr.weight)],
            bias=self.conv_corr.bias, bias_target=self.conv_corr.bias, act="relu",
        )
        flo = pw_fused(
            [(self.conv_flow.weight.squeeze(-1), flow_t, self.conv_flow.weight)],
            bias=self.conv_flow.bias, bias_target=self.conv_flow.bias, act="relu",
        )
        wp = wcache.get("motion") if wcache else None
        out = _split_mm(self.conv.weight, self.conv.bias, [cor, flo], wp, act="relu")
        return out, flow_t


class ConvGRU(nn.Module):
    def __init__(self, input_dim: int = 128, hidden_dim: int = 64):
        super().__init__()
        self.convz = PwConv1d(input_dim + hidden_dim, hidden_dim, 1)
        self.convr = PwConv1d(input_dim + hidden_dim, hidden_dim, 1)
        self.convq = PwConv1d(input_dim + hidden_dim, hidden_dim, 1)

    def forward(self, h: Tensor, x: Tensor) -> Tensor:
        """Reference formulation (update.py:31-40) for direct use."""
        return self.forward_parts(h, [x])

    def gate_weights(self, part_sizes):
        """Cross-gate fused weight slices.

        All three gates consume the same non-hidden inputs, so their weight
        column slices stack row-wise into ONE GEMM per input part
        (z/r/q = rows [0:64/64:128/128:192]); only the hidden operand
        differs (z,r read h; q reads r*h), so the h columns fuse z+r only.

        When deferred wgrad is active every stacked tensor is paired with a
        fp32 buffer ("t_*") whose flush-time callback scatters it back into
        convz/convr/convq's .grad column blocks.
        """
        from . import pointwise

        hd = self.convz.out_channels
        wz = self.convz.weight.squeeze(-1)
        wr = self.convr.weight.squeeze(-1)
        wq = self.convq.weight.squeeze(-1)
        out = {"zr_h": torch.cat([wz[:, :hd], wr[:, :hd]], dim=0).contiguous(),
               "q_h": wq[:, :hd].contiguous(),
               "b_zr": torch.cat([self.convz.bias, self.convr.bias], dim=0),
               "b_q": self.convq.bias,
               "parts": [],
               "t_zr_h": None, "t_q_h": None, "t_b_zr": None, "t_parts": []}
        out["zr_h"]._cast_key = (id(self), "zr_h")
        out["q_h"]._cast_key = (id(self), "q_h")
        out["b_zr"]._cast_key = (id(self), "b_zr")
        lo = hd
        for sz in part_sizes:
            part = torch.cat([wz[:, lo : lo + sz], wr[:, lo : lo + sz], wq[:, lo : lo + sz]], dim=0).contiguous()
            part._cast_key = (id(self), "part", lo, sz)
            out["parts"].append(part)
            lo += sz
        if pointwise.wgrad_defer_active():
            dev = wz.device
            gbuf = pointwise._grad_buffer
            zw, rw, qw = self.convz.weight, self.convr.weight, self.convq.weight

            def scatter_h(buf):
                gbuf(zw).view(hd, -1)[:, :hd].add_(buf[:hd])
                gbuf(rw).view(hd, -1)[:, :hd].add_(buf[hd:])

            out["t_zr_h"] = pointwise.defer_buffer((2 * hd, hd), dev, scatter_h)
            out["t_q_h"] = pointwise.defer_buffer(
                (hd, hd), dev,
                lambda buf: gbuf(qw).view(hd, -1)[:, :hd].add_(buf))
            zb, rb = self.convz.bias, self.convr.bias
            out["t_b_zr"] = pointwise.defer_buffer(
                (2 * hd,), dev,
                lambda buf: (gbuf(zb).add_(buf[:hd]), gbuf(rb).add_(buf[hd:])))
            lo = hd
            for sz in part_sizes:
                def scatter_part(buf, lo_c=lo, s=sz):
                    gbuf(zw).view(hd, -1)[:, lo_c:lo_c + s].add_(buf[:hd])
                    gbuf(rw).view(hd, -1)[:, lo_c:lo_c + s].add_(buf[hd:2 * hd])
                    gbuf(qw).view(hd, -1)[:, lo_c:lo_c + s].add_(buf[2 * hd:])
                out["t_parts"].append(
                    pointwise.defer_buffer((3 * hd, sz), dev, scatter_part))
                lo += sz
        else:
            out["t_parts"] = [None] * len(part_sizes)
        return out

    def precompute_inp(self, inp: Tensor, gw) -> Tensor:
        """Contribution of the iteration-constant context features to all
        three gate preactivations -- hoisted out of the GRU loop."""
        from .pointwise import pw_fused

        return pw_fused([(gw["parts"][0], inp, gw["t_parts"][0])])

    def forward_parts(self, h: Tensor, x_parts, wcache=None, gw=None, pre=None) -> Tensor:
        """Gates from the concat parts (cross-gate fused GEMMs); the gate
        elementwise math (sigmoid/tanh/r*h/blend + backward) runs as two
        fused HIP kernels per direction (ops.gru_zr / ops.gru_q)."""
        from pvraft_amd import ops

        from .pointwise import pw_fused

        if gw is None:
            gw = self.gate_weights([p.shape[1] for p in x_parts])
        hd = self.convz.out_channels
        # shared-input contribution for all gates: (B, 3*hd, N).  The
        # iteration-varying parts (motion, flow) sum in ONE fused kernel
        # with the hoisted context contribution as the carried addend.
        if pre is not None:
            m = pw_fused(
                [(w_i, p, t) for w_i, t, p in
                 zip(gw["parts"][1:], gw["t_parts"][1:], x_parts[1:])],
                addend=pre,
            )
        else:
            m = pw_fused(
                [(w_i, p, t) for w_i, t, p in
                 zip(gw["parts"], gw["t_parts"], x_parts)],
            )
        # gate preactivations: hidden-operand GEMM + the matching slice of
        # m, again one kernel each
        pre_zr = pw_fused([(gw["zr_h"], h, gw["t_zr_h"])], bias=gw["b_zr"],
                          bias_target=gw["t_b_zr"], addend=m[:, : 2 * hd])
        z, rh = ops.gru_zr(pre_zr, h)
        pre_q = pw_fused([(gw["q_h"], rh, gw["t_q_h"])], bias=gw["b_q"],
                         bias_target=gw["b_q"], addend=m[:, 2 * hd :])
        return ops.gru_q(pre_q, z, h)


class ConvRNN(nn.Module):
    """Vanilla tanh RNN alternative to ConvGRU (reference update.py:43-54;
    defined for parity -- the reference never instantiates it either)."""

    def __init__(self, input_dim: int = 128, hidden_dim: int = 64):
        super().__init__()
        self.convx = PwConv1d(input_dim, hidden_dim, 1)
        self.convh = PwConv1d(hidden_dim, hidden_dim, 1)

    def forward(self, h: Tensor, x: Tensor) -> Tensor:
        return torch.tanh(self.convx(x) + self.convh(h))


class FlowHead(nn.Module):
    def __init__(self, input_dim: int = 64):
        super().__init__()
        self.conv1 = PwConv1d(input_dim, 64, 1)
        self.setconv = SetConv(64, 64)
        self.out_conv = nn.Sequential(
            PwConv1d(128, 64, 1),
            nn.ReLU(),
            PwConv1d(64, 3, 1),
        )

    def forward(self, x: Tensor, graph: Graph, wcache=None) -> Tensor:
        from pvraft_amd import ops

        out = self.conv1(x)
        out_set = ops.transpose_last2(self.setconv(ops.transpose_last2(x), graph))
        # out_conv(cat([out_set, out])) without the cat; ReLU fused
        wp = wcache.get("flowhead") if wcache else None
        mid = _split_mm(self.out_conv[0].weight, self.out_conv[0].bias,
                        [out_set, out], wp, act="relu")
        return self.out_conv[2](mid)


class UpdateBlock(nn.Module):
    def __init__(self, input_dim: int = 128, hidden_dim: int = 64):
        super().__init__()
        self.motion_encoder = MotionEncoder()
        self.gru = ConvGRU(input_dim=input_dim, hidden_dim=hidden_dim)
        self.flow_head = FlowHead(input_dim=hidden_dim)

    def make_wcache(self):
        """Prepare the concat-free weights once per forward (the GRU loop
        reuses them across all iterations): cross-gate fused GRU weights
        plus the motion/flow-head column slices."""
        # slices stay in the weights' dtype: pw_matmul's per-forward cast
        # cache dedupes the autocast cast across the GRU iterations, and
        # keeping the Function inputs fp32 keeps the weight grads fp32
        # end-to-end (no bf16 round-trip per iteration)
        return {
            "gru": self.gru.gate_weights([64, 61, 3]),  # [inp, motion61, flow3]
            "motion": _slice_weight(self.motion_encoder.conv.weight, [64, 64]),
            "flowhead": _slice_weight(self.flow_head.out_conv[0].weight, [64, 64]),
        }

    def forward(self, net: Tensor, inp: Tensor, corr: Tensor, flow: Tensor, graph: Graph,
                wcache=None, inp_pre=None):
        from pvraft_amd import ops

        motion, flow_t = self.motion_encoder(flow, corr, wcache)
        # gru input = cat(inp, motion, flow_t) (reference update.py:84), as parts
        gw = wcache.get("gru") if wcache else None
        net = self.gru.forward_parts(net, [inp, motion, flow_t], wcache, gw=gw, pre=inp_pre)
        delta_flow = ops.transpose_last2(self.flow_head(net, graph, wcache))
        return net, delta_flow
