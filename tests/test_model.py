"""Model-level tests: shapes, iteration behaviour, reference state-dict
parity, refine head, gradients end-to-end (CPU, small N)."""

import torch

from pvraft_amd.model import PVRaft, PVRaftRefine


def tiny_model(refine=False, truncate_k=16):
    cls = PVRaftRefine if refine else PVRaft
    return cls(corr_levels=3, base_scales=0.25, truncate_k=truncate_k)


def clouds(B=2, N=48):
    xyz1 = torch.randn(B, N, 3)
    xyz2 = xyz1 + 0.1 * torch.randn(B, N, 3)
    return [xyz1, xyz2]


def test_forward_returns_per_iteration_flows():
    model = tiny_model()
    p = clouds()
    flows = model(p, num_iters=3)
    assert len(flows) == 3
    for f in flows:
        assert f.shape == (2, 48, 3)
        assert torch.isfinite(f).all()


def test_backward_reaches_all_parameters():
    model = tiny_model()
    flows = model(clouds(), num_iters=2)
    loss = sum(f.abs().mean() for f in flows)
    loss.backward()
    missing = [n for n, p in model.named_parameters() if p.grad is None]
    assert missing == [], f"params without grad: {missing}"


def test_state_dict_matches_reference_layout():
    """Key names/shapes follow the reference (checkpoint interchange)."""
    model = tiny_model(refine=True)
    sd = model.state_dict()
    expected = {
        "feature_extractor.feat_conv1.fc1.weight": (16, 6, 1, 1),
        "feature_extractor.feat_conv2.fc2.weight": (64, 48, 1, 1)[:2] + (1,),
        "context_extractor.feat_conv3.gn3.weight": (128,),
        "corr_block.out_conv.0.weight": (128, 81, 1),
        "corr_block.out_conv.2.weight": (1,),  # PReLU
        "corr_block.out_conv.3.weight": (64, 128, 1),
        "corr_block.knn_conv.0.weight": (64, 4, 1, 1),
        "corr_block.knn_out.weight": (64, 64, 1),
        "update_block.motion_encoder.conv.weight": (61, 128, 1),
        "update_block.gru.convz.weight": (64, 192, 1),
        "update_block.flow_head.conv1.weight": (64, 64, 1),
        "update_block.flow_head.setconv.fc1.weight": (64, 67, 1, 1),
        "update_block.flow_head.out_conv.2.weight": (3, 64, 1),
        "refine_block.ref_conv1.fc1.weight": (16, 6, 1, 1),
        "refine_block.fc.weight": (3, 128),
    }
    for key, shape in expected.items():
        assert key in sd, f"missing {key}"
        assert tuple(sd[key].shape) == tuple(shape), (key, sd[key].shape, shape)


def test_setconv_mid_width_rule():
    """gconv.py:21-24: mid = out//2 if in odd else (in+out)//2."""
    from pvraft_amd.model import SetConv

    assert SetConv(3, 32).fc1.out_channels == 16
    assert SetConv(32, 64).fc1.out_channels == 48
    assert SetConv(64, 128).fc1.out_channels == 96
    assert SetConv(64, 64).fc1.out_channels == 64


def test_refine_model_returns_single_flow_and_only_refine_grads():
    model = tiny_model(refine=True)
    model.freeze_backbone()
    flow = model(clouds(), num_iters=2)
    assert flow.shape == (2, 48, 3)
    flow.abs().mean().backward()
    for n, p in model.named_parameters():
        if n.startswith("refine_block"):
            assert p.grad is not None, n
        else:
            assert p.grad is None, n


def test_stage1_checkpoint_loads_into_refine_model():
    m1 = tiny_model(refine=False)
    m2 = tiny_model(refine=True)
    missing, unexpected = m2.load_state_dict(m1.state_dict(), strict=False)
    assert unexpected == []
    assert all(k.startswith("refine_block") for k in missing)


def test_iteration_zero_of_longer_run_matches_shorter_run():
    """The GRU loop is causal: first k flows do not depend on later iters."""
    model = tiny_model()
    model.eval()
    p = clouds(B=1, N=40)
    with torch.no_grad():
        f2 = model(p, num_iters=2)
        f4 = model(p, num_iters=4)
    assert torch.allclose(f2[0], f4[0], atol=1e-5)
    assert torch.allclose(f2[1], f4[1], atol=1e-5)


def test_pw_conv_matches_nn_conv():
    """PwConv1d/2d (matmul + split-K wgrad) vs stock nn.Conv forward/backward."""
    from pvraft_amd.model.pointwise import PwConv1d, PwConv2d

    torch.manual_seed(0)
    conv = torch.nn.Conv1d(6, 10, 1)
    pw = PwConv1d(6, 10, 1)
    pw.load_state_dict(conv.state_dict())
    x1 = torch.randn(2, 6, 32, requires_grad=True)
    x2 = x1.detach().clone().requires_grad_(True)
    y1 = conv(x1)
    y2 = pw(x2)
    assert torch.allclose(y1, y2, atol=1e-6)
    g = torch.randn_like(y1)
    y1.backward(g)
    y2.backward(g)
    assert torch.allclose(x1.grad, x2.grad, atol=1e-6)
    assert torch.allclose(conv.weight.grad.squeeze(-1), pw.weight.grad.squeeze(-1), atol=1e-5)
    assert torch.allclose(conv.bias.grad, pw.bias.grad, atol=1e-5)

    conv2 = torch.nn.Conv2d(5, 7, 1, bias=False)
    pw2 = PwConv2d(5, 7, 1, bias=False)
    pw2.load_state_dict(conv2.state_dict())
    x = torch.randn(2, 5, 4, 16, requires_grad=True)
    xb = x.detach().clone().requires_grad_(True)
    ya = conv2(x)
    yb = pw2(xb)
    assert torch.allclose(ya, yb, atol=1e-6)
    ya.sum().backward()
    yb.sum().backward()
    assert torch.allclose(x.grad, xb.grad, atol=1e-6)
    assert torch.allclose(conv2.weight.grad, pw2.weight.grad, atol=1e-5)


def test_concat_free_gru_matches_cat_formulation():
    """forward_parts (summed weight-slice GEMMs) == conv on torch.cat."""
    from pvraft_amd.model.update import ConvGRU, _split_mm

    torch.manual_seed(1)
    gru = ConvGRU(input_dim=128, hidden_dim=64)
    h = torch.randn(2, 64, 40)
    inp = torch.randn(2, 64, 40)
    motion = torch.randn(2, 61, 40)
    flow_t = torch.randn(2, 3, 40)

    out_parts = gru.forward_parts(h, [inp, motion, flow_t])
    out_cat = gru(h, torch.cat([inp, motion, flow_t], dim=1))
    assert torch.allclose(out_parts, out_cat, atol=1e-5)

    # _split_mm == conv1d on the concatenation
    conv = torch.nn.Conv1d(128, 61, 1)
    got = _split_mm(conv.weight, conv.bias, [inp, motion, flow_t])
    want = conv(torch.cat([inp, motion, flow_t], dim=1))
    assert torch.allclose(got, want, atol=1e-5)


def test_predictor_cpu_fallback_exposes_flows():
    """Predictor without a GPU serves the eager path and exposes the full
    per-iteration flow list for sequence-loss evaluation."""
    import torch

    from pvraft_amd.engine import Predictor
    from pvraft_amd.model import PVRaft

    model = PVRaft(truncate_k=16)
    pred = Predictor(model, points=64, batch=1, iters=3, use_graph=False)
    xyz1 = torch.randn(1, 64, 3)
    xyz2 = xyz1 + 0.01 * torch.randn(1, 64, 3)
    final = pred(xyz1, xyz2)
    assert final.shape == (1, 64, 3)
    assert len(pred.last_flows) == 3
    assert torch.equal(pred.last_flows[-1], final)


def test_cast_cache_invalidated_by_inplace_update():
    """the cast mirror must not serve stale bf16 values after an
    optimizer step mutates the fp32 weight in place (version-checked on
    hit, batch-refreshed by refresh_casts)."""
    from pvraft_amd.model.pointwise import (
        _cast_cached,
        clear_step_cache,
        refresh_casts,
    )

    clear_step_cache()
    w = torch.randn(4, 4)
    c1 = _cast_cached(w, torch.bfloat16)
    assert torch.equal(c1, w.to(torch.bfloat16))
    w.add_(1.0)  # in-place update bumps w._version
    c2 = _cast_cached(w, torch.bfloat16)
    assert torch.equal(c2, w.to(torch.bfloat16))
    # the batched per-step refresh path re-fills the mirror too
    w.mul_(2.0)
    refresh_casts()
    assert torch.equal(_cast_cached(w, torch.bfloat16), w.to(torch.bfloat16))
    clear_step_cache()

def test_gn_defer_targets_active_with_grad_disabled():
    """Regression: Function.backward runs with grad mode DISABLED; the
    deferred-gradient gate must key on the explicit _DEFER flag, not
    is_grad_enabled() -- the old gate silently rerouted every GN weight
    grad through AccumulateGrad, whose side-stream-pinned nodes race
    under hipGraph replay (graphed training stopped learning)."""
    import torch.nn as nn

    from pvraft_amd.model import pointwise
    from pvraft_amd.ops import _gn_defer_targets

    w = nn.Parameter(torch.randn(8))
    b = nn.Parameter(torch.randn(8))
    pointwise.wgrad_defer_begin()
    try:
        with torch.no_grad():  # the grad mode backward actually runs in
            tgt = _gn_defer_targets(w, b, None, act=1)
        assert tgt is not None
        assert tgt[0].shape == (8,) and tgt[1].shape == (8,)
    finally:
        pointwise.wgrad_defer_end()
        w.grad = None
        b.grad = None


def test_cast_registry_does_not_grow_across_steps():
    """Regression: per-forward weight stacks and fresh views must REPLACE
    their registry entries (stable keys), not accumulate one entry per
    step (unbounded eager-mode growth)."""
    from pvraft_amd.model.pointwise import _CASTS, _cast_cached, clear_step_cache

    clear_step_cache()
    base = torch.randn(4, 8, 1)
    for _ in range(5):
        v = base.squeeze(-1)          # fresh view object each "step"
        _cast_cached(v, torch.bfloat16)
        stack = torch.cat([base[:, :4, 0], base[:, 4:, 0]], dim=1).contiguous()
        stack._cast_key = (id(base), "stack")
        _cast_cached(stack, torch.bfloat16)
    assert len(_CASTS) == 2, len(_CASTS)
    clear_step_cache()

