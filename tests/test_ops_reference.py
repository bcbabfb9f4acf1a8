"""Unit tests for the pure-torch op oracles (CPU path).

These pin the semantics that the HIP kernels are tested against on GPU:
kNN selection, edge gather, correlation truncation, voxel pyramid and kNN
correlation lookup (reference behaviours cited in pvraft_amd/ops/reference.py).
"""

import math

import pytest
import torch

from pvraft_amd.ops import reference as R


def brute_knn(xyz, k):
    # literal reference formulation (graph.py:53-60): full matrix + argsort
    d = (xyz ** 2).sum(-1, keepdim=True) + (xyz ** 2).sum(-1, keepdim=True).transpose(1, 2) \
        - 2 * torch.bmm(xyz, xyz.transpose(1, 2))
    return torch.argsort(d, -1)[..., :k]


def test_knn_idx_matches_brute_force():
    xyz = torch.randn(2, 64, 3, dtype=torch.float64)
    got = R.knn_idx(xyz.float(), 8, chunk=17)
    want = brute_knn(xyz, 8)
    # same neighbour SETS (tie order may differ)
    assert torch.equal(got.sort(-1).values, want.sort(-1).values)


def test_knn_self_included():
    xyz = torch.randn(1, 32, 3)
    idx = R.knn_idx(xyz, 4)
    # nearest neighbour of a point is itself (distance 0)
    assert torch.equal(idx[..., 0], torch.arange(32).unsqueeze(0))


def test_gather_edge_concat_values():
    B, N, C, K = 2, 16, 5, 4
    feats = torch.randn(B, N, C)
    xyz = torch.randn(B, N, 3)
    idx = torch.randint(0, N, (B, N, K))
    out = R.gather_edge_concat(feats, idx, xyz)
    assert out.shape == (B, C + 3, K, N)
    b, n, j = 1, 3, 2
    nb = idx[b, n, j]
    assert torch.allclose(out[b, :C, j, n], feats[b, nb] - feats[b, n])
    assert torch.allclose(out[b, C:, j, n], xyz[b, nb] - xyz[b, n])


def test_corr_truncate_matches_full():
    B, C, N, M, K = 2, 16, 33, 49, 12
    f1 = torch.randn(B, C, N)
    f2 = torch.randn(B, C, M)
    xyz2 = torch.randn(B, M, 3)
    corr, idx, txyz = R.corr_truncate(f1, f2, xyz2, K, chunk=10)
    full = torch.bmm(f1.transpose(1, 2), f2) / math.sqrt(C)
    want_v, want_i = full.topk(K, dim=2, sorted=True)
    assert torch.allclose(corr, want_v, atol=1e-5)
    assert torch.equal(idx, want_i)
    # gathered xyz must match the indices
    for b in range(B):
        assert torch.allclose(txyz[b], xyz2[b][idx[b]])


def test_corr_truncate_grads_flow():
    B, C, N, M, K = 1, 8, 12, 20, 5
    f1 = torch.randn(B, C, N, requires_grad=True)
    f2 = torch.randn(B, C, M, requires_grad=True)
    xyz2 = torch.randn(B, M, 3)
    corr, _, _ = R.corr_truncate(f1, f2, xyz2, K)
    corr.sum().backward()
    assert f1.grad is not None and f1.grad.abs().sum() > 0
    assert f2.grad is not None and f2.grad.abs().sum() > 0


def voxel_corr_scatter_oracle(corr, xyz, coords, base_scale, L, R_=3):
    """Literal per-element loop oracle for the voxel pyramid."""
    B, N, K = corr.shape
    out = torch.zeros(B, L * 27, N)
    for b in range(B):
        for n in range(N):
            for l in range(L):
                r = base_scale * (2 ** l)
                sums = torch.zeros(27)
                cnts = torch.zeros(27)
                for j in range(K):
                    dv = torch.round((xyz[b, n, j] - coords[b, n]) / r)
                    if (dv.abs() <= 1).all():
                        cell = int(dv[0] + 1) * 9 + int(dv[1] + 1) * 3 + int(dv[2] + 1)
                        sums[cell] += corr[b, n, j]
                        cnts[cell] += 1
                out[b, l * 27 : (l + 1) * 27, n] = sums / cnts.clamp(min=1)
    return out


def test_voxel_corr_against_loop_oracle():
    B, N, K, L = 1, 6, 24, 2
    corr = torch.randn(B, N, K)
    coords = torch.randn(B, N, 3)
    xyz = coords.unsqueeze(2) + torch.randn(B, N, K, 3) * 0.4
    got = R.voxel_corr(corr, xyz, coords, 0.25, L)
    want = voxel_corr_scatter_oracle(corr, xyz, coords, 0.25, L)
    assert got.shape == (B, L * 27, N)
    assert torch.allclose(got, want, atol=1e-5)


def test_voxel_corr_gradcheck():
    B, N, K = 1, 3, 8
    corr = torch.randn(B, N, K, dtype=torch.float64, requires_grad=True)
    coords = torch.randn(B, N, 3, dtype=torch.float64)
    xyz = coords.unsqueeze(2) + torch.randn(B, N, K, 3, dtype=torch.float64) * 0.3
    torch.autograd.gradcheck(
        lambda c: R.voxel_corr(c, xyz, coords, 0.25, 2), (corr,), eps=1e-6, atol=1e-4
    )


def test_knn_corr_values():
    B, N, K, k = 1, 5, 16, 4
    corr = torch.randn(B, N, K)
    coords = torch.randn(B, N, 3)
    xyz = coords.unsqueeze(2) + torch.randn(B, N, K, 3)
    out = R.knn_corr(corr, xyz, coords, k)
    assert out.shape == (B, 4, k, N)
    # check point 0: k nearest candidates by distance
    d = ((xyz[0, 0] - coords[0, 0]) ** 2).sum(-1)
    nbr = d.topk(k, largest=False).indices
    assert torch.allclose(out[0, 0, :, 0].sort().values, corr[0, 0, nbr].sort().values)
    rel = xyz[0, 0, nbr] - coords[0, 0]
    assert torch.allclose(
        out[0, 1:, :, 0].t().sort(0).values, rel.sort(0).values
    )


def test_knn_corr_grad_is_gather():
    B, N, K, k = 1, 4, 10, 3
    corr = torch.randn(B, N, K, requires_grad=True)
    coords = torch.randn(B, N, 3)
    xyz = coords.unsqueeze(2) + torch.randn(B, N, K, 3)
    out = R.knn_corr(corr, xyz, coords, k)
    out[:, 0].sum().backward()  # only channel 0 touches corr
    # each row has exactly k ones scattered at the selected indices
    assert corr.grad.sum().item() == pytest.approx(B * N * k)
    assert ((corr.grad == 0) | (corr.grad == 1)).all()


def test_corr_truncate_custom_backward_matches_autograd():
    """The gather-based backward (_CorrTruncate, GPU path) vs plain autograd."""
    from pvraft_amd.ops import _CorrTruncate

    B, C, N, M, K = 2, 8, 15, 21, 6
    f1 = torch.randn(B, C, N, requires_grad=True)
    f2 = torch.randn(B, C, M, requires_grad=True)
    xyz2 = torch.randn(B, M, 3)
    f1r = f1.detach().clone().requires_grad_(True)
    f2r = f2.detach().clone().requires_grad_(True)

    corr, idx, txyz = _CorrTruncate.apply(f1, f2, xyz2, K)
    corr_r, idx_r, txyz_r = R.corr_truncate(f1r, f2r, xyz2, K)
    assert torch.allclose(corr, corr_r, atol=1e-6)
    assert torch.equal(idx, idx_r)

    g = torch.randn_like(corr)
    corr.backward(g)
    corr_r.backward(g)
    assert torch.allclose(f1.grad, f1r.grad, atol=1e-5)
    assert torch.allclose(f2.grad, f2r.grad, atol=1e-5)
