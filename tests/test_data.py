"""Dataset/collate tests: synthetic pairs, FT3D/KITTI directory layouts
(fake on-disk trees), subsampling and the Batch collate."""

import os

import numpy as np
import pytest
import torch

from pvraft_amd.data import FT3D, Batch, Kitti, SyntheticSceneFlow


def test_synthetic_shapes_and_flow():
    ds = SyntheticSceneFlow(nb_points=64, length=4)
    item = ds[0]
    pc1, pc2 = item["sequence"]
    mask, flow = item["ground_truth"]
    assert pc1.shape == (1, 64, 3) and pc2.shape == (1, 64, 3)
    assert mask.shape == (1, 64, 1) and flow.shape == (1, 64, 3)
    # pc1/pc2 are independently permuted by subsampling, so compare as sets:
    # pc1 + flow must equal pc2 up to row order
    warped = (pc1 + flow)[0]
    diff = (warped.unsqueeze(1) - pc2[0].unsqueeze(0)).norm(dim=-1)
    assert (diff.min(dim=1).values < 1e-5).all()  # exact row match pre-shuffle
    assert (mask == 1).all()


def test_synthetic_deterministic_per_index():
    ds = SyntheticSceneFlow(nb_points=32, length=4, seed=3)
    a = ds.load_sequence(1)[0][0]
    b = ds.load_sequence(1)[0][0]
    assert np.array_equal(a, b)


def test_batch_collate():
    ds = SyntheticSceneFlow(nb_points=16, length=3)
    batch = Batch([ds[i] for i in range(3)])
    assert batch["sequence"][0].shape == (3, 16, 3)
    assert batch["ground_truth"][1].shape == (3, 16, 3)
    assert batch.batch_size == 3
    batch2 = batch.to(torch.device("cpu"))
    assert batch2["sequence"][0].device.type == "cpu"


def _write_ft3d_tree(root, n_train=10, n_test=4, points=32):
    rng = np.random.default_rng(0)
    for split, count in (("train", n_train), ("val", n_test)):
        for i in range(count):
            d = os.path.join(root, split, f"{i:07d}")
            os.makedirs(d)
            pc1 = rng.random((points, 3)).astype(np.float32)
            pc2 = rng.random((points, 3)).astype(np.float32)
            np.save(os.path.join(d, "pc1.npy"), pc1)
            np.save(os.path.join(d, "pc2.npy"), pc2)


def test_ft3d_splits_and_flip(tmp_path):
    _write_ft3d_tree(str(tmp_path))
    tr = FT3D(str(tmp_path), nb_points=16, mode="train", strict_sizes=False)
    va = FT3D(str(tmp_path), nb_points=16, mode="val", strict_sizes=False)
    te = FT3D(str(tmp_path), nb_points=16, mode="test", strict_sizes=False)
    assert len(tr) + len(va) == 10 and len(va) >= 1
    assert set(tr.filenames).isdisjoint(va.filenames)
    assert len(te) == 4
    item = tr[0]
    pc1, pc2 = item["sequence"]
    assert pc1.shape == (1, 16, 3)
    # x and z are sign-flipped on load -> all coords of the random tree <= 0
    assert (pc1[0, :, 0] <= 0).all() and (pc1[0, :, 2] <= 0).all()
    assert (pc1[0, :, 1] >= 0).all()


def test_ft3d_strict_size_enforced(tmp_path):
    _write_ft3d_tree(str(tmp_path))
    with pytest.raises(RuntimeError):
        FT3D(str(tmp_path), nb_points=16, mode="train", strict_sizes=True)


def test_kitti_filters(tmp_path):
    rng = np.random.default_rng(1)
    for i in range(5):
        d = os.path.join(str(tmp_path), f"{i:06d}")
        os.makedirs(d)
        pc1 = rng.normal(size=(64, 3)).astype(np.float32)
        pc2 = pc1 + 0.01
        pc1[:5, 1] = -2.0  # ground in both clouds
        pc2[:5, 1] = -2.0
        pc1[5:8, 2] = 40.0  # too far
        pc2[5:8, 2] = 40.0
        np.save(os.path.join(d, "pc1.npy"), pc1)
        np.save(os.path.join(d, "pc2.npy"), pc2)
    ds = Kitti(str(tmp_path), nb_points=16, strict_sizes=False, mapping_file="")
    assert len(ds) == 5
    seq, gt = ds.load_sequence(0)
    assert seq[0].shape[0] <= 64 - 5 - 3
    assert (seq[0][:, 2] < 35).all() and (seq[1][:, 2] < 35).all()
    # no point is ground (y < -1.4) in both clouds
    assert not np.logical_and(seq[0][:, 1] < -1.4, seq[1][:, 1] < -1.4).any()
    assert gt[0].shape == (seq[0].shape[0], 1)


def test_kitti_bundled_mapping_default(tmp_path):
    """The HPLFlowNet 142-scene protocol is the default (reference
    datasets/KITTI_mapping.txt consumed at kitti_hplflownet.py:44-50)."""
    from pvraft_amd.data.kitti import BUNDLED_MAPPING

    with open(BUNDLED_MAPPING) as fd:
        lines = [line.strip() for line in fd.readlines()]
    assert len(lines) == 200
    assert sum(1 for line in lines if line) == 142

    rng = np.random.default_rng(2)
    for i in range(200):
        d = os.path.join(str(tmp_path), f"{i:06d}")
        os.makedirs(d)
        pc = rng.normal(size=(32, 3)).astype(np.float32)
        np.save(os.path.join(d, "pc1.npy"), pc)
        np.save(os.path.join(d, "pc2.npy"), pc + 0.01)
    ds = Kitti(str(tmp_path), nb_points=16)  # no mapping arg: bundled default
    assert len(ds) == 142
    kept = {int(os.path.split(p)[-1]) for p in ds.paths}
    assert kept == {i for i, line in enumerate(lines) if line}


def test_subsample_skips_small_samples():
    class TinyThenBig(SyntheticSceneFlow):
        def load_sequence(self, idx):
            seq, gt = super().load_sequence(idx)
            if idx == 0:  # too few points
                seq = [s[:4] for s in seq]
                gt = [g[:4] for g in gt]
            return seq, gt

    ds = TinyThenBig(nb_points=8, length=3)
    item = ds[0]  # skips forward to idx 1
    assert item["sequence"][0].shape == (1, 8, 3)
