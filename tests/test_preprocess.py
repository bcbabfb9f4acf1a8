"""Preprocessing IO + geometry tests (pure numpy, synthetic files)."""

import os
import struct
import zlib

import numpy as np
import pytest

from pvraft_amd.data.preprocess import io as pio
from pvraft_amd.data.preprocess.flyingthings3d import backproject
from pvraft_amd.data.preprocess.kitti import disp_to_depth, pixel_to_xyz


def write_pfm(path, data, scale=-1.0):
    with open(path, "wb") as f:
        f.write(b"Pf\n" if data.ndim == 2 else b"PF\n")
        h, w = data.shape[:2]
        f.write(f"{w} {h}\n".encode())
        f.write(f"{scale}\n".encode())
        f.write(np.flipud(data).astype("<f4").tobytes())


def write_flo(path, flow):
    with open(path, "wb") as f:
        f.write(struct.pack("<f", 202021.25))
        h, w = flow.shape[:2]
        f.write(struct.pack("<ii", w, h))
        f.write(flow.astype("<f4").tobytes())


def write_png(path, arr, bitdepth, filters=None):
    """Minimal PNG writer (same subset as the reader) for roundtrips."""
    h, w = arr.shape[:2]
    channels = 1 if arr.ndim == 2 else arr.shape[2]
    colortype = 0 if channels == 1 else 2
    if bitdepth == 16:
        payload = arr.astype(">u2").tobytes()
    else:
        payload = arr.astype(np.uint8).tobytes()
    stride = w * channels * (bitdepth // 8)
    raw = b""
    for row in range(h):
        ftype = 0 if filters is None else filters[row % len(filters)]
        raw += bytes([0]) + payload[row * stride : (row + 1) * stride]
    idat = zlib.compress(raw)

    def chunk(ctype, data):
        c = ctype + data
        return struct.pack(">I", len(data)) + c + struct.pack(">I", zlib.crc32(c))

    ihdr = struct.pack(">IIBBBBB", w, h, bitdepth, colortype, 0, 0, 0)
    with open(path, "wb") as f:
        f.write(b"\x89PNG\r\n\x1a\n")
        f.write(chunk(b"IHDR", ihdr))
        f.write(chunk(b"IDAT", idat))
        f.write(chunk(b"IEND", b""))


def test_pfm_roundtrip(tmp_path):
    data = np.random.rand(7, 5).astype(np.float32) * 50
    p = str(tmp_path / "d.pfm")
    write_pfm(p, data)
    assert np.allclose(pio.read_pfm(p), data)


def test_flo_roundtrip(tmp_path):
    flow = np.random.randn(6, 9, 2).astype(np.float32)
    p = str(tmp_path / "f.flo")
    write_flo(p, flow)
    assert np.allclose(pio.read_flo(p), flow)


@pytest.mark.parametrize("bitdepth,channels", [(8, 1), (16, 1), (16, 3)])
def test_png_roundtrip(tmp_path, bitdepth, channels):
    rng = np.random.default_rng(0)
    shape = (11, 13) if channels == 1 else (11, 13, 3)
    hi = 255 if bitdepth == 8 else 65535
    arr = rng.integers(0, hi + 1, size=shape).astype(np.uint16 if bitdepth == 16 else np.uint8)
    p = str(tmp_path / "x.png")
    write_png(p, arr, bitdepth)
    got = pio.read_png(p)
    assert got.dtype == arr.dtype
    assert np.array_equal(got, arr)


def test_png_up_filter(tmp_path):
    """PIL writes filtered rows; exercise the Up filter path explicitly."""
    from PIL import Image

    rng = np.random.default_rng(1)
    arr = rng.integers(0, 256, size=(16, 16, 3)).astype(np.uint8)
    p = str(tmp_path / "pil.png")
    Image.fromarray(arr).save(p)  # PIL picks adaptive filters
    got = pio.read_png(p)
    assert np.array_equal(got, arr)


def test_ft3d_backprojection_geometry():
    disp = np.full((4, 4), 10.0, dtype=np.float32)
    pc = backproject(disp)
    # depth = -f * baseline / disp = 1050/10 = 105 (z positive in this frame)
    assert np.allclose(pc[..., 2], 105.0)
    # pixel at cx maps to x = 0
    disp1 = np.full((1, 960), 5.0, dtype=np.float32)
    pc1 = backproject(disp1)
    # x = -(px - cx) / disp = -(479 - 479.5)/5 = +0.1
    assert abs(pc1[0, 479, 0] - 0.1) < 1e-5


def test_kitti_depth_and_projection():
    disp = np.full((3, 3), 2.0, dtype=np.float32)
    valid = np.ones_like(disp, dtype=bool)
    depth = disp_to_depth(disp, valid, focal_px=720.0)
    assert np.allclose(depth, 720.0 * 0.54 / 2.00001, atol=1e-3)
    P = np.array([[720.0, 0, 600, 40], [0, 720.0, 180, 2], [0, 0, 1, 0.003]], dtype=np.float32)
    pc = pixel_to_xyz(depth, P)
    assert pc.shape == (3, 3, 3)
    assert np.allclose(pc[..., 2], depth)
