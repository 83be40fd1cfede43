"""Raw-image I/O: whole-frame, per-rank block offsets, parallel writers."""

import numpy as np

from parallel_convolution_amd.parallel.decomp import make_blocks
from parallel_convolution_amd.spec import DType
from parallel_convolution_amd.utils import rawio


def test_roundtrip_gray(tmp_path):
    img = rawio.synthetic_image(31, 47, 1, DType.U8)
    p = str(tmp_path / "img.raw")
    rawio.write_image(p, img)
    back = rawio.read_image(p, 31, 47, 1, DType.U8)
    np.testing.assert_array_equal(img, back)


def test_roundtrip_rgb_f16(tmp_path):
    img = rawio.synthetic_image(10, 12, 3, DType.F16)
    p = str(tmp_path / "img.raw")
    rawio.write_image(p, img)
    back = rawio.read_image(p, 10, 12, 3, DType.F16)
    np.testing.assert_array_equal(img, back)


def test_block_read_matches_slice(tmp_path):
    H, W, C = 40, 56, 3
    img = rawio.synthetic_image(H, W, C, DType.U8)
    p = str(tmp_path / "img.raw")
    rawio.write_image(p, img)
    for b in make_blocks(H, W, 8, C):
        blk = rawio.read_block(p, H, W, C, DType.U8, b.y0, b.x0, b.h, b.w)
        np.testing.assert_array_equal(
            blk, img[b.y0 : b.y0 + b.h, b.x0 * C : (b.x0 + b.w) * C]
        )


def test_block_write_reassembles(tmp_path):
    H, W, C = 33, 29, 1
    img = rawio.synthetic_image(H, W, C, DType.F32)
    p = str(tmp_path / "out.raw")
    for b in make_blocks(H, W, 4, C):
        rawio.write_block(
            p, H, W, C, DType.F32, b.y0, b.x0,
            img[b.y0 : b.y0 + b.h, b.x0 * C : (b.x0 + b.w) * C],
        )
    back = rawio.read_image(p, H, W, C, DType.F32)
    np.testing.assert_array_equal(img, back)


def test_synthetic_block_world_size_invariant():
    """Any decomposition of the synthetic image reproduces the exact same
    global pixels (per-rank seeding made world sizes
    incomparable; the counter-hash generator fixes that by construction)."""
    for dt in (DType.U8, DType.F16, DType.F32):
        img = rawio.synthetic_image(48, 56, 1, dt, seed=11)
        for p in (2, 4, 8):
            for b in make_blocks(48, 56, p, 1):
                blk = rawio.synthetic_block(
                    48, 56, 1, dt, b.y0, b.x0, b.h, b.w, seed=11
                )
                np.testing.assert_array_equal(
                    blk, img[b.y0 : b.y0 + b.h, b.x0 : b.x0 + b.w]
                )


def test_synthetic_block_rgb_window():
    img = rawio.synthetic_image(32, 40, 3, DType.U8, seed=5)
    blk = rawio.synthetic_block(32, 40, 3, DType.U8, 4, 6, 12, 20, seed=5)
    np.testing.assert_array_equal(blk, img[4:16, 18:78])


def test_synthetic_values_change_with_seed():
    a = rawio.synthetic_image(16, 16, 1, DType.U8, seed=1)
    b = rawio.synthetic_image(16, 16, 1, DType.U8, seed=2)
    assert not np.array_equal(a, b)
