"""Oracle sanity: our fp32 reference vs scipy.ndimage."""

import numpy as np
import pytest
import scipy.ndimage

from parallel_convolution_amd.models.filters import (
    GAUSSIAN_BLUR_3,
    SHARPEN_5,
    parse_filter_text,
)
from parallel_convolution_amd.ops.reference import run_reference
from parallel_convolution_amd.spec import DType, ProblemSpec
from parallel_convolution_amd.utils.rawio import synthetic_image


def scipy_iter(img_f32, filt, iters, quant):
    x = img_f32.astype(np.float64)
    for _ in range(iters):
        # our contract is cross-correlation: correlate, zero boundary
        x = scipy.ndimage.correlate(x, filt.astype(np.float64), mode="constant", cval=0.0)
        x = quant(x)
    return x


def test_oracle_matches_scipy_u8_gray():
    spec = ProblemSpec(64, 48, 1, DType.U8, GAUSSIAN_BLUR_3, 5)
    img = synthetic_image(48, 64, 1, DType.U8)
    ours = run_reference(spec, img)

    def quant(x):
        return np.clip(np.rint(x), 0, 255)

    ref = scipy_iter(img.astype(np.float64), spec.filter, 5, quant)
    assert ours.dtype == np.uint8
    np.testing.assert_array_equal(ours.astype(np.float64), ref)


def test_oracle_matches_scipy_f32_sharpen5():
    spec = ProblemSpec(40, 40, 1, DType.F32, SHARPEN_5, 3)
    img = synthetic_image(40, 40, 1, DType.F32)
    ours = run_reference(spec, img)
    ref = scipy_iter(img, spec.filter, 3, lambda x: x)
    np.testing.assert_allclose(ours, ref, rtol=1e-4, atol=1e-2)


def test_oracle_rgb_channels_independent():
    spec = ProblemSpec(16, 16, 3, DType.U8, GAUSSIAN_BLUR_3, 2)
    img = synthetic_image(16, 16, 3, DType.U8)
    out = run_reference(spec, img)
    # each channel must equal the gray-path result on that channel
    for c in range(3):
        chan = np.ascontiguousarray(img.reshape(16, 16, 3)[:, :, c])
        gspec = ProblemSpec(16, 16, 1, DType.U8, GAUSSIAN_BLUR_3, 2)
        gout = run_reference(gspec, chan)
        np.testing.assert_array_equal(out.reshape(16, 16, 3)[:, :, c], gout)


def test_filter_parse_roundtrip(tmp_path):
    from parallel_convolution_amd.models.filters import read_filter, write_filter

    p = tmp_path / "f.txt"
    write_filter(str(p), SHARPEN_5)
    f2 = read_filter(str(p))
    np.testing.assert_array_equal(f2, SHARPEN_5)
    f3 = parse_filter_text("# comment\n3\n0 0 0  0 1 0\n0 0 0\n")
    assert f3[1, 1] == 1.0 and f3.sum() == 1.0


def test_filter_parse_errors():
    with pytest.raises(ValueError):
        parse_filter_text("2\n1 1 1 1")
    with pytest.raises(ValueError):
        parse_filter_text("3\n1 2 3")
    with pytest.raises(ValueError):
        parse_filter_text("")
