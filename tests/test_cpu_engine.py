"""CPU engine vs oracle: single rank, local multi-block, property tests.

SURVEY.md §4 halo property test: conv of a block-decomposed image with halo
exchange must equal conv of the whole image — byte-exact for u8 with dyadic
filters.
"""

import numpy as np
import pytest

from parallel_convolution_amd.cpu_engine import run_local_blocks, run_single_cpu
from parallel_convolution_amd.models.filters import (
    GAUSSIAN_BLUR_3,
    GAUSSIAN_BLUR_5,
    SHARPEN_5,
)
from parallel_convolution_amd.ops.reference import run_reference
from parallel_convolution_amd.spec import DType, ProblemSpec
from parallel_convolution_amd.utils.rawio import synthetic_image


def test_config1_1024_gray_10iters():
    """BASELINE.json:7 config 1: 1024² gray, 3×3 Gaussian, 10 iters, 1 CPU
    rank — byte-exact vs oracle."""
    spec = ProblemSpec(1024, 1024, 1, DType.U8, GAUSSIAN_BLUR_3, 10)
    img = synthetic_image(1024, 1024, 1, DType.U8)
    out = run_single_cpu(spec, img)
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(out, ref)


@pytest.mark.parametrize("p", [2, 4, 8])
def test_local_blocks_match_oracle_u8(p):
    spec = ProblemSpec(96, 80, 1, DType.U8, GAUSSIAN_BLUR_3, 20)
    img = synthetic_image(80, 96, 1, DType.U8)
    out = run_local_blocks(spec, img, p)
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(out, ref)


@pytest.mark.parametrize("p", [4, 8])
def test_local_blocks_k5(p):
    spec = ProblemSpec(100, 72, 1, DType.U8, GAUSSIAN_BLUR_5, 9)
    img = synthetic_image(72, 100, 1, DType.U8)
    out = run_local_blocks(spec, img, p)
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(out, ref)


@pytest.mark.parametrize("p", [2, 8])
def test_local_blocks_rgb_sharpen(p):
    spec = ProblemSpec(64, 56, 3, DType.U8, SHARPEN_5, 6)
    img = synthetic_image(56, 64, 3, DType.U8)
    out = run_local_blocks(spec, img, p)
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(out, ref)


def test_local_blocks_f16_tolerance():
    # fp16 storage: order-of-summation LSB drift possible with a
    # non-dyadic filter; dyadic blur stays exact.
    spec = ProblemSpec(64, 64, 1, DType.F16, GAUSSIAN_BLUR_3, 10)
    img = synthetic_image(64, 64, 1, DType.F16)
    out = run_local_blocks(spec, img, 4)
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(out, ref)


def test_local_blocks_f32():
    # fp32 storage with real-valued inputs: products are inexact, so
    # summation order shifts the LSB — tolerance, not bitwise (spec.py).
    spec = ProblemSpec(64, 64, 1, DType.F32, GAUSSIAN_BLUR_3, 10)
    img = synthetic_image(64, 64, 1, DType.F32)
    out = run_local_blocks(spec, img, 4)
    ref = run_reference(spec, img)
    np.testing.assert_allclose(out, ref, rtol=1e-5, atol=1e-4)


def test_uneven_sizes_and_tiny_blocks():
    # block dims not divisible by grid, chunk T reduced by small blocks
    spec = ProblemSpec(37, 53, 1, DType.U8, GAUSSIAN_BLUR_3, 13)
    img = synthetic_image(53, 37, 1, DType.U8)
    out = run_local_blocks(spec, img, 8)
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(out, ref)


def test_zero_iters_identity():
    spec = ProblemSpec(16, 16, 1, DType.U8, GAUSSIAN_BLUR_3, 0)
    img = synthetic_image(16, 16, 1, DType.U8)
    out = run_single_cpu(spec, img)
    np.testing.assert_array_equal(out, img)


def test_random_float_filter_tolerance():
    """Non-dyadic filter: u8 results may differ at exact-.5 rounding
    boundaries between summation orders; require ≤1 gray-level and make
    sure it is rare."""
    rng = np.random.default_rng(7)
    filt = rng.random((3, 3), dtype=np.float32)
    filt /= filt.sum()
    spec = ProblemSpec(64, 64, 1, DType.U8, filt, 8)
    img = synthetic_image(64, 64, 1, DType.U8)
    out = run_local_blocks(spec, img, 4)
    ref = run_reference(spec, img)
    diff = np.abs(out.astype(np.int32) - ref.astype(np.int32))
    assert diff.max() <= 1
    assert (diff > 0).mean() < 0.01


def test_large_kernel_k35():
    """k=35 (radius 17 > default halo): halo_for_radius must widen the
    padding so nothing reads past the buffer."""
    rng = np.random.default_rng(5)
    filt = rng.random((35, 35), dtype=np.float32)
    filt /= filt.sum()
    spec = ProblemSpec(96, 80, 1, DType.F32, filt, 2)
    img = synthetic_image(80, 96, 1, DType.F32)
    out = run_local_blocks(spec, img, 2)
    ref = run_reference(spec, img)
    np.testing.assert_allclose(out, ref, rtol=1e-4, atol=1e-2)
