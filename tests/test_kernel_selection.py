"""Kernel-path selection logic (host-side, no GPU needed)."""

import numpy as np
import pytest

from parallel_convolution_amd.hip_engine import mfma_eligible, u8dot_weights
from parallel_convolution_amd.models.filters import (
    EDGE_3,
    GAUSSIAN_BLUR_3,
    GAUSSIAN_BLUR_5,
    IDENTITY_3,
    SHARPEN_5,
)
from parallel_convolution_amd.spec import DType, ProblemSpec


def spec(filt, dt=DType.U8, c=1):
    return ProblemSpec(64, 64, c, dt, filt, 1)


def test_dot_blur3_eligible():
    assert u8dot_weights(spec(GAUSSIAN_BLUR_3)) == [4, 1, 2, 1, 2, 4, 2, 1, 2, 1]


def test_dot_identity_eligible():
    w = u8dot_weights(spec(IDENTITY_3))
    assert w[0] >= 1 and sum(w[1:]) == (1 << w[0])


def test_dot_ineligible_cases():
    assert u8dot_weights(spec(EDGE_3)) == []  # negative taps
    assert u8dot_weights(spec(GAUSSIAN_BLUR_3, dt=DType.F16)) == []
    assert u8dot_weights(spec(GAUSSIAN_BLUR_5)) == []  # k=5 (no dot template)
    rng = np.random.default_rng(1)
    f = rng.random((3, 3), dtype=np.float32)
    f /= f.sum()
    assert u8dot_weights(spec(f)) == []  # non-dyadic
    # sum > 2^shift (unnormalized smoothing): gain > 1 would need clamping
    f2 = GAUSSIAN_BLUR_3 * np.float32(1.5)
    assert u8dot_weights(spec(f2)) == []


def test_dot_env_override(monkeypatch):
    monkeypatch.setenv("PCONV_DISABLE_DOT", "1")
    assert u8dot_weights(spec(GAUSSIAN_BLUR_3)) == []


def test_mfma_policy():
    # k=5 fp16-exact filters select MFMA; k=3 defaults to VALU/dot
    assert mfma_eligible(spec(GAUSSIAN_BLUR_5))
    assert mfma_eligible(spec(SHARPEN_5, c=3))
    assert not mfma_eligible(spec(GAUSSIAN_BLUR_3))
    assert not mfma_eligible(spec(GAUSSIAN_BLUR_5, dt=DType.F32))
    rng = np.random.default_rng(2)
    f = rng.random((5, 5), dtype=np.float32)  # not fp16-exact
    f /= f.sum()
    assert not mfma_eligible(spec(f))


def test_mfma_force(monkeypatch):
    monkeypatch.setenv("PCONV_FORCE_MFMA", "1")
    assert mfma_eligible(spec(GAUSSIAN_BLUR_3))
    monkeypatch.setenv("PCONV_DISABLE_MFMA", "1")
    assert not mfma_eligible(spec(GAUSSIAN_BLUR_3))
