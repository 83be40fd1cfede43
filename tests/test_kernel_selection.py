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


# ---- separable integer path (round 2 headline kernel) ----------------------

from parallel_convolution_amd.hip_engine import u8sep_weights  # noqa: E402


def test_sep_blur3_eligible():
    assert u8sep_weights(spec(GAUSSIAN_BLUR_3)) == [
        4, 1, 2, 1, 0, 0, 1, 2, 1, 0, 0
    ]


def test_sep_blur5_eligible():
    w = u8sep_weights(spec(GAUSSIAN_BLUR_5))
    assert w == [8, 1, 4, 6, 4, 1, 1, 4, 6, 4, 1]
    # u16 accumulator bound holds
    assert 255 * 16 * 16 + (1 << 7) <= 65535


def test_sep_identity_eligible():
    w = u8sep_weights(spec(IDENTITY_3))
    assert w and w[0] >= 1


def test_sep_ineligible_cases():
    assert u8sep_weights(spec(EDGE_3)) == []        # negative taps
    assert u8sep_weights(spec(SHARPEN_5)) == []     # negative taps
    assert u8sep_weights(spec(GAUSSIAN_BLUR_3, c=3)) == []  # RGB
    assert u8sep_weights(spec(GAUSSIAN_BLUR_3, dt=DType.F16)) == []
    # dyadic but NOT separable: blur3 with a dented center stays dyadic
    # (sum 15/16 <= 1) but is rank 2
    f = GAUSSIAN_BLUR_3.copy()
    f[1, 1] -= np.float32(1.0 / 16.0)
    assert u8sep_weights(spec(f)) == []
    assert u8dot_weights(spec(f)) != []  # ...but the dot path covers it


def test_sep_disable_env(monkeypatch):
    monkeypatch.setenv("PCONV_DISABLE_SEP", "1")
    assert u8sep_weights(spec(GAUSSIAN_BLUR_3)) == []


def _sep_simulate(img, w9, k):
    """Simulate the conv_u8sep_k integer arithmetic (horizontal udot4,
    u16 vertical SWAR, round-half-even fix) in numpy — the exactness
    argument of csrc/conv_kernels.hip conv_u8sep_k, testable on CPU."""
    shift = w9[0]
    a = np.array(w9[1 : 1 + k], dtype=np.uint32)
    b = np.array(w9[6 : 6 + k], dtype=np.uint32)
    r = k // 2
    p = np.pad(img.astype(np.uint32), r)
    H, W = img.shape
    # horizontal pass on all input rows
    h = np.zeros((H + 2 * r, W), dtype=np.uint32)
    for i in range(k):
        h += a[i] * p[:, i : i + W]
    assert h.max() < 1 << 16
    # vertical pass + rounding
    acc = np.full((H, W), (1 << (shift - 1)) - 1, dtype=np.uint32)
    for j in range(k):
        acc += b[j] * h[j : j + H, :]
    assert acc.max() < 1 << 16
    t = (acc >> shift) & 1
    q = (acc + t) >> shift
    assert q.max() <= 255
    return q.astype(np.uint8)


@pytest.mark.parametrize("filt,k", [(GAUSSIAN_BLUR_3, 3), (GAUSSIAN_BLUR_5, 5)])
def test_sep_integer_arithmetic_matches_oracle(filt, k):
    from parallel_convolution_amd.ops.reference import run_reference
    from parallel_convolution_amd.utils.rawio import synthetic_image

    s = ProblemSpec(96, 64, 1, DType.U8, filt, 1)
    img = synthetic_image(64, 96, 1, DType.U8, seed=17)
    w9 = u8sep_weights(s)
    assert w9
    got = _sep_simulate(img, w9, k)
    ref = run_reference(s, img)
    np.testing.assert_array_equal(got, ref)


def test_sep_rounding_edge_values():
    """All 256^3 vertical-window values would be slow; sweep the full
    s-range for blur3 instead: every achievable integer sum s in
    [0, 255*16] must round exactly like fp32 RNE."""
    shift = 4
    s = np.arange(0, 255 * 16 + 1, dtype=np.uint32)
    v = s + (1 << (shift - 1)) - 1
    t = (v >> shift) & 1
    q = (v + t) >> shift
    ref = np.rint(s.astype(np.float64) / (1 << shift)).astype(np.uint32)
    np.testing.assert_array_equal(q, ref)


def test_sep_factorization_property():
    """500 random cases: whenever u8sep_weights accepts, the returned
    integer factorization must reconstruct the filter EXACTLY and satisfy
    every kernel precondition (u16 SWAR bound, byte-sized horizontal
    taps); random separable non-negative dyadic filters within bounds
    must always be accepted."""
    rng = np.random.default_rng(42)
    accepted = 0
    for trial in range(500):
        k = int(rng.choice([3, 5]))
        mode = rng.integers(0, 3)
        if mode == 0:
            # separable dyadic within bounds
            a = rng.integers(0, 4, k)
            b = rng.integers(0, 4, k)
            a[k // 2] += 1
            b[k // 2] += 1
            n = np.outer(b, a).astype(np.float64)
            shift = 2 ** int(np.ceil(np.log2(max(n.sum(), 1))))
            f = (n / shift).astype(np.float32)
            must_accept = 255 * int(a.sum()) * int(b.sum()) + shift // 2 <= 65535
        elif mode == 1:
            # random dyadic, usually rank>1
            n = rng.integers(0, 5, (k, k)).astype(np.float64)
            n[k // 2, k // 2] += 1
            f = (n / 2 ** int(np.ceil(np.log2(max(n.sum(), 1))))).astype(np.float32)
            must_accept = None
        else:
            f = rng.random((k, k), dtype=np.float32)
            f /= f.sum()
            must_accept = None
        s = ProblemSpec(32, 32, 1, DType.U8, f, 1)
        w = u8sep_weights(s)
        if must_accept and not w:
            raise AssertionError(f"trial {trial}: in-bounds separable rejected: {f}")
        if w:
            accepted += 1
            shift_, a_, b_ = w[0], np.array(w[1:6][:k]), np.array(w[6:11][:k])
            assert (a_ >= 0).all() and (b_ >= 0).all()
            assert a_.max() <= 255
            assert 255 * int(a_.sum()) * int(b_.sum()) + (1 << (shift_ - 1)) <= 65535
            np.testing.assert_array_equal(
                np.outer(b_, a_) / np.float64(1 << shift_),
                np.asarray(f, dtype=np.float64),
            )
    assert accepted >= 100  # the separable class must actually be exercised
