"""Property-based tests (hypothesis): decomposition invariants and the
halo property over random problem shapes."""

import numpy as np
from hypothesis import given, settings, strategies as st

from parallel_convolution_amd.cpu_engine import run_local_blocks, run_single_cpu
from parallel_convolution_amd.models.filters import GAUSSIAN_BLUR_3, GAUSSIAN_BLUR_5
from parallel_convolution_amd.ops.reference import run_reference
from parallel_convolution_amd.parallel.decomp import (
    make_blocks,
    max_chunk_iters,
    plan_chunks,
)
from parallel_convolution_amd.spec import DType, ProblemSpec


@given(
    h=st.integers(1, 300),
    w=st.integers(1, 300),
    p=st.integers(1, 16),
)
@settings(max_examples=200, deadline=None)
def test_decomposition_partitions_exactly(h, w, p):
    try:
        blocks = make_blocks(h, w, p)
    except ValueError:
        # image smaller than the process grid: rejected by design
        from parallel_convolution_amd.parallel.decomp import factor_grid

        R, C = factor_grid(p)
        assert h < R or w < C
        return
    assert sum(b.h * b.w for b in blocks) == h * w
    # neighbor symmetry
    by = {b.rank: b for b in blocks}
    for b in blocks:
        for side, opp in (("north", "south"), ("west", "east")):
            n = getattr(b, side)
            if n is not None:
                assert getattr(by[n], opp) == b.rank


@given(
    iters=st.integers(0, 60),
    radius=st.integers(0, 3),
    chunk=st.integers(1, 20),
)
@settings(max_examples=200, deadline=None)
def test_chunk_plans_cover_iters(iters, radius, chunk):
    plans = plan_chunks(iters, radius, chunk)
    assert sum(p.iters for p in plans) == iters
    for p in plans:
        assert p.depth == radius * p.iters
        for t in range(p.iters):
            assert p.ext(t) >= 0
        if p.iters:
            assert p.ext(p.iters - 1) == 0


@given(
    w=st.integers(20, 90),
    h=st.integers(20, 90),
    p=st.sampled_from([2, 3, 4, 6, 8]),
    iters=st.integers(1, 12),
    k5=st.booleans(),
)
@settings(max_examples=25, deadline=None)
def test_halo_property_random_shapes(w, h, p, iters, k5):
    """Block-decomposed run == whole-image oracle, byte-exact, for random
    shapes/rank counts/iteration counts."""
    filt = GAUSSIAN_BLUR_5 if k5 else GAUSSIAN_BLUR_3
    spec = ProblemSpec(w, h, 1, DType.U8, filt, iters)
    rng = np.random.default_rng(w * 1000 + h * 10 + p)
    img = rng.integers(0, 256, size=(h, w), dtype=np.uint8)
    out = run_local_blocks(spec, img, p)
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(out, ref)


def test_k1_filter():
    """1×1 filter (radius 0): no exchange at all, pure scaling."""
    filt = np.array([[0.5]], dtype=np.float32)
    spec = ProblemSpec(40, 32, 1, DType.U8, filt, 3)
    img = np.full((32, 40), 200, dtype=np.uint8)
    out = run_local_blocks(spec, img, 4)
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(out, ref)
    out1 = run_single_cpu(spec, img)
    np.testing.assert_array_equal(out1, ref)
