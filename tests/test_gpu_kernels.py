"""HIP kernel + engine numerics on a real MI355X (pytest -m gpu).

Every test compares against the plain PyTorch fp32 oracle
(ops/reference.py) — byte-exact for u8/f16 with dyadic filters per the
spec.py contract, tolerance otherwise.
"""

import numpy as np
import pytest
import torch

from parallel_convolution_amd.models.filters import (
    EDGE_3,
    GAUSSIAN_BLUR_3,
    GAUSSIAN_BLUR_5,
    SHARPEN_5,
)
from parallel_convolution_amd.ops.reference import run_reference
from parallel_convolution_amd.spec import DType, ProblemSpec
from parallel_convolution_amd.utils.rawio import synthetic_image

pytestmark = pytest.mark.gpu


def _engine_mod():
    from parallel_convolution_amd import hip_engine

    return hip_engine


@pytest.fixture(scope="module", autouse=True)
def _need_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")


def assert_matches(spec, out, ref):
    if spec.dtype == DType.U8:
        np.testing.assert_array_equal(out, ref)
    elif spec.dtype == DType.F16:
        np.testing.assert_allclose(
            out.astype(np.float32), ref.astype(np.float32), rtol=2e-3, atol=2e-1
        )
    else:
        np.testing.assert_allclose(out, ref, rtol=1e-5, atol=1e-3)


CASES = [
    # (W, H, C, dtype, filter, iters)
    (512, 384, 1, DType.U8, GAUSSIAN_BLUR_3, 10),
    (256, 256, 1, DType.U8, GAUSSIAN_BLUR_5, 6),
    (256, 200, 1, DType.F16, GAUSSIAN_BLUR_3, 10),
    (256, 200, 1, DType.F32, GAUSSIAN_BLUR_3, 10),
    (200, 160, 3, DType.U8, GAUSSIAN_BLUR_3, 5),
    (200, 160, 3, DType.U8, SHARPEN_5, 5),
    (192, 128, 3, DType.F32, SHARPEN_5, 4),
    (131, 97, 1, DType.U8, GAUSSIAN_BLUR_3, 7),  # odd sizes, tile clipping
]


@pytest.mark.parametrize("W,H,C,dt,filt,iters", CASES)
def test_single_gpu_matches_oracle(W, H, C, dt, filt, iters):
    spec = ProblemSpec(W, H, C, dt, filt, iters)
    img = synthetic_image(H, W, C, dt)
    out = _engine_mod().run_single_gpu(spec, img)
    ref = run_reference(spec, img)
    assert_matches(spec, out, ref)


@pytest.mark.parametrize("p", [2, 4, 8])
def test_local_blocks_gpu_protocol(p):
    """Multi-block D2D exchange on one GPU == whole-image oracle. This is
    the single-box stand-in for the 8-GPU RCCL run (same engine loop, same
    pack geometry; only ncclSend/Recv replaced by D2D copies)."""
    spec = ProblemSpec(96, 80, 1, DType.U8, GAUSSIAN_BLUR_3, 20)
    img = synthetic_image(80, 96, 1, DType.U8)
    out = _engine_mod().run_local_blocks_gpu(spec, img, p)
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(out, ref)


@pytest.mark.parametrize("p", [4, 8])
def test_local_blocks_gpu_rgb_k5(p):
    spec = ProblemSpec(128, 96, 3, DType.U8, SHARPEN_5, 8)
    img = synthetic_image(96, 128, 3, DType.U8)
    out = _engine_mod().run_local_blocks_gpu(spec, img, p)
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(out, ref)


def test_local_blocks_gpu_uneven():
    spec = ProblemSpec(237, 101, 1, DType.U8, GAUSSIAN_BLUR_3, 11)
    img = synthetic_image(101, 237, 1, DType.U8)
    out = _engine_mod().run_local_blocks_gpu(spec, img, 8)
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(out, ref)


def test_generic_kernel_k7():
    rng = np.random.default_rng(3)
    filt = rng.random((7, 7), dtype=np.float32)
    filt /= filt.sum()
    spec = ProblemSpec(128, 96, 1, DType.F32, filt, 3)
    img = synthetic_image(96, 128, 1, DType.F32)
    out = _engine_mod().run_single_gpu(spec, img)
    ref = run_reference(spec, img)
    np.testing.assert_allclose(out, ref, rtol=1e-4, atol=1e-2)


def test_edge_filter_negative_values():
    # EDGE_3 drives accumulators negative: exercises the u8 clamp-low path
    spec = ProblemSpec(128, 128, 1, DType.U8, EDGE_3, 3)
    img = synthetic_image(128, 128, 1, DType.U8)
    out = _engine_mod().run_single_gpu(spec, img)
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(out, ref)


def test_diff_max():
    from parallel_convolution_amd.hip_engine import HipEngine
    from parallel_convolution_amd.parallel.decomp import make_blocks

    spec = ProblemSpec(64, 64, 1, DType.U8, GAUSSIAN_BLUR_3, 1)
    img = synthetic_image(64, 64, 1, DType.U8)
    blocks = make_blocks(64, 64, 1, 1)
    eng = HipEngine(spec, blocks)
    eng.load_block(0, img)
    eng.run(1)
    d = eng.diff_max()
    ref0 = run_reference(spec, img, iters=0).astype(np.int32)
    ref1 = run_reference(spec, img, iters=1).astype(np.int32)
    assert d == float(np.abs(ref1 - ref0).max())


def test_identity_filter_many_iters():
    from parallel_convolution_amd.models.filters import IDENTITY_3

    spec = ProblemSpec(96, 64, 1, DType.U8, IDENTITY_3, 50)
    img = synthetic_image(64, 96, 1, DType.U8)
    out = _engine_mod().run_local_blocks_gpu(spec, img, 4)
    np.testing.assert_array_equal(out, img)


def test_native_extension_is_loaded():
    """Guard against a silent eager/PyTorch fallback: the in-tree _C.so
    must be the module that ran."""
    import parallel_convolution_amd._C as C

    assert C.HIP_BUILT_FOR == "gfx950"
    assert "/parallel_convolution_amd/" in C.__file__


def test_rccl_world1_init_and_run():
    """RCCL communicator creation + engine run on a 1-GPU world: validates
    the NCCL-mode code path (comm init, exchange skip) on a real box."""
    import parallel_convolution_amd._C as C
    from parallel_convolution_amd.hip_engine import HipEngine
    from parallel_convolution_amd.parallel.decomp import make_blocks

    spec = ProblemSpec(128, 96, 1, DType.U8, GAUSSIAN_BLUR_3, 6)
    img = synthetic_image(96, 128, 1, DType.U8)
    blocks = make_blocks(96, 128, 1, 1)
    eng = HipEngine(spec, blocks)
    uid = C.get_nccl_unique_id()
    eng.engine.init_nccl(uid, 0, 1)
    eng.load_block(0, img)
    eng.run()
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(eng.store_block(0), ref)


def test_cli_hip_backend(tmp_path):
    """Reference CLI end-to-end on the GPU (hip backend)."""
    import subprocess
    import sys
    import os

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    img = synthetic_image(64, 96, 1, DType.U8)
    ip = str(tmp_path / "img.raw")
    img.tofile(ip)
    out = str(tmp_path / "out.raw")
    env = dict(os.environ)
    env["PYTHONPATH"] = repo + os.pathsep + env.get("PYTHONPATH", "")
    r = subprocess.run(
        [sys.executable, "-m", "parallel_convolution_amd", ip, "96", "64",
         "blur3", "8", "--backend", "hip", "--output", out],
        cwd=repo, env=env, capture_output=True, text=True, timeout=300,
    )
    assert r.returncode == 0, r.stderr
    got = np.fromfile(out, dtype=np.uint8).reshape(64, 96)
    spec = ProblemSpec(96, 64, 1, DType.U8, GAUSSIAN_BLUR_3, 8)
    np.testing.assert_array_equal(got, run_reference(spec, img))


def test_mfma_probe_layout():
    """Verify the assumed mfma_f32_16x16x32_f16 fragment layouts with
    asymmetric A and B (catches any transposed mapping)."""
    rng = np.random.default_rng(11)
    # small integers: exact in fp16, exact fp32 accumulation
    A = torch.tensor(rng.integers(-8, 8, (16, 32)), dtype=torch.float32, device="cuda")
    B = torch.tensor(rng.integers(-8, 8, (32, 16)), dtype=torch.float32, device="cuda")
    import parallel_convolution_amd._C as C

    D = C.mfma_probe(A, B)
    ref = (A.cpu() @ B.cpu())
    np.testing.assert_array_equal(D.cpu().numpy(), ref.numpy())


def test_mfma_engine_selected_and_byte_exact():
    """MFMA and VALU paths must agree bitwise with the fp32 oracle for a
    fp16-exact filter (blur3); k=5 filters select MFMA by default."""
    import os

    from parallel_convolution_amd.hip_engine import mfma_eligible

    assert mfma_eligible(ProblemSpec(64, 64, 1, DType.U8, GAUSSIAN_BLUR_5, 1))
    assert mfma_eligible(ProblemSpec(64, 64, 3, DType.U8, SHARPEN_5, 1))
    spec = ProblemSpec(512, 384, 1, DType.U8, GAUSSIAN_BLUR_3, 10)
    img = synthetic_image(384, 512, 1, DType.U8)
    ref = run_reference(spec, img)
    # the integer sep/dot paths outrank MFMA in launch_region — disable
    # both so FORCE_MFMA really exercises conv_mfma_k (without this the
    # test silently reran the integer kernel)
    os.environ["PCONV_FORCE_MFMA"] = "1"
    os.environ["PCONV_DISABLE_SEP"] = "1"
    os.environ["PCONV_DISABLE_DOT"] = "1"
    try:
        assert mfma_eligible(spec)
        eng_mod = _engine_mod()
        from parallel_convolution_amd.parallel.decomp import make_blocks

        probe = eng_mod.HipEngine(spec, make_blocks(384, 512, 1, 1))
        assert probe.kernel_path() == "mfma-f16"
        del probe
        out_mfma = eng_mod.run_single_gpu(spec, img)
    finally:
        del os.environ["PCONV_FORCE_MFMA"]
        del os.environ["PCONV_DISABLE_SEP"]
        del os.environ["PCONV_DISABLE_DOT"]
    np.testing.assert_array_equal(out_mfma, ref)
    os.environ["PCONV_DISABLE_MFMA"] = "1"
    os.environ["PCONV_DISABLE_SEP"] = "1"
    os.environ["PCONV_DISABLE_DOT"] = "1"
    os.environ["PCONV_FUSE"] = "0"
    try:
        out_valu = _engine_mod().run_single_gpu(spec, img)
    finally:
        for v in ("PCONV_DISABLE_MFMA", "PCONV_DISABLE_SEP",
                  "PCONV_DISABLE_DOT", "PCONV_FUSE"):
            del os.environ[v]
    np.testing.assert_array_equal(out_valu, out_mfma)


@pytest.mark.parametrize("W,H,C,dt,filt,iters", [
    (256, 256, 1, DType.U8, GAUSSIAN_BLUR_5, 6),
    (256, 200, 1, DType.F16, GAUSSIAN_BLUR_3, 10),
    (200, 160, 3, DType.U8, GAUSSIAN_BLUR_3, 5),
    (200, 160, 3, DType.U8, SHARPEN_5, 5),
    (131, 97, 1, DType.U8, GAUSSIAN_BLUR_3, 7),
])
def test_mfma_variants_match_oracle(W, H, C, dt, filt, iters):
    """Every (dtype, k, channels) MFMA variant vs the oracle (forced on)."""
    import os

    spec = ProblemSpec(W, H, C, dt, filt, iters)
    img = synthetic_image(H, W, C, dt)
    os.environ["PCONV_FORCE_MFMA"] = "1"
    os.environ["PCONV_DISABLE_SEP"] = "1"  # sep/dot outrank MFMA
    os.environ["PCONV_DISABLE_DOT"] = "1"
    try:
        out = _engine_mod().run_single_gpu(spec, img)
    finally:
        for v in ("PCONV_FORCE_MFMA", "PCONV_DISABLE_SEP", "PCONV_DISABLE_DOT"):
            del os.environ[v]
    ref = run_reference(spec, img)
    assert_matches(spec, out, ref)


def test_mfma_multiblock_protocol():
    import os

    spec = ProblemSpec(96, 80, 1, DType.U8, GAUSSIAN_BLUR_3, 20)
    img = synthetic_image(80, 96, 1, DType.U8)
    os.environ["PCONV_FORCE_MFMA"] = "1"
    os.environ["PCONV_DISABLE_SEP"] = "1"  # sep/dot outrank MFMA
    os.environ["PCONV_DISABLE_DOT"] = "1"
    try:
        out = _engine_mod().run_local_blocks_gpu(spec, img, 8)
    finally:
        for v in ("PCONV_FORCE_MFMA", "PCONV_DISABLE_SEP", "PCONV_DISABLE_DOT"):
            del os.environ[v]
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(out, ref)


def test_local_blocks_gpu_f32():
    spec = ProblemSpec(128, 96, 1, DType.F32, GAUSSIAN_BLUR_3, 12)
    img = synthetic_image(96, 128, 1, DType.F32)
    out = _engine_mod().run_local_blocks_gpu(spec, img, 4)
    ref = run_reference(spec, img)
    np.testing.assert_allclose(out, ref, rtol=1e-5, atol=1e-3)


def test_phase_timing_accumulates():
    from parallel_convolution_amd.hip_engine import HipEngine
    from parallel_convolution_amd.parallel.decomp import make_blocks

    spec = ProblemSpec(256, 128, 1, DType.U8, GAUSSIAN_BLUR_3, 20)
    img = synthetic_image(128, 256, 1, DType.U8)
    blocks = make_blocks(128, 256, 2, 1)
    eng = HipEngine(spec, blocks)
    eng.set_phase_timing(True)
    for i, bs in enumerate(blocks):
        eng.load_block(i, img[bs.y0 : bs.y0 + bs.h, bs.x0 : bs.x0 + bs.w])
    eng.run()
    pt = eng.phase_times()
    assert pt["chunks"] >= 1
    assert pt["compute_ms"] > 0
    assert pt["exchange_ms"] >= 0


def test_cli_hip_checkpoint_resume(tmp_path):
    import subprocess
    import sys
    import os

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    img = synthetic_image(96, 128, 1, DType.U8)
    ip = str(tmp_path / "img.raw")
    img.tofile(ip)
    out = str(tmp_path / "out.raw")
    env = dict(os.environ)
    env["PYTHONPATH"] = repo + os.pathsep + env.get("PYTHONPATH", "")
    def cmd(iters, extra):
        return [sys.executable, "-m", "parallel_convolution_amd", ip, "128",
                "96", "blur3", str(iters), "--backend", "hip", "--output",
                out] + extra

    r = subprocess.run(cmd(6, ["--checkpoint-every", "6"]), cwd=repo, env=env,
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr
    r = subprocess.run(cmd(14, ["--checkpoint-every", "6", "--resume"]),
                       cwd=repo, env=env, capture_output=True, text=True,
                       timeout=300)
    assert r.returncode == 0, r.stderr
    got = np.fromfile(out, dtype=np.uint8).reshape(96, 128)
    spec = ProblemSpec(128, 96, 1, DType.U8, GAUSSIAN_BLUR_3, 14)
    np.testing.assert_array_equal(got, run_reference(spec, img))


def test_generic_kernel_k35_gpu():
    """Generic-kernel fallback with radius 17 (> default halo): the halo
    must widen so the kernel never reads outside the padded buffer."""
    rng = np.random.default_rng(5)
    filt = rng.random((35, 35), dtype=np.float32)
    filt /= filt.sum()
    spec = ProblemSpec(128, 96, 1, DType.F32, filt, 2)
    img = synthetic_image(96, 128, 1, DType.F32)
    out = _engine_mod().run_single_gpu(spec, img)
    ref = run_reference(spec, img)
    np.testing.assert_allclose(out, ref, rtol=1e-4, atol=1e-2)


@pytest.mark.parametrize("extra,dt,C", [
    (["--dtype", "fp16"], DType.F16, 1),
    (["--rgb"], DType.U8, 3),
])
def test_cli_hip_dtype_variants(tmp_path, extra, dt, C):
    import subprocess
    import sys
    import os

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    img = synthetic_image(48, 64, C, dt)
    ip = str(tmp_path / "img.raw")
    img.tofile(ip)
    out = str(tmp_path / "out.raw")
    env = dict(os.environ)
    env["PYTHONPATH"] = repo + os.pathsep + env.get("PYTHONPATH", "")
    r = subprocess.run(
        [sys.executable, "-m", "parallel_convolution_amd", ip, "64", "48",
         "blur3", "6", "--backend", "hip", "--output", out] + extra,
        cwd=repo, env=env, capture_output=True, text=True, timeout=300,
    )
    assert r.returncode == 0, r.stderr
    got = np.fromfile(out, dtype=dt.numpy_dtype).reshape(48, 64 * C)
    spec = ProblemSpec(64, 48, C, dt, GAUSSIAN_BLUR_3, 6)
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(got, ref)


@pytest.mark.parametrize("p", [1, 4])
def test_fused_double_iteration_byte_exact(p):
    """Fused two-iterations-per-launch path (PCONV_FUSE=1) must agree
    bitwise with the oracle for u8 (mid image quantized per iteration)."""
    import os

    spec = ProblemSpec(96, 80, 1, DType.U8, GAUSSIAN_BLUR_3, 21)  # odd iters
    img = synthetic_image(80, 96, 1, DType.U8)
    os.environ["PCONV_FUSE"] = "1"
    try:
        out = _engine_mod().run_local_blocks_gpu(spec, img, p)
    finally:
        del os.environ["PCONV_FUSE"]
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(out, ref)


def test_fused_rgb_k5():
    import os

    spec = ProblemSpec(128, 96, 3, DType.U8, SHARPEN_5, 7)
    img = synthetic_image(96, 128, 3, DType.U8)
    os.environ["PCONV_FUSE"] = "1"
    os.environ["PCONV_DISABLE_MFMA"] = "1"
    try:
        out = _engine_mod().run_local_blocks_gpu(spec, img, 4)
    finally:
        del os.environ["PCONV_FUSE"]
        del os.environ["PCONV_DISABLE_MFMA"]
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(out, ref)


def test_fused_f16_tie_rounding_consistency():
    """fp16 exact-tie regression: the fused path's LDS quantizer must
    round exactly like the store path (packed v_cvt_pk_f16_f32 == torch
    round-half-even). Inputs are crafted so iteration 1 produces exact
    fp16 ties: w = [.5, .5] row filter over alternating values whose mean
    sits exactly between two fp16 neighbors."""
    import os

    filt = np.zeros((3, 3), dtype=np.float32)
    filt[1, 1] = 0.5
    filt[1, 2] = 0.5
    a, b = 1.0, 1.0 + 3 * 2.0**-10  # mean = 1 + 1.5*2^-10: exact f16 tie
    img = np.empty((64, 64), dtype=np.float16)
    img[:, 0::2] = a
    img[:, 1::2] = b
    spec = ProblemSpec(64, 64, 1, DType.F16, filt, 2)
    ref = run_reference(spec, img)
    outs = {}
    for fuse in ("0", "1"):
        os.environ["PCONV_FUSE"] = fuse
        try:
            outs[fuse] = _engine_mod().run_single_gpu(spec, img)
        finally:
            del os.environ["PCONV_FUSE"]
    np.testing.assert_array_equal(outs["0"], outs["1"])
    np.testing.assert_array_equal(outs["0"], ref)


@pytest.mark.parametrize("p", [1, 4])
def test_fused_dot_byte_exact(p):
    """Fused integer-dot pair path (PCONV_DOT_FUSE=1) bitwise vs oracle,
    odd iteration count, multi-block exchange."""
    import os

    spec = ProblemSpec(96, 80, 1, DType.U8, GAUSSIAN_BLUR_3, 21)
    img = synthetic_image(80, 96, 1, DType.U8)
    # PCONV_DISABLE_SEP: since r02 the separable path outranks the dot
    # path for blur3 and disables dot fusion — without this the test
    # would silently exercise conv_u8sep_k instead of conv_u8dot2_k
    os.environ["PCONV_DOT_FUSE"] = "1"
    os.environ["PCONV_DISABLE_SEP"] = "1"
    try:
        out = _engine_mod().run_local_blocks_gpu(spec, img, p)
    finally:
        del os.environ["PCONV_DOT_FUSE"]
        del os.environ["PCONV_DISABLE_SEP"]
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(out, ref)


# ---- separable integer path (round-2 headline kernel) ----------------------


def test_sep_selected_for_headline_filter():
    from parallel_convolution_amd.hip_engine import HipEngine
    from parallel_convolution_amd.parallel.decomp import make_blocks

    spec = ProblemSpec(64, 64, 1, DType.U8, GAUSSIAN_BLUR_3, 1)
    eng = HipEngine(spec, make_blocks(64, 64, 1, 1))
    assert eng.kernel_path() == "u8sep"
    spec5 = ProblemSpec(64, 64, 1, DType.U8, GAUSSIAN_BLUR_5, 1)
    eng5 = HipEngine(spec5, make_blocks(64, 64, 1, 1))
    assert eng5.kernel_path() == "u8sep"


@pytest.mark.parametrize("W,H,filt,iters", [
    (512, 384, GAUSSIAN_BLUR_3, 10),
    (1031, 517, GAUSSIAN_BLUR_3, 7),   # odd sizes, 1024-wide tile clipping
    (512, 384, GAUSSIAN_BLUR_5, 6),    # k=5 separable path
    (1031, 517, GAUSSIAN_BLUR_5, 5),
    (900, 40, GAUSSIAN_BLUR_3, 3),     # fewer rows than one tile chain
])
def test_sep_matches_oracle(W, H, filt, iters):
    spec = ProblemSpec(W, H, 1, DType.U8, filt, iters)
    img = synthetic_image(H, W, 1, DType.U8)
    out = _engine_mod().run_single_gpu(spec, img)
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(out, ref)


@pytest.mark.parametrize("p,filt", [(2, GAUSSIAN_BLUR_3), (8, GAUSSIAN_BLUR_3),
                                    (4, GAUSSIAN_BLUR_5), (8, GAUSSIAN_BLUR_5)])
def test_sep_multiblock_protocol(p, filt):
    """Halo-deepened chunked exchange + sep kernel == oracle (the border
    strip regions exercise narrow-region launches of the 1024-wide tile)."""
    spec = ProblemSpec(160, 144, 1, DType.U8, filt, 20)
    img = synthetic_image(144, 160, 1, DType.U8)
    out = _engine_mod().run_local_blocks_gpu(spec, img, p)
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(out, ref)


def test_sep_vs_dot_identical(monkeypatch):
    """A/B: the separable kernel and the non-separable dot kernel are
    bit-identical on the headline filter (both exact integer RNE)."""
    spec = ProblemSpec(640, 480, 1, DType.U8, GAUSSIAN_BLUR_3, 9)
    img = synthetic_image(480, 640, 1, DType.U8, seed=23)
    out_sep = _engine_mod().run_single_gpu(spec, img)
    monkeypatch.setenv("PCONV_DISABLE_SEP", "1")
    out_dot = _engine_mod().run_single_gpu(spec, img)
    np.testing.assert_array_equal(out_sep, out_dot)


# ---- exchange-overlap pyramid (round-2 scheduler) --------------------------


@pytest.mark.parametrize("depth", [1, 2, 4])
@pytest.mark.parametrize("p", [2, 8])
def test_overlap_depth_byte_exact(depth, p, monkeypatch):
    """The m-deep interior/band overlap schedule must be bit-identical to
    the oracle for every depth: interiors of iterations 0..m-1 run while
    the exchange flies, their border bands replay afterwards against the
    historical buffer pairs."""
    monkeypatch.setenv("PCONV_OVERLAP_DEPTH", str(depth))
    spec = ProblemSpec(160, 144, 1, DType.U8, GAUSSIAN_BLUR_3, 21)
    img = synthetic_image(144, 160, 1, DType.U8, seed=7)
    out = _engine_mod().run_local_blocks_gpu(spec, img, p)
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(out, ref)


def test_overlap_depth_fp16_bit_reproducible(monkeypatch):
    """f16 storage: multi-block with bands must match the 1-block run
    bitwise (the generic strip kernel rounds via the same packed cvt)."""
    spec = ProblemSpec(256, 192, 1, DType.F16, GAUSSIAN_BLUR_3, 12)
    img = synthetic_image(192, 256, 1, DType.F16, seed=9)
    ref1 = _engine_mod().run_single_gpu(spec, img)
    monkeypatch.setenv("PCONV_OVERLAP_DEPTH", "3")
    out = _engine_mod().run_local_blocks_gpu(spec, img, 4)
    np.testing.assert_array_equal(
        out.view(np.uint16), ref1.view(np.uint16)
    )
