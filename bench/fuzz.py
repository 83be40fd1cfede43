"""Randomized correctness campaign: random problems vs the fp32 oracle.

Each trial draws a random (size, dtype, channels, filter, iterations,
block count) within engine limits, runs the decomposed GPU engine (local
D2D exchange — the full chunked/overlap/kernel-selection stack) and
compares against ops/reference.py: byte-exact for u8/f16 with dyadic
filters, tolerance otherwise.

    python bench/fuzz.py [minutes] [seed]
"""

import sys
import time

sys.path.insert(0, ".")
import numpy as np

from parallel_convolution_amd.hip_engine import run_local_blocks_gpu
from parallel_convolution_amd.models import filters as F
from parallel_convolution_amd.ops.reference import run_reference
from parallel_convolution_amd.spec import DType, ProblemSpec
from parallel_convolution_amd.utils.rawio import synthetic_image


def random_filter(rng):
    kind = rng.integers(0, 6)
    if kind == 0:
        return F.GAUSSIAN_BLUR_3, True
    if kind == 1:
        return F.GAUSSIAN_BLUR_5, True
    if kind == 2:
        return F.SHARPEN_5, True
    if kind == 3:
        # random dyadic non-negative k=3 (dot-path eligible shapes)
        n = rng.integers(0, 8, (3, 3)).astype(np.float32)
        n[1, 1] += 1
        shift = 2 ** int(np.ceil(np.log2(max(n.sum(), 1))))
        return (n / shift).astype(np.float32), True
    if kind == 4:
        # random separable non-negative dyadic (sep-path eligible:
        # sum(a)*sum(b) <= 256 keeps the u16 SWAR bound)
        k = int(rng.choice([3, 5]))
        a = rng.integers(0, 4, k)
        b = rng.integers(0, 4, k)
        a[k // 2] += 1
        b[k // 2] += 1
        n = np.outer(b, a).astype(np.float64)
        shift = 2 ** int(np.ceil(np.log2(max(n.sum(), 1))))
        return (n / shift).astype(np.float32), True
    k = int(rng.choice([3, 5, 7]))
    f = rng.random((k, k), dtype=np.float32)
    f /= f.sum()
    return f, False


def one_trial(rng, i):
    w = int(rng.integers(33, 700))
    h = int(rng.integers(33, 700))
    c = int(rng.choice([1, 1, 1, 3]))
    dt = DType(str(rng.choice(["u8", "u8", "fp16", "fp32"])))
    filt, dyadic = random_filter(rng)
    iters = int(rng.integers(1, 40))
    # expansive filters (sum|w| > 1, e.g. sharpen) amplify any 1-ulp
    # rounding difference by ~3x per iteration on fp16 (and overflow to
    # inf within ~10 iterations IN THE ORACLE TOO) — two correct
    # implementations legitimately diverge within a few iterations
    # (the u8 sharpen tests are byte-exact: integer re-quantization
    # stops fractional drift, so the kernel arithmetic is verified
    # there). Keep such fp16 trials to 3 iterations.
    if dt == DType.F16 and float(np.abs(filt).sum()) > 1.001:
        iters = min(iters, 3)
    p = int(rng.choice([1, 2, 3, 4, 6, 8]))
    grid_r = min(p, h)  # make_blocks raises on degenerate grids; skip them
    spec = ProblemSpec(w, h, c, dt, filt, iters)
    img = synthetic_image(h, w, c, dt, seed=int(rng.integers(0, 1 << 30)))
    try:
        out = run_local_blocks_gpu(spec, img, p)
    except ValueError:
        return "skip"
    ref = run_reference(spec, img)
    if dt != DType.U8 and not np.isfinite(ref.astype(np.float32)).all():
        return "diverged"  # problem blew up (legitimately) — nothing to compare
    if dt == DType.U8 and dyadic:
        ok = np.array_equal(out, ref)
        kind = "exact"
    elif dt == DType.U8:
        ok = np.abs(out.astype(int) - ref.astype(int)).max() <= 1
        kind = "u8±1"
    elif dt == DType.F16:
        ok = np.allclose(out.astype(np.float32), ref.astype(np.float32),
                         rtol=3e-3, atol=0.5)
        kind = "f16tol"
    else:
        ok = np.allclose(out, ref, rtol=1e-4, atol=1e-2)
        kind = "f32tol"
    if not ok:
        print(f"FAIL trial {i}: {w}x{h}x{c} {dt.value} k={spec.k} "
              f"iters={iters} p={p} ({kind})", flush=True)
        return "fail"
    return kind


def main():
    minutes = float(sys.argv[1]) if len(sys.argv) > 1 else 5.0
    seed = int(sys.argv[2]) if len(sys.argv) > 2 else 12345
    rng = np.random.default_rng(seed)
    t0 = time.time()
    stats = {}
    i = 0
    fails = 0
    while time.time() - t0 < minutes * 60:
        r = one_trial(rng, i)
        stats[r] = stats.get(r, 0) + 1
        fails += r == "fail"
        i += 1
    print(f"FUZZ: {i} trials, {fails} failures, breakdown {stats}", flush=True)
    return 1 if fails else 0


if __name__ == "__main__":
    sys.exit(main())
