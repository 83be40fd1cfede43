"""Per-rank-shape timing (the 8-GPU launch-overhead regime).

The 2x4-decomposed headline block is 8192 rows x 4096 px per rank — 8x
less work per launch than the 1-GPU bench. This measures, on one GPU:

  1. pure compute at the per-rank shape (no neighbors, real chunked
     engine loop) — the per-rank ceiling the 8-GPU scaling point divides
     against, and the µs/iter the exchange must hide inside;
  2. a chunk-T sweep at that shape;
  3. the LOCAL-mode 2x4 decomposition (8 blocks on this GPU, D2D
     exchange standing in for RCCL): schedule + pack overhead per iter.

Usage: python bench/perrank.py [--iters N]
"""

import argparse
import sys
import time

sys.path.insert(0, ".")
import torch  # noqa: E402

from parallel_convolution_amd.hip_engine import HipEngine  # noqa: E402
from parallel_convolution_amd.models.filters import GAUSSIAN_BLUR_3  # noqa: E402
from parallel_convolution_amd.parallel.decomp import halo_for_radius, make_blocks  # noqa: E402
from parallel_convolution_amd.spec import DType, ProblemSpec  # noqa: E402
from parallel_convolution_amd.utils.rawio import synthetic_image  # noqa: E402


def time_single(h, w, iters, chunk=None, label=""):
    spec = ProblemSpec(w, h, 1, DType.U8, GAUSSIAN_BLUR_3, iters)
    blocks = make_blocks(h, w, 1, 1, halo=halo_for_radius(1))
    eng = HipEngine(spec, blocks, chunk_iters=chunk)
    eng.load_block(0, synthetic_image(h, w, 1, DType.U8, seed=3))
    eng.run(20)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    eng.run(iters)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(
        f"[perrank] {label:24s} {h}x{w} chunk={eng.chunk_iters:3d} "
        f"{dt / iters * 1e6:8.1f} us/iter  {h * w * iters / dt / 1e9:6.0f} Gpx/s "
        f"kernel={eng.kernel_path()}",
        flush=True,
    )
    return dt / iters


def time_local_grid(n, p, iters, chunk=None):
    spec = ProblemSpec(n, n, 1, DType.U8, GAUSSIAN_BLUR_3, iters)
    blocks = make_blocks(n, n, p, 1, halo=halo_for_radius(1))
    eng = HipEngine(spec, blocks, chunk_iters=chunk)
    for i, bs in enumerate(blocks):
        eng.load_block(
            i, synthetic_image(bs.h, bs.w, 1, DType.U8, seed=100 + i)
        )
    eng.run(10)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    eng.run(iters)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(
        f"[perrank] local-{p}blk {n}x{n} chunk={eng.chunk_iters:3d} "
        f"{dt / iters * 1e6:8.1f} us/iter  {float(n) * n * iters / dt / 1e9:6.0f} Gpx/s",
        flush=True,
    )
    return dt / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=300)
    ap.add_argument("--all-shapes", action="store_true",
                    help="also time the N=2/N=4 rank shapes and the fp16 "
                    "weak-scaling rank shape")
    args = ap.parse_args()

    # 1. per-rank block of the 2x4 headline decomposition
    time_single(8192, 4096, args.iters, label="per-rank(2x4 headline)")
    # 2. chunk-T sweep at that shape
    for chunk in (8, 16, 32):
        time_single(8192, 4096, args.iters, chunk=chunk, label=f"per-rank chunk={chunk}")
    # whole-image reference point
    time_single(16384, 16384, max(50, args.iters // 4), label="whole 16384^2")
    # 3. LOCAL 2x4 grid on one GPU (D2D exchange, real chunk schedule)
    for chunk in (16, 32):
        time_local_grid(16384, 8, max(50, args.iters // 4), chunk=chunk)
    if args.all_shapes:
        # N=2 (1x2 grid) and N=4 (2x2) headline rank shapes
        time_single(16384, 8192, max(50, args.iters // 2), label="per-rank(1x2, N=2)")
        time_single(8192, 8192, max(50, args.iters // 2), label="per-rank(2x2, N=4)")
        # config-5 weak-scaling rank shape: 65536^2 fp16 on 2x4
        from parallel_convolution_amd.spec import DType as _DT

        spec = ProblemSpec(16384, 32768, 1, _DT.F16, GAUSSIAN_BLUR_3, 1)
        blocks = make_blocks(32768, 16384, 1, 1, halo=halo_for_radius(1))
        eng = HipEngine(spec, blocks)
        eng.load_block(0, synthetic_image(32768, 16384, 1, _DT.F16, seed=5))
        eng.run(10)
        torch.cuda.synchronize()
        import time as _t

        t0 = _t.perf_counter()
        eng.run(40)
        torch.cuda.synchronize()
        dt = _t.perf_counter() - t0
        print(
            f"[perrank] fp16-rank(2x4 of 65536^2) 32768x16384 "
            f"{dt / 40 * 1e6:8.1f} us/iter  {32768 * 16384 * 40 / dt / 1e9:6.0f} Gpx/s "
            f"kernel={eng.kernel_path()}",
            flush=True,
        )


if __name__ == "__main__":
    main()
