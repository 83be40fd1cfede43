"""Isolate halo-exchange overhead (pack + copy + schedule) on one GPU.

LOCAL-mode runs with 1x2 (E/W exchange: strided pack via
hipMemcpy2DAsync) and 2x1 (N/S exchange: contiguous, no pack) grids at
the N=2 headline decomposition, swept over chunk T. The per-iteration
cost difference vs the no-neighbor single block isolates what one
exchange round costs — the quantity the 8-GPU scaling point must hide
("the hipMemcpy2DAsync column staging ... never been timed").

Usage: python bench/exchange_cost.py [--iters N]
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

N = 16384


def run(grid, chunk, iters):
    p = grid[0] * grid[1]
    spec = ProblemSpec(N, N, 1, DType.U8, GAUSSIAN_BLUR_3, iters)
    blocks = make_blocks(N, N, p, 1, halo=halo_for_radius(1), grid=grid if p > 1 else None)
    eng = HipEngine(spec, blocks, chunk_iters=chunk)
    for i, bs in enumerate(blocks):
        eng.load_block(i, synthetic_image(bs.h, bs.w, 1, DType.U8, seed=i))
    eng.run(16)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    eng.run(iters)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters * 1e6
    return dt


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=200)
    args = ap.parse_args()

    base = run((1, 1), 32, args.iters)
    print(f"[xcost] 1-block (no exchange)          {base:8.1f} us/iter", flush=True)
    for grid, tag in (((1, 2), "E/W strided pack"), ((2, 1), "N/S contiguous ")):
        for chunk in (1, 4, 8, 32):
            dt = run(grid, chunk, args.iters)
            # both blocks run on ONE GPU: compute cost == base; the rest
            # is exchange+schedule, amortized over the chunk
            print(
                f"[xcost] {grid[0]}x{grid[1]} {tag} chunk={chunk:2d} "
                f"{dt:8.1f} us/iter  (+{dt - base:6.1f} over compute; "
                f"~{(dt - base) * chunk:6.1f} us per exchange round)",
                flush=True,
            )


if __name__ == "__main__":
    main()
