"""Headline-geometry validation on one GPU.

Runs the exact BASELINE.json:9 decomposition — 16384² gray u8, 3×3,
100 iterations, 2×4 grid — in single-process local mode (D2D copies in
place of ncclSend/Recv; identical chunk schedule, regions, pack
geometry, overlap path) and checks it is bit-identical to the 1-block
run. This is the 1-GPU-box stand-in for the driver's 8-GPU RCCL run.
"""

import sys
import time

sys.path.insert(0, ".")
import torch

from parallel_convolution_amd.hip_engine import HipEngine
from parallel_convolution_amd.models.filters import GAUSSIAN_BLUR_3
from parallel_convolution_amd.parallel.decomp import make_blocks
from parallel_convolution_amd.spec import DType, ProblemSpec
from parallel_convolution_amd.utils.rawio import synthetic_image

N, ITERS, P = 16384, 100, 8


def main():
    spec = ProblemSpec(N, N, 1, DType.U8, GAUSSIAN_BLUR_3, ITERS)
    img = synthetic_image(N, N, 1, DType.U8, seed=7)

    t0 = time.time()
    b1 = make_blocks(N, N, 1, 1)
    e1 = HipEngine(spec, b1)
    e1.load_block(0, img)
    e1.run()
    bb = b1[0]
    hold = (
        (e1.blocks[0].dst if e1._out_is_dst else e1.blocks[0].src)[
            bb.halo : bb.halo + N, bb.halo : bb.halo + N
        ]
        .clone()
    )
    del e1
    torch.cuda.empty_cache()
    print(f"1-block x{ITERS} done in {time.time()-t0:.1f}s", flush=True)

    bp = make_blocks(N, N, P, 1)
    ep = HipEngine(spec, bp)
    print("grid:", bp[0].grid, "chunk T:", ep.chunk_iters, flush=True)
    for i, bs in enumerate(bp):
        ep.load_block(i, img[bs.y0 : bs.y0 + bs.h, bs.x0 : bs.x0 + bs.w])
    t0 = time.time()
    ep.run()
    print(f"{P}-block x{ITERS} done in {time.time()-t0:.1f}s", flush=True)
    ok = True
    for i, bs in enumerate(bp):
        t = ep.blocks[i].dst if ep._out_is_dst else ep.blocks[i].src
        out = t[bs.halo : bs.halo + bs.h, bs.halo : bs.halo + bs.w]
        ref = hold[bs.y0 : bs.y0 + bs.h, bs.x0 : bs.x0 + bs.w]
        same = bool(torch.equal(out, ref))
        print(f"rank {i} ({bs.h}x{bs.w} at {bs.y0},{bs.x0}): bit-identical = {same}", flush=True)
        ok &= same
    print("VALIDATE_HEADLINE_GEOMETRY:", "PASS" if ok else "FAIL", flush=True)
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
