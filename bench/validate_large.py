"""Config-5-scale validation: 65536² fp16 (17.2 GB working set per copy).

Runs the whole image as ONE block and as TWO decomposed blocks with halo
exchange on a single GPU, and asserts the outputs are bit-identical —
the halo property at the 288-GB-HBM sizing point (BASELINE.json:11),
also exercising >2^31-element buffers (size_t indexing paths).
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

N = 65536
ITERS = 10


def main():
    global N
    if len(sys.argv) > 1:
        N = int(sys.argv[1])
    dt = DType.U8 if len(sys.argv) > 2 and sys.argv[2] == "u8" else DType.F16
    spec = ProblemSpec(N, N, 1, dt, GAUSSIAN_BLUR_3, ITERS)
    print("generating synthetic block data...", flush=True)

    t0 = time.time()
    # single block run; keep the result on-device
    blocks1 = make_blocks(N, N, 1, 1)
    eng1 = HipEngine(spec, blocks1)
    img = synthetic_image(N, N, 1, dt, seed=42)
    eng1.load_block(0, img)
    eng1.run()
    b = blocks1[0]
    hold = (
        (eng1.blocks[0].dst if eng1._out_is_dst else eng1.blocks[0].src)[
            b.halo : b.halo + b.h, b.halo : b.halo + b.w
        ]
        .clone()
    )
    del eng1
    torch.cuda.empty_cache()
    print(f"single-block run done in {time.time()-t0:.1f}s", flush=True)

    blocks2 = make_blocks(N, N, 2, 1)
    eng2 = HipEngine(spec, blocks2)
    for i, bs in enumerate(blocks2):
        eng2.load_block(i, img[bs.y0 : bs.y0 + bs.h, bs.x0 : bs.x0 + bs.w])
    eng2.run()
    ok = True
    for i, bs in enumerate(blocks2):
        t = eng2.blocks[i].dst if eng2._out_is_dst else eng2.blocks[i].src
        out = t[bs.halo : bs.halo + bs.h, bs.halo : bs.halo + bs.w]
        ref = hold[bs.y0 : bs.y0 + bs.h, bs.x0 : bs.x0 + bs.w]
        same = bool(torch.equal(out, ref))
        print(f"block {i}: bit-identical = {same}", flush=True)
        if not same:
            mism = (out != ref).any(dim=1)
            rows = torch.nonzero(mism).flatten()
            cols = torch.nonzero((out != ref).any(dim=0)).flatten()
            print(
                f"  rows [{int(rows[0])}, {int(rows[-1])}] n={int(mism.sum())}, "
                f"cols [{int(cols[0])}, {int(cols[-1])}]",
                flush=True,
            )
            r0 = int(rows[0])
            d = (out[r0] != ref[r0]).nonzero().flatten()
            print(f"  first row {r0} (+y0={bs.y0}): {int(d.numel())} diffs, first col {int(d[0])}", flush=True)
        ok &= same
    print("VALIDATE_LARGE:", "PASS" if ok else "FAIL", flush=True)
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
