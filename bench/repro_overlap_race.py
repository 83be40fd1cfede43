"""Deterministic regression for the overlap-pyramid send-capture race.

The r02 fuzz campaign caught intermittent corruption at overlap depth
>= 2 on 2D process grids (write-after-read: interior 1 overwrote the
chunk-start buffer's edge strips while the exchange was still reading
them for sends). These fixed cases reproduced it reliably before the
staged-send fix (csrc/engine_api.hpp exchange capture phase +
evPacked_); they must stay at BAD: 0.

    python bench/repro_overlap_race.py            # default depth (2)
    PCONV_OVERLAP_DEPTH=4 python bench/repro_overlap_race.py
"""

import sys

sys.path.insert(0, ".")
import numpy as np
from parallel_convolution_amd.hip_engine import run_local_blocks_gpu
from parallel_convolution_amd.models.filters import GAUSSIAN_BLUR_3, GAUSSIAN_BLUR_5
from parallel_convolution_amd.ops.reference import run_reference
from parallel_convolution_amd.spec import DType, ProblemSpec
from parallel_convolution_amd.utils.rawio import synthetic_image

CASES = [
    (55, 662, 1, DType.U8, GAUSSIAN_BLUR_3, 21, 8),   # h,w order: fuzz prints WxH
    (662, 55, 1, DType.U8, GAUSSIAN_BLUR_3, 21, 8),
    (129, 393, 1, DType.U8, GAUSSIAN_BLUR_3, 35, 4),
    (393, 129, 1, DType.U8, GAUSSIAN_BLUR_3, 35, 4),
    (626, 83, 1, DType.F32, GAUSSIAN_BLUR_5, 38, 8),
    (83, 626, 1, DType.F32, GAUSSIAN_BLUR_5, 38, 8),
]
bad = 0
for (h, w, c, dt, filt, iters, p) in CASES:
    spec = ProblemSpec(w, h, c, dt, filt, iters)
    img = synthetic_image(h, w, c, dt, seed=123)
    try:
        out = run_local_blocks_gpu(spec, img, p)
    except ValueError:
        print("skip", h, w); continue
    ref = run_reference(spec, img)
    if dt == DType.U8:
        ok = np.array_equal(out, ref)
    else:
        ok = np.allclose(out, ref, rtol=1e-4, atol=1e-2)
    print(("OK  " if ok else "FAIL"), f"{h}x{w} {dt.value} k={spec.k} it={iters} p={p}")
    bad += 0 if ok else 1
print("BAD:", bad)
