"""Direct conv-kernel sweep across image sizes (cache-residency probe).

Note: drives `_C.conv_region` (the VALU / MFMA single-kernel entry), not
the engine's per-problem path selection — the engine may pick the
integer-dot or fused kernels instead (see hip_engine.kernel_path).

Sizes chosen so src+dst land in L2 (~32 MB), L3 (~256 MB), or HBM:
2048²=8 MB, 4096²=33 MB, 8192²=134 MB, 16384²=537 MB (u8, two buffers).
"""

import sys
import time

sys.path.insert(0, ".")
import numpy as np
import torch

from parallel_convolution_amd.hip_engine import load_extension
from parallel_convolution_amd.models.filters import GAUSSIAN_BLUR_3

C = load_extension()


def sweep(n, dtype=torch.uint8, iters=200, k=3, channels=1, use_mfma=False, w=None):
    halo = 16
    w = w or n
    pitch = ((w + 2 * halo) * channels + 63) // 64 * 64
    bufh = n + 2 * halo
    if dtype == torch.uint8:
        src = torch.randint(0, 256, (bufh, pitch), dtype=dtype, device="cuda")
    else:
        src = (torch.rand((bufh, pitch), device="cuda") * 256).to(dtype)
    dst = torch.zeros_like(src)
    wgt = torch.from_numpy(GAUSSIAN_BLUR_3[:k, :k].copy()).cuda().float().flatten()
    region = [halo, halo + n, halo * channels, (halo + w) * channels]
    for _ in range(10):
        C.conv_region(src, dst, wgt, channels, k, halo, region, use_mfma)
        src, dst = dst, src
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        C.conv_region(src, dst, wgt, channels, k, halo, region, use_mfma)
        src, dst = dst, src
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    px = float(n) * w * iters
    import os
    print(
        f"n={n:6d}x{w:6d} dtype={str(dtype)[6:]:8s} mfma={int(use_mfma)} "
        f"chain={os.environ.get('PCONV_CHAIN','auto'):>4s} {px/dt/1e12:.3f} Tpx/s  "
        f"{dt/iters*1e6:8.1f} us/iter  footprint={2*n*w*src.element_size()/1e6:.0f} MB",
        flush=True,
    )


if __name__ == "__main__":
    import sys

    if len(sys.argv) > 1 and sys.argv[1] == "blocks":
        # 8-GPU headline block shape (2x4 grid of 16384²) + 4-GPU (2x2)
        sweep(8192, w=4096)
        sweep(8192, w=8192)
        sweep(4096)
    else:
        for n in (2048, 4096, 8192, 16384):
            sweep(n)
            sweep(n, use_mfma=True)
        sweep(8192, torch.float16)
        sweep(8192, torch.float16, use_mfma=True)
        sweep(8192, torch.float32)
