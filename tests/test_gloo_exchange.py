"""Multi-process halo exchange over torch.distributed gloo (CPU).

Covers the distributed exchange-ordering logic that the RCCL path reuses
(SURVEY.md §4 'multi-rank without a cluster'): world_size 2 and 4 in this
GPU-less sandbox, rendezvous on 127.0.0.1.
"""

import multiprocessing as mp
import os

import numpy as np
import pytest

from parallel_convolution_amd.models.filters import GAUSSIAN_BLUR_3, SHARPEN_5
from parallel_convolution_amd.ops.reference import run_reference
from parallel_convolution_amd.spec import DType, ProblemSpec
from parallel_convolution_amd.utils.rawio import synthetic_image


def _worker(rank, world, port, spec_args, img_bytes, shape, out_q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    import torch.distributed as dist

    from parallel_convolution_amd.cpu_engine import CpuBlock, CpuEngine, GlooExchanger
    from parallel_convolution_amd.parallel.decomp import make_blocks

    dist.init_process_group("gloo")
    try:
        w, h, c, dt, filt_list, k, iters = spec_args
        filt = np.array(filt_list, dtype=np.float32).reshape(k, k)
        spec = ProblemSpec(w, h, c, DType(dt), filt, iters)
        img = np.frombuffer(img_bytes, dtype=spec.dtype.numpy_dtype).reshape(shape)
        blocks = make_blocks(h, w, world, c)
        bs = blocks[rank]
        blk = CpuBlock(spec, bs)
        blk.load(img[bs.y0 : bs.y0 + bs.h, bs.x0 * c : (bs.x0 + bs.w) * c])
        eng = CpuEngine(spec, [blk], exchanger=GlooExchanger(blk))
        eng.run()
        out_q.put((rank, bs.y0, bs.x0, blk.store().tobytes()))
    finally:
        dist.destroy_process_group()


def _run_world(world, spec, img, port):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    spec_args = (
        spec.width,
        spec.height,
        spec.channels,
        spec.dtype.value,
        [float(v) for v in spec.filter.flat],
        spec.k,
        spec.iters,
    )
    procs = [
        ctx.Process(
            target=_worker,
            args=(r, world, port, spec_args, img.tobytes(), img.shape, q),
        )
        for r in range(world)
    ]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    out = np.zeros_like(img)
    from parallel_convolution_amd.parallel.decomp import make_blocks

    blocks = {b.rank: b for b in make_blocks(spec.height, spec.width, world, spec.channels)}
    for rank, y0, x0, data in results:
        b = blocks[rank]
        blk = np.frombuffer(data, dtype=spec.dtype.numpy_dtype).reshape(
            b.h, b.w * spec.channels
        )
        out[y0 : y0 + b.h, x0 * spec.channels : (x0 + b.w) * spec.channels] = blk
    return out


@pytest.mark.parametrize("world,port", [(2, 29611), (4, 29612)])
def test_gloo_world_matches_oracle(world, port):
    spec = ProblemSpec(96, 72, 1, DType.U8, GAUSSIAN_BLUR_3, 18)
    img = synthetic_image(72, 96, 1, DType.U8)
    out = _run_world(world, spec, img, port)
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(out, ref)


def test_gloo_world2_rgb_k5(port=29613):
    spec = ProblemSpec(48, 40, 3, DType.U8, SHARPEN_5, 5)
    img = synthetic_image(40, 48, 3, DType.U8)
    out = _run_world(2, spec, img, port)
    ref = run_reference(spec, img)
    np.testing.assert_array_equal(out, ref)
