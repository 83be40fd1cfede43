"""Python wrapper of the C++/HIP/RCCL engine (csrc/).

The thin PyTorch-ROCm front end of BASELINE.json:5: torch owns the device
buffers (its caching allocator), the C++ ConvEngine owns streams, the
iteration loop and the RCCL communicator. This module never touches pixels
on the hot path — run() is one C++ call for the whole timed loop.

The extension is REQUIRED on a GPU box: if `parallel_convolution_amd._C`
is missing there we raise instead of silently falling back to eager
PyTorch (the native path must be the one that runs).
"""

from __future__ import annotations

import dataclasses
from typing import List, Optional

import numpy as np
import torch

from .parallel.decomp import BlockSpec, halo_for_radius, make_blocks, max_chunk_iters
from .spec import DType, ProblemSpec

_DT_CODE = {DType.U8: 0, DType.F16: 1, DType.F32: 2}


def mfma_eligible(spec: ProblemSpec) -> bool:
    """MFMA (fp16-input, fp32-accumulate) path selection.

    Numerically valid only when every filter coefficient round-trips
    through fp16 (u8/f16 pixels always do; see csrc/conv_kernels.hip MFMA
    notes). Measured A/B (profiles/r01_single_gpu.md): parity with the
    VALU kernel at k=3 (9 taps — staging-bound either way), +72-83 % at
    k=5 (25 taps — the VALU path is fma-issue-bound there), so the
    default picks MFMA for k=5 only. PCONV_DISABLE_MFMA=1 /
    PCONV_FORCE_MFMA=1 override for A/B.
    """
    import os

    if os.environ.get("PCONV_DISABLE_MFMA"):
        return False
    if spec.dtype not in (DType.U8, DType.F16):
        return False
    if spec.k not in (3, 5) or spec.channels not in (1, 3):
        return False
    import torch as _t

    f = _t.from_numpy(spec.filter.copy())
    if not bool((f.half().float() == f).all()):
        return False
    if os.environ.get("PCONV_FORCE_MFMA"):
        return True
    return spec.k == 5


def u8sep_weights(spec: ProblemSpec):
    """Integer taps [shift, a0..a4, b0..b4] for the separable integer path
    (csrc conv_u8sep_k): k in {3,5} grayscale u8 with a non-negative
    SEPARABLE dyadic filter w = outer(b, a)/2^shift (integer a, b >= 0,
    a_i <= 255) whose worst-case accumulator fits u16:
    255*sum(a)*sum(b) + 2^(shift-1) <= 65535. Both mandated Gaussian
    blurs qualify; the horizontal pass is then shared across the 2R+1
    output rows that read it (instructions/pixel).
    Empty list = ineligible. PCONV_DISABLE_SEP=1 forces the next path."""
    import math
    import os

    if os.environ.get("PCONV_DISABLE_SEP"):
        return []
    if spec.dtype != DType.U8 or spec.k not in (3, 5) or spec.channels != 1:
        return []
    w = spec.filter.astype("float64").reshape(spec.k, spec.k)
    if (w < 0).any():
        return []
    for shift in range(1, 9):
        n = w * (1 << shift)
        if not (n == n.round()).all():
            continue
        n = n.round().astype("int64")
        # integer rank-1 factorization n = outer(b, a)
        sums = n.sum(axis=1)
        i0 = int(sums.argmax())
        if sums[i0] == 0:
            return []
        row = n[i0]
        g = int(math.gcd(*[int(v) for v in row]))
        if g == 0:
            return []
        a = row // g
        j0 = int(a.argmax())
        if a[j0] == 0:
            return []
        b = n[:, j0] // a[j0]
        if (np.outer(b, a) != n).any():
            return []
        if a.max() > 255:
            return []
        if 255 * int(a.sum()) * int(b.sum()) + (1 << (shift - 1)) > 65535:
            return []
        pad = [0] * (5 - spec.k)
        return (
            [shift]
            + [int(v) for v in a]
            + pad
            + [int(v) for v in b]
            + pad
        )
    return []


def u8dot_weights(spec: ProblemSpec):
    """Integer taps [shift, n0..n8] for the exact u8 dot-product path
    (csrc conv_u8dot_k): k=3 grayscale with a non-negative dyadic filter
    w = n/2^shift, n in [0,255], sum(n) <= 2^shift — the mandated
    Gaussian blur qualifies. Empty list = ineligible.
    PCONV_DISABLE_DOT=1 forces the fp32 VALU path for A/B."""
    import os

    if os.environ.get("PCONV_DISABLE_DOT"):
        return []
    if spec.dtype != DType.U8 or spec.k != 3 or spec.channels != 1:
        return []
    w = spec.filter.astype("float64").reshape(-1)
    if (w < 0).any():
        return []
    for shift in range(1, 16):
        n = w * (1 << shift)
        if (n == n.round()).all():
            n = n.round().astype("int64")
            if n.max() <= 255 and n.sum() <= (1 << shift):
                return [shift] + [int(v) for v in n]
            return []
    return []


def load_extension():
    try:
        from . import _C  # noqa: F401

        return _C
    except ImportError as e:
        raise ImportError(
            "parallel_convolution_amd._C is not built. Run "
            "`PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace` "
            f"(import error: {e})"
        ) from e


@dataclasses.dataclass
class _HipBlock:
    block: BlockSpec
    src: torch.Tensor  # (bufh, pitchE) storage dtype
    dst: torch.Tensor


class HipEngine:
    """One process's view of the run: 1 block (distributed/RCCL) or the
    whole decomposition (single-process local mode, D2D exchange)."""

    def __init__(
        self,
        spec: ProblemSpec,
        blocks: List[BlockSpec],
        world_blocks: Optional[List[BlockSpec]] = None,
        device: str = "cuda",
        chunk_iters: Optional[int] = None,
    ):
        self._C = load_extension()
        self.spec = spec
        self.device = torch.device(device)
        self.blocks: List[_HipBlock] = []
        filt = [float(v) for v in spec.filter.flat]
        self.use_mfma = mfma_eligible(spec)
        self.dot_weights = u8dot_weights(spec)
        self.sep_weights = u8sep_weights(spec)
        self.engine = self._C.ConvEngine(
            _DT_CODE[spec.dtype],
            spec.k,
            filt,
            spec.channels,
            self.use_mfma,
            self.dot_weights,
            self.sep_weights,
        )
        for bs in blocks:
            shape = (bs.buf_h, bs.buf_w_elems)
            src = torch.zeros(shape, dtype=spec.dtype.torch_dtype, device=self.device)
            dst = torch.zeros(shape, dtype=spec.dtype.torch_dtype, device=self.device)
            nbr = [
                -1 if bs.north is None else bs.north,
                -1 if bs.south is None else bs.south,
                -1 if bs.west is None else bs.west,
                -1 if bs.east is None else bs.east,
            ]
            idx = self.engine.add_block(
                src, dst, bs.h, bs.w, bs.halo, spec.channels, nbr
            )
            assert idx == len(self.blocks)
            self.blocks.append(_HipBlock(bs, src, dst))
        # chunk T must be identical on every rank: derive from the WORLD
        # decomposition, not just local blocks.
        wb = world_blocks if world_blocks is not None else blocks
        t_cap = max_chunk_iters(wb, spec.radius, wb[0].halo)
        self.chunk_iters = min(chunk_iters or t_cap, t_cap)
        self._iters_done = 0

    # -- local-mode neighbor remap: BlockSpec neighbors are ranks; in local
    # mode ranks == local block indices already (make_blocks order), so
    # nothing to remap.

    def init_distributed(self, rank: int, world: int) -> None:
        """Create the RCCL communicator; ncclUniqueId broadcast over the
        (gloo) control group."""
        import torch.distributed as dist

        from .parallel.bootstrap import broadcast_bytes

        uid = self._C.get_nccl_unique_id() if rank == 0 else None
        uid = broadcast_bytes(uid, src=0, length=128)
        assert dist.get_rank() == rank and dist.get_world_size() == world
        self.engine.init_nccl(uid, rank, world)

    @property
    def _out_is_dst(self) -> bool:
        return bool(self.engine.swapped())

    def load_block(self, i: int, img_block: np.ndarray) -> None:
        hb = self.blocks[i]
        b, S = hb.block, self.spec.channels
        t = torch.from_numpy(np.ascontiguousarray(img_block))
        assert t.shape == (b.h, b.w * S), (t.shape, b)
        target = hb.dst if self._out_is_dst else hb.src
        target[b.halo : b.halo + b.h, b.halo * S : (b.halo + b.w) * S].copy_(
            t.to(self.device)
        )

    def store_block(self, i: int) -> np.ndarray:
        hb = self.blocks[i]
        b, S = hb.block, self.spec.channels
        t = hb.dst if self._out_is_dst else hb.src
        out = t[b.halo : b.halo + b.h, b.halo * S : (b.halo + b.w) * S]
        return out.contiguous().cpu().numpy()

    def run(self, iters: Optional[int] = None) -> None:
        """Enqueue + complete `iters` iterations (C++ loop; returns after
        device sync)."""
        iters = self.spec.iters if iters is None else iters
        if iters <= 0:
            return
        # The engine runs on its own HIP streams: torch-side work (the
        # buffer zero-fill at construction, load_block's H2D copies) must
        # complete first or it races the conv kernels — at 65536² the
        # 8.6 GB fill demonstrably overwrote conv output (see
        # bench/validate_large.py, which caught this).
        torch.cuda.synchronize()
        self.engine.run(iters, self.chunk_iters)
        self._iters_done += iters

    def diff_max(self) -> float:
        """max |last - previous| over owned regions of local blocks (the
        convergence test of SURVEY.md §2.1 R10; allreduce is the caller's
        job in distributed mode). When the last step ran a fused
        double-iteration kernel, "previous" is two iterations back — a
        conservative (larger) delta, so convergence may be detected one
        check later but never falsely early."""
        torch.cuda.synchronize()
        return float(self.engine.diff_max())

    def set_phase_timing(self, on: bool) -> None:
        """hipEvent pairs around each chunk's exchange/compute (SURVEY.md
        §5.1); adds one device sync per chunk — keep off in timed bench."""
        self.engine.set_phase_timing(on)

    def kernel_path(self) -> str:
        """Which compute path run() takes (u8dot / mfma-f16 / valu-fused2
        / valu) — recorded in bench output."""
        return str(self.engine.kernel_path())

    def phase_times(self) -> dict:
        ex, comp, chunks = self.engine.phase_times()
        return {"exchange_ms": ex, "compute_ms": comp, "chunks": int(chunks)}

    def synchronize(self) -> None:
        self.engine.synchronize()


def run_single_gpu(spec: ProblemSpec, image: np.ndarray, device: str = "cuda") -> np.ndarray:
    """Whole image on one GPU (BASELINE.json:8 config 2 path)."""
    blocks = make_blocks(
        spec.height, spec.width, 1, spec.channels, halo=halo_for_radius(spec.radius)
    )
    eng = HipEngine(spec, blocks)
    eng.load_block(0, image)
    eng.run()
    return eng.store_block(0)


def run_local_blocks_gpu(spec: ProblemSpec, image: np.ndarray, p: int) -> np.ndarray:
    """P blocks on ONE GPU with D2D exchange — the 1-GPU-box test mode for
    the multi-GPU protocol."""
    specs = make_blocks(
        spec.height, spec.width, p, spec.channels, halo=halo_for_radius(spec.radius)
    )
    eng = HipEngine(spec, specs)
    S = spec.channels
    for i, bs in enumerate(specs):
        eng.load_block(i, image[bs.y0 : bs.y0 + bs.h, bs.x0 * S : (bs.x0 + bs.w) * S])
    eng.run()
    out = np.zeros_like(image)
    for i, bs in enumerate(specs):
        out[bs.y0 : bs.y0 + bs.h, bs.x0 * S : (bs.x0 + bs.w) * S] = eng.store_block(i)
    return out
