"""CPU engine: the full decomposition/halo/chunk protocol on host tensors.

This is the device-free twin of the HIP engine ("the 'fake
backend' is the CPU path … making all index math testable in this GPU-less
sandbox"). It implements BASELINE.json:7 config 1 (1024² gray, 3×3, 10
iters, 1 rank on CPU) and serves as the protocol oracle for the C++/RCCL
engine: identical decomposition (parallel/decomp.py), identical two-phase
exchange, identical chunked halo-deepening schedule.

Buffers are kept as fp32 planes (C, buf_h, buf_w); per-iteration storage
quantization (u8 round-half-even / f16 cast — spec.py contract) is applied
in fp32, which is exactly equivalent to storing in the narrow type because
every representable u8/f16 value round-trips through fp32.
"""

from __future__ import annotations

import dataclasses
import time
from typing import List, Optional

import numpy as np
import torch
import torch.nn.functional as F

from .parallel.decomp import (
    BlockSpec,
    halo_for_radius,
    make_blocks,
    max_chunk_iters,
    plan_chunks,
)
from .spec import DType, ProblemSpec


def _quantize_(x: torch.Tensor, dtype: DType) -> torch.Tensor:
    if dtype == DType.U8:
        return torch.clamp(torch.round(x), 0.0, 255.0)
    if dtype == DType.F16:
        return x.to(torch.float16).to(torch.float32)
    return x


class CpuBlock:
    """One rank's padded double buffer + region compute."""

    def __init__(self, spec: ProblemSpec, block: BlockSpec):
        self.spec = spec
        self.block = block
        C = spec.channels
        bh = block.h + 2 * block.halo
        bw = block.w + 2 * block.halo
        self.src = torch.zeros((C, bh, bw), dtype=torch.float32)
        self.dst = torch.zeros((C, bh, bw), dtype=torch.float32)
        self.weight = (
            torch.from_numpy(np.ascontiguousarray(spec.filter))
            .reshape(1, 1, spec.k, spec.k)
            .expand(C, 1, spec.k, spec.k)
            .contiguous()
        )

    def load(self, img_block: np.ndarray) -> None:
        """img_block: (h, w*channels) in storage dtype, interleaved."""
        b, C = self.block, self.spec.channels
        t = torch.from_numpy(np.ascontiguousarray(img_block)).reshape(b.h, b.w, C)
        t = t.permute(2, 0, 1).to(torch.float32)
        self.src[:, b.halo : b.halo + b.h, b.halo : b.halo + b.w] = t

    def store(self) -> np.ndarray:
        b, C = self.block, self.spec.channels
        t = self.src[:, b.halo : b.halo + b.h, b.halo : b.halo + b.w]
        t = t.permute(1, 2, 0).reshape(b.h, b.w * C).contiguous()
        return t.numpy().astype(self.spec.dtype.numpy_dtype)

    def conv_region(self, region) -> None:
        """dst[region] = quantize(conv(src)) over buffer-pixel region."""
        r = self.spec.radius
        y0, y1, x0, x1 = region
        xin = self.src[:, y0 - r : y1 + r, x0 - r : x1 + r].unsqueeze(0)
        y = F.conv2d(xin, self.weight, groups=self.spec.channels).squeeze(0)
        self.dst[:, y0:y1, x0:x1] = _quantize_(y, self.spec.dtype)

    def swap(self) -> None:
        self.src, self.dst = self.dst, self.src


# --- exchangers -----------------------------------------------------------

# message tags (gloo): phase letter + direction of travel
_TAG_A_EAST = 11  # payload travelling eastward (sender's east strip -> east rank)
_TAG_A_WEST = 12
_TAG_B_SOUTH = 21
_TAG_B_NORTH = 22


class LocalExchanger:
    """All blocks live in one process: exchange = tensor copies.

    This is also the model for the HIP engine's single-process multi-block
    test mode (D2D copies standing in for ncclSend/Recv)."""

    def __init__(self, blocks: List[CpuBlock]):
        self.blocks = {b.block.rank: b for b in blocks}

    def exchange(self, depth: int) -> None:
        d = depth
        # Phase A: East/West column strips, owned rows only.
        for blk in self.blocks.values():
            b = blk.block
            H = b.halo
            if b.east is not None:
                e = self.blocks[b.east]
                # my east strip -> east rank's west halo
                e.src[:, H : H + b.h, H - d : H] = blk.src[
                    :, H : H + b.h, H + b.w - d : H + b.w
                ]
                # east rank's west strip -> my east halo
                blk.src[:, H : H + b.h, H + b.w : H + b.w + d] = e.src[
                    :, H : H + e.block.h, H : H + d
                ]
        # Phase B: North/South full-width row strips (corners ride along).
        for blk in self.blocks.values():
            b = blk.block
            H = b.halo
            if b.south is not None:
                s = self.blocks[b.south]
                s.src[:, H - d : H, :] = blk.src[:, H + b.h - d : H + b.h, :]
                blk.src[:, H + b.h : H + b.h + d, :] = s.src[:, H : H + d, :]


class GlooExchanger:
    """torch.distributed point-to-point exchange (gloo backend on CPU).

    Mirrors the RCCL grouped send/recv the C++ engine performs
    (SURVEY.md §2.3 mapping table): phase A strided column strips via
    contiguous staging, phase B contiguous full-width row strips."""

    def __init__(self, blk: CpuBlock):
        import torch.distributed as dist

        self.dist = dist
        self.blk = blk

    def _p2p(self, ops) -> None:
        works = []
        for kind, tensor, peer, tag in ops:
            if kind == "send":
                works.append(self.dist.isend(tensor, peer, tag=tag))
            else:
                works.append(self.dist.irecv(tensor, peer, tag=tag))
        for w in works:
            w.wait()

    def exchange(self, depth: int) -> None:
        d = depth
        blk = self.blk
        b = blk.block
        H = b.halo
        src = blk.src
        # Phase A: E/W columns (staged contiguous)
        ops = []
        recv_slots = []
        if b.east is not None:
            send_e = src[:, H : H + b.h, H + b.w - d : H + b.w].contiguous()
            recv_e = torch.empty_like(send_e)
            ops.append(("send", send_e, b.east, _TAG_A_EAST))
            ops.append(("recv", recv_e, b.east, _TAG_A_WEST))
            recv_slots.append((recv_e, (slice(None), slice(H, H + b.h), slice(H + b.w, H + b.w + d))))
        if b.west is not None:
            send_w = src[:, H : H + b.h, H : H + d].contiguous()
            recv_w = torch.empty_like(send_w)
            ops.append(("send", send_w, b.west, _TAG_A_WEST))
            ops.append(("recv", recv_w, b.west, _TAG_A_EAST))
            recv_slots.append((recv_w, (slice(None), slice(H, H + b.h), slice(H - d, H))))
        if ops:
            self._p2p(ops)
            for t, sl in recv_slots:
                src[sl] = t
        # Phase B: N/S full-width rows (contiguous in the (C,bh,bw) layout
        # per channel-plane; stage anyway for C>1 simplicity)
        ops = []
        recv_slots = []
        if b.south is not None:
            send_s = src[:, H + b.h - d : H + b.h, :].contiguous()
            recv_s = torch.empty_like(send_s)
            ops.append(("send", send_s, b.south, _TAG_B_SOUTH))
            ops.append(("recv", recv_s, b.south, _TAG_B_NORTH))
            recv_slots.append((recv_s, (slice(None), slice(H + b.h, H + b.h + d), slice(None))))
        if b.north is not None:
            send_n = src[:, H : H + d, :].contiguous()
            recv_n = torch.empty_like(send_n)
            ops.append(("send", send_n, b.north, _TAG_B_NORTH))
            ops.append(("recv", recv_n, b.north, _TAG_B_SOUTH))
            recv_slots.append((recv_n, (slice(None), slice(H - d, H), slice(None))))
        if ops:
            self._p2p(ops)
            for t, sl in recv_slots:
                src[sl] = t


# --- engine ---------------------------------------------------------------


@dataclasses.dataclass
class RunStats:
    iters: int
    elapsed_s: float
    exchanges: int


class CpuEngine:
    """Runs the chunked iterate-exchange loop on 1..P local blocks, or on
    one block of a gloo-distributed world."""

    def __init__(
        self,
        spec: ProblemSpec,
        blocks: List[CpuBlock],
        exchanger=None,
        chunk_iters: Optional[int] = None,
    ):
        self.spec = spec
        self.blocks = blocks
        self.exchanger = exchanger
        all_specs = [b.block for b in blocks]
        t_cap = max_chunk_iters(all_specs, spec.radius, all_specs[0].halo)
        self.chunk_iters = min(chunk_iters or t_cap, t_cap)

    def diff_max(self) -> float:
        """max |src - dst| over owned regions."""
        m = 0.0
        for blk in self.blocks:
            b = blk.block
            s = blk.src[:, b.halo : b.halo + b.h, b.halo : b.halo + b.w]
            d = blk.dst[:, b.halo : b.halo + b.h, b.halo : b.halo + b.w]
            m = max(m, float((s - d).abs().max()))
        return m

    def run(self, iters: Optional[int] = None) -> RunStats:
        iters = self.spec.iters if iters is None else iters
        r = self.spec.radius
        plans = plan_chunks(iters, r, self.chunk_iters)
        t0 = time.perf_counter()
        nex = 0
        for plan in plans:
            if self.exchanger is not None and r > 0:
                self.exchanger.exchange(plan.depth)
                nex += 1
            for t in range(plan.iters):
                for blk in self.blocks:
                    blk.conv_region(plan.region(blk.block, t))
                for blk in self.blocks:
                    blk.swap()
        return RunStats(iters=iters, elapsed_s=time.perf_counter() - t0, exchanges=nex)


def run_single_cpu(spec: ProblemSpec, image: np.ndarray) -> np.ndarray:
    """Convenience: whole image, 1 rank, CPU (BASELINE.json:7 config 1)."""
    blocks = make_blocks(
        spec.height, spec.width, 1, spec.channels, halo=halo_for_radius(spec.radius)
    )
    blk = CpuBlock(spec, blocks[0])
    blk.load(image)
    eng = CpuEngine(spec, [blk], exchanger=None)
    eng.run()
    return blk.store()


def run_local_blocks(spec: ProblemSpec, image: np.ndarray, p: int) -> np.ndarray:
    """P blocks in one process with LocalExchanger — protocol test path."""
    specs = make_blocks(
        spec.height, spec.width, p, spec.channels, halo=halo_for_radius(spec.radius)
    )
    blocks = []
    for bs in specs:
        blk = CpuBlock(spec, bs)
        blk.load(
            image[bs.y0 : bs.y0 + bs.h, bs.x0 * spec.channels : (bs.x0 + bs.w) * spec.channels]
        )
        blocks.append(blk)
    eng = CpuEngine(spec, blocks, exchanger=LocalExchanger(blocks))
    eng.run()
    out = np.zeros_like(image)
    for blk in blocks:
        bs = blk.block
        out[bs.y0 : bs.y0 + bs.h, bs.x0 * spec.channels : (bs.x0 + bs.w) * spec.channels] = (
            blk.store()
        )
    return out
