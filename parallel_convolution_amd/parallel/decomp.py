"""2D block decomposition and halo-exchange geometry.

MI355X-native counterpart of the reference's r×c MPI Cartesian grid with
ghost-cell halo exchange (BASELINE.json:5,9; SURVEY.md §2.3 — the reference
tree itself was unmounted). One rank per GPU; the grid for
P ranks is the near-square factorization of P (2×4 at P=8, matching the
mandated BASELINE.json:9 "2×4 block decomposition").

Design differences from a plain MPI port, chosen for xGMI latency (SURVEY.md
§7 "hard parts" — per-iteration latency, not bandwidth, is the scaling
constraint on MI355X):

* **Halo deepening**: we exchange a halo of depth d = r·T once per T
  iterations (a "chunk") instead of depth r every iteration, and compute a
  shrinking apron redundantly. The redundant compute is O(d/h) (<1 % at the
  mandated configs) and divides RCCL p2p latency by T.
* **Two-phase corner-free exchange** (SURVEY.md §2.3): phase A exchanges
  East/West column strips (packed on device — strided), phase B exchanges
  North/South row strips at FULL buffer pitch (contiguous, no packing).
  Because phase B rows include the phase-A-received E/W halo columns,
  diagonal-corner data propagates without 8-way p2p.

Coordinates: `y` is the row (height) axis, `x` the column (width) axis, in
PIXEL units. Buffers are padded by `halo` pixels on each side; the owned
block lives at [halo:halo+h, halo:halo+w] (pixel coords; multiply x by
`channels` for storage-element coords).
"""

from __future__ import annotations

import dataclasses
from typing import List, Optional, Tuple

#: Default allocated halo width (pixels). Also the default max exchange
#: depth: with radius-1 filters the engine exchanges every 32 iterations
#: (exchange latency, not bandwidth, bounds stencil scaling on xGMI — the
#: redundant apron compute this buys is <1 % at the mandated configs).
#: A multiple of 16 keeps tile x-origins 16-byte aligned for u8 kernels.
HALO_CAP = 32


def halo_for_radius(radius: int) -> int:
    """Allocated halo width for a filter radius: at least HALO_CAP, and at
    least the radius itself (rounded up to 16 to keep alignment), so large
    odd kernels (k > 33) never read past the padded buffer."""
    need = max(HALO_CAP, radius)
    return (need + 15) // 16 * 16

#: Row pitch alignment in storage ELEMENTS (64 elems → 64/128/256 B for
#: u8/f16/f32 rows: every row start is at least 64-byte aligned).
PITCH_ALIGN = 64


def factor_grid(p: int) -> Tuple[int, int]:
    """Near-square factorization p = rows × cols with rows <= cols.

    Matches the reference's process-grid choice and the
    mandated 2×4 grid at 8 GPUs (BASELINE.json:9).
    """
    if p <= 0:
        raise ValueError("p must be positive")
    best = (1, p)
    r = 1
    while r * r <= p:
        if p % r == 0:
            best = (r, p // r)
        r += 1
    return best


def split_extents(n: int, parts: int) -> List[Tuple[int, int]]:
    """Split n into `parts` contiguous extents [(start, size), ...].

    The first n % parts blocks get one extra element (reference-style
    block distribution).
    """
    base, rem = divmod(n, parts)
    out = []
    start = 0
    for i in range(parts):
        size = base + (1 if i < rem else 0)
        out.append((start, size))
        start += size
    return out


@dataclasses.dataclass(frozen=True)
class BlockSpec:
    """Geometry of one rank's block (pixel units)."""

    rank: int
    pr: int  # process-grid row
    pc: int  # process-grid col
    grid: Tuple[int, int]  # (R, C)
    y0: int  # owned-block origin in image coords
    x0: int
    h: int  # owned-block size
    w: int
    halo: int  # allocated halo width (pixels)
    channels: int

    # neighbor ranks (None at a physical edge)
    north: Optional[int] = None
    south: Optional[int] = None
    west: Optional[int] = None
    east: Optional[int] = None

    @property
    def buf_h(self) -> int:
        """Padded buffer height in rows."""
        return self.h + 2 * self.halo

    @property
    def buf_w_elems(self) -> int:
        """Padded buffer row width in storage elements, pitch-aligned."""
        w = (self.w + 2 * self.halo) * self.channels
        return (w + PITCH_ALIGN - 1) // PITCH_ALIGN * PITCH_ALIGN

    def neighbor(self, side: str) -> Optional[int]:
        return getattr(self, side)


def make_blocks(
    height: int,
    width: int,
    p: int,
    channels: int = 1,
    halo: int = HALO_CAP,
    grid: Optional[Tuple[int, int]] = None,
) -> List[BlockSpec]:
    """Build the full decomposition for p ranks (rank-major over rows)."""
    R, C = grid if grid is not None else factor_grid(p)
    if R * C != p:
        raise ValueError(f"grid {R}x{C} != {p} ranks")
    rows = split_extents(height, R)
    cols = split_extents(width, C)
    blocks = []
    for rank in range(p):
        pr, pc = divmod(rank, C)
        y0, h = rows[pr]
        x0, w = cols[pc]
        if h <= 0 or w <= 0:
            raise ValueError(f"empty block at rank {rank}: {h}x{w}")
        blocks.append(
            BlockSpec(
                rank=rank,
                pr=pr,
                pc=pc,
                grid=(R, C),
                y0=y0,
                x0=x0,
                h=h,
                w=w,
                halo=halo,
                channels=channels,
                north=(rank - C) if pr > 0 else None,
                south=(rank + C) if pr < R - 1 else None,
                west=(rank - 1) if pc > 0 else None,
                east=(rank + 1) if pc < C - 1 else None,
            )
        )
    return blocks


def max_chunk_iters(blocks: List[BlockSpec], radius: int, halo: int) -> int:
    """Largest T so that exchange depth d = radius*T is legal everywhere.

    d must be ≤ the allocated halo and ≤ every block dimension that has a
    neighbor on that axis (the sent strip must lie in the sender's OWNED
    region).
    """
    if radius <= 0:
        return 1_000_000_000
    d_max = halo
    for b in blocks:
        if b.north is not None or b.south is not None:
            d_max = min(d_max, b.h)
        if b.west is not None or b.east is not None:
            d_max = min(d_max, b.w)
    return max(1, d_max // radius)


@dataclasses.dataclass(frozen=True)
class ChunkPlan:
    """Compute-region schedule for one chunk of T iterations at depth d.

    Iteration t (0-based) computes the owned block extended by
    ext(t) = d - (t+1)*r on every side that has a neighbor (0 at physical
    edges); it reads src up to ext(t)+r = d - t*r, which is exactly the
    post-exchange validity. ext(T-1) == d - T*r == 0 when d == r*T.
    """

    iters: int
    depth: int
    radius: int

    def ext(self, t: int) -> int:
        e = self.depth - (t + 1) * self.radius
        assert e >= 0, (self.iters, self.depth, self.radius, t)
        return e

    def region(self, block: BlockSpec, t: int) -> Tuple[int, int, int, int]:
        """Compute region for iteration t as (y_lo, y_hi, x_lo, x_hi) in
        buffer PIXEL coords (halo offset included)."""
        e = self.ext(t)
        y_lo = block.halo - (e if block.north is not None else 0)
        y_hi = block.halo + block.h + (e if block.south is not None else 0)
        x_lo = block.halo - (e if block.west is not None else 0)
        x_hi = block.halo + block.w + (e if block.east is not None else 0)
        return (y_lo, y_hi, x_lo, x_hi)


def plan_chunks(total_iters: int, radius: int, chunk_iters: int) -> List[ChunkPlan]:
    """Split total_iters into chunks of ≤ chunk_iters, each with its own
    exchange depth d = r*T_chunk (the final partial chunk exchanges less)."""
    plans = []
    left = total_iters
    while left > 0:
        t = min(chunk_iters, left)
        plans.append(ChunkPlan(iters=t, depth=radius * t, radius=radius))
        left -= t
    return plans
