"""Raw (headerless) image I/O: whole-frame and per-rank block access.

Mirrors the reference's raw-image contract (BASELINE.json:5: "raw
input/output image format"; SURVEY.md §2.1 R6: each rank reads/writes its
own block at computed offsets). Host-side I/O sits OUTSIDE the timed
iteration loop, exactly as the reference times only the MPI loop
(SURVEY.md §5.8c).

A GPU-box path with the same block-offset semantics also exists natively in
C++ (csrc/io.cpp, `_C.read_block`/`_C.write_block`) for the engine's
scatter/gather; this module is the portable implementation and the oracle
for it.
"""

from __future__ import annotations

import os

import numpy as np

from ..spec import DType


def read_image(path: str, height: int, width: int, channels: int, dtype: DType) -> np.ndarray:
    """Read a whole raw frame as (height, width*channels) of dtype."""
    n = height * width * channels
    arr = np.fromfile(path, dtype=dtype.numpy_dtype, count=n)
    if arr.size != n:
        raise ValueError(
            f"{path}: expected {n} elements ({height}x{width}x{channels}), got {arr.size}"
        )
    return arr.reshape(height, width * channels)


def write_image(path: str, img: np.ndarray) -> None:
    np.ascontiguousarray(img).tofile(path)


def _native_io():
    """The C++ block-I/O (csrc/io.cpp) when the extension is built; the
    Python fallback keeps this module importable without it."""
    try:
        import torch  # noqa: F401 — extension links against torch

        from .. import _C

        return _C
    except ImportError:
        return None


def read_block(
    path: str,
    height: int,
    width: int,
    channels: int,
    dtype: DType,
    y0: int,
    x0: int,
    h: int,
    w: int,
) -> np.ndarray:
    """Read one rank's (h, w) pixel block: returns (h, w*channels).

    Per-rank row-offset reads like the reference's fread+fseek scatter
    (SURVEY.md §2.3 'input scatter' row); served by the native pread
    implementation (csrc/io.cpp) when the extension is available.
    """
    row_elems = width * channels
    itemsize = dtype.itemsize
    C = _native_io()
    if C is not None:
        import torch

        out_t = torch.empty((h, w * channels), dtype=dtype.torch_dtype)
        C.read_block(path, row_elems * itemsize, x0 * channels * itemsize, y0, out_t)
        return out_t.numpy()
    out = np.empty((h, w * channels), dtype=dtype.numpy_dtype)
    with open(path, "rb") as f:
        for r in range(h):
            f.seek(((y0 + r) * row_elems + x0 * channels) * itemsize)
            buf = f.read(w * channels * itemsize)
            if len(buf) != w * channels * itemsize:
                raise ValueError(f"{path}: short read at row {y0+r}")
            out[r] = np.frombuffer(buf, dtype=dtype.numpy_dtype)
    return out


def write_block(
    path: str,
    height: int,
    width: int,
    channels: int,
    dtype: DType,
    y0: int,
    x0: int,
    block: np.ndarray,
) -> None:
    """Write one rank's block into a (pre-sized) raw file at its offsets."""
    h = block.shape[0]
    w_elems = block.shape[1]
    row_elems = width * channels
    itemsize = dtype.itemsize
    total = height * row_elems * itemsize
    # Ensure the file exists at full size (rank-parallel writers each own
    # disjoint regions; os.O_CREAT + pwrite keeps them independent).
    C = _native_io()
    if C is not None:
        import torch

        blk_t = torch.from_numpy(np.ascontiguousarray(block))
        C.write_block(
            path, row_elems * itemsize, total, x0 * channels * itemsize, y0, blk_t
        )
        return
    fd = os.open(path, os.O_RDWR | os.O_CREAT, 0o644)
    try:
        if os.fstat(fd).st_size < total:
            os.ftruncate(fd, total)
        blk = np.ascontiguousarray(block)
        for r in range(h):
            off = ((y0 + r) * row_elems + x0 * channels) * itemsize
            os.pwrite(fd, blk[r].tobytes(), off)
    finally:
        os.close(fd)


_U64 = np.uint64


def _splitmix64(x: np.ndarray) -> np.ndarray:
    """Vectorized splitmix64 finalizer (counter-based RNG: value depends
    only on the input word, so any sub-rectangle of the global image can
    be generated independently)."""
    z = (x + _U64(0x9E3779B97F4A7C15)).astype(np.uint64)
    z ^= z >> _U64(30)
    z *= _U64(0xBF58476D1CE4E5B9)
    z ^= z >> _U64(27)
    z *= _U64(0x94D049BB133111EB)
    z ^= z >> _U64(31)
    return z


def synthetic_block(
    height: int,
    width: int,
    channels: int,
    dtype: DType,
    y0: int,
    x0: int,
    h: int,
    w: int,
    seed: int = 1234,
) -> np.ndarray:
    """One (h, w)-pixel window of the seeded synthetic global image.

    World-size-INVARIANT: pixel (y, x, c) of the global height×width frame
    has the same value no matter which rank generates it or how the frame
    is decomposed, so 1-GPU and N-GPU runs on `synthetic` input produce
    comparable outputs (counter-based hash of the global element index).
    Returns (h, w*channels).
    """
    wE = w * channels
    row_e = width * channels
    out = np.empty((h, wE), dtype=dtype.numpy_dtype)
    sk = _splitmix64(np.array([seed & 0xFFFFFFFFFFFFFFFF], dtype=np.uint64))[0]
    base_x = np.arange(x0 * channels, x0 * channels + wE, dtype=np.uint64)
    # row slabs bound the u64 intermediate to ~1 GB (the 65536² config
    # would otherwise need a 34 GB temporary)
    slab = max(1, (1 << 27) // max(wE, 1))
    for r0 in range(0, h, slab):
        r1 = min(r0 + slab, h)
        ys = np.arange(y0 + r0, y0 + r1, dtype=np.uint64) * _U64(row_e)
        hv = _splitmix64((ys[:, None] + base_x[None, :]) ^ sk)
        if dtype == DType.U8:
            out[r0:r1] = (hv & _U64(0xFF)).astype(np.uint8)
        else:
            # floats in [0, 256) — same dynamic range as u8
            out[r0:r1] = (
                (hv >> _U64(40)).astype(np.float32) * np.float32(256.0 / (1 << 24))
            ).astype(dtype.numpy_dtype)
    return out


def synthetic_image(
    height: int, width: int, channels: int, dtype: DType, seed: int = 1234
) -> np.ndarray:
    """Seeded random-pixel frame (BASELINE.json north-star: benchmarks run
    'on synthetic random-pixel images'); (height, width*channels)."""
    return synthetic_block(
        height, width, channels, dtype, 0, 0, height, width, seed=seed
    )
