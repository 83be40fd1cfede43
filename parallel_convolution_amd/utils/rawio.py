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


def synthetic_image(
    height: int, width: int, channels: int, dtype: DType, seed: int = 1234
) -> np.ndarray:
    """Seeded random-pixel frame (BASELINE.json north-star: benchmarks run
    'on synthetic random-pixel images'); (height, width*channels)."""
    rng = np.random.default_rng(seed)
    shape = (height, width * channels)
    if dtype == DType.U8:
        return rng.integers(0, 256, size=shape, dtype=np.uint8)
    # float images in [0, 256) — same dynamic range as u8; generated in
    # row slabs so the fp32 intermediate never exceeds ~1 GB (the 65536²
    # fp16 config would otherwise need a 17 GB temporary)
    out = np.empty(shape, dtype=dtype.numpy_dtype)
    slab = max(1, (1 << 28) // max(shape[1], 1))
    for y0 in range(0, height, slab):
        y1 = min(y0 + slab, height)
        out[y0:y1] = (rng.random((y1 - y0, shape[1]), dtype=np.float32) * 256).astype(
            dtype.numpy_dtype
        )
    return out
