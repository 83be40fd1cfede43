"""Problem specification shared by every engine backend.

The behavioral contract this implements is BASELINE.json:5 (the reference
repo jimouris/parallel-convolution was not mounted — see SURVEY.md §0):
iterative k×k 2D convolution over raw images, grayscale (1 byte/px) and
packed-RGB (3 interleaved bytes/px), CLI surface = (image path, W×H,
kernel/filter file, iteration count), raw headerless input/output.

Semantics defined HERE (the reference tree was unreadable, SURVEY.md §7
"u8 arithmetic path" mandates defining and documenting them):

* Boundary: zero padding. Pixels outside the image read as 0; every image
  pixel is convolved (no skipped border strip). This makes the kernel
  uniform over the whole domain and is exactly reproducible by the oracle
  (`ops/reference.py`).
* Accumulation: fp32, taps applied in row-major (dy, dx) order.
* u8 storage: result = clamp(round_half_to_even(acc), 0, 255) per
  iteration (round-half-even == torch.round == np.rint == device
  __float2int_rn, so oracle and kernels agree bit-exactly for dyadic
  filters).
* fp16 storage: acc rounded to nearest-even half per iteration.
* fp32 storage: acc stored as-is.
"""

from __future__ import annotations

import dataclasses
import enum


class DType(str, enum.Enum):
    U8 = "u8"
    F16 = "fp16"
    F32 = "fp32"

    @property
    def itemsize(self) -> int:
        return {DType.U8: 1, DType.F16: 2, DType.F32: 4}[self]

    @property
    def torch_dtype(self):
        import torch

        return {
            DType.U8: torch.uint8,
            DType.F16: torch.float16,
            DType.F32: torch.float32,
        }[self]

    @property
    def numpy_dtype(self):
        import numpy as np

        return {DType.U8: np.uint8, DType.F16: np.float16, DType.F32: np.float32}[self]


@dataclasses.dataclass(frozen=True)
class ProblemSpec:
    """One iterative-convolution problem instance.

    width/height are in PIXELS; channels is 1 (gray) or 3 (packed RGB,
    3 interleaved storage elements per pixel, BASELINE.json:5).
    """

    width: int
    height: int
    channels: int  # 1 or 3
    dtype: DType
    filter: "object"  # numpy (k, k) float32 array
    iters: int

    def __post_init__(self):
        import numpy as np

        if self.channels not in (1, 3):
            raise ValueError(f"channels must be 1 or 3, got {self.channels}")
        f = np.asarray(self.filter, dtype=np.float32)
        if f.ndim != 2 or f.shape[0] != f.shape[1] or f.shape[0] % 2 != 1:
            raise ValueError(f"filter must be odd square 2D, got shape {f.shape}")
        object.__setattr__(self, "filter", f)
        if self.width <= 0 or self.height <= 0:
            raise ValueError("image dims must be positive")
        if self.iters < 0:
            raise ValueError("iters must be >= 0")

    @property
    def k(self) -> int:
        return int(self.filter.shape[0])

    @property
    def radius(self) -> int:
        return self.k // 2

    @property
    def elems_per_row(self) -> int:
        """Storage elements per image row (channels interleaved)."""
        return self.width * self.channels

    @property
    def nbytes(self) -> int:
        return self.height * self.elems_per_row * self.dtype.itemsize
