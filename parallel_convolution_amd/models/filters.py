"""Built-in convolution filters and the filter-file format.

The reference hardcodes (or reads from a kernel file named on the CLI,
BASELINE.json:5) a normalized k×k filter; the classic assignment filter is
the 3×3 Gaussian-style blur 1/16·{1,2,1;2,4,2;1,2,1}.

Filter-file format (defined here; the reference tree was unreadable,
SURVEY.md §0): whitespace-separated text — first token is the odd kernel
size k, followed by k*k floats in row-major order. `#` starts a comment to
end of line.
"""

from __future__ import annotations

import numpy as np

# 3×3 Gaussian-style blur — the BASELINE.json:7-9 configs' filter.
# Dyadic coefficients (n/16) => fp32 accumulation is exact for u8 inputs,
# so parallel/GPU runs are byte-exact vs the oracle (spec.py docstring).
GAUSSIAN_BLUR_3 = np.array(
    [[1, 2, 1], [2, 4, 2], [1, 2, 1]], dtype=np.float32
) / np.float32(16.0)

# 5×5 blur (dyadic, /256 binomial) for the k=5 paths.
_b5 = np.array([1, 4, 6, 4, 1], dtype=np.float32)
GAUSSIAN_BLUR_5 = np.outer(_b5, _b5) / np.float32(256.0)

# 5×5 sharpen — BASELINE.json:10 names "5×5 sharpen" for the RGB config.
# Unsharp mask: identity + (identity - blur5): dyadic, sums to 1.
SHARPEN_5 = (-GAUSSIAN_BLUR_5).copy()
SHARPEN_5[2, 2] += np.float32(2.0)

IDENTITY_3 = np.zeros((3, 3), dtype=np.float32)
IDENTITY_3[1, 1] = 1.0

EDGE_3 = np.array(
    [[-1, -1, -1], [-1, 8, -1], [-1, -1, -1]], dtype=np.float32
)

BUILTIN = {
    "blur3": GAUSSIAN_BLUR_3,
    "blur5": GAUSSIAN_BLUR_5,
    "sharpen5": SHARPEN_5,
    "identity3": IDENTITY_3,
    "edge3": EDGE_3,
}


def parse_filter_text(text: str) -> np.ndarray:
    toks = []
    for line in text.splitlines():
        line = line.split("#", 1)[0]
        toks.extend(line.split())
    if not toks:
        raise ValueError("empty filter file")
    k = int(toks[0])
    if k <= 0 or k % 2 == 0:
        raise ValueError(f"filter size must be odd positive, got {k}")
    vals = [float(t) for t in toks[1:]]
    if len(vals) != k * k:
        raise ValueError(f"expected {k*k} coefficients, got {len(vals)}")
    return np.array(vals, dtype=np.float32).reshape(k, k)


def read_filter(path: str) -> np.ndarray:
    """Load a filter: builtin name (e.g. 'blur3') or filter-file path."""
    if path in BUILTIN:
        return BUILTIN[path].copy()
    with open(path, "r") as f:
        return parse_filter_text(f.read())


def write_filter(path: str, filt: np.ndarray) -> None:
    filt = np.asarray(filt, dtype=np.float32)
    k = filt.shape[0]
    with open(path, "w") as f:
        f.write(f"{k}\n")
        for row in filt:
            f.write(" ".join(repr(float(v)) for v in row) + "\n")
