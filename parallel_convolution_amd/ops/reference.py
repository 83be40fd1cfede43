"""Correctness oracle: plain PyTorch fp32 iterative convolution.

Plays the role the reference's serial variant plays for its MPI variants
(serial run = correctness oracle, outputs byte-compared).
Every engine backend (CPU-torch, HIP single-GPU, HIP multi-GPU over RCCL)
must reproduce this bit-exactly for u8 with dyadic filters, and to fp
tolerance otherwise (spec.py documents the arithmetic contract).
"""

from __future__ import annotations

import numpy as np
import torch
import torch.nn.functional as F

from ..spec import DType, ProblemSpec


def _quantize(acc: torch.Tensor, dtype: DType) -> torch.Tensor:
    """Per-iteration storage rounding — the single definition of the
    arithmetic contract (spec.py): round-half-even, clamp for u8."""
    if dtype == DType.U8:
        return torch.clamp(torch.round(acc), 0.0, 255.0)
    if dtype == DType.F16:
        return acc.to(torch.float16).to(torch.float32)
    return acc


def conv_once(img_f32: torch.Tensor, filt: torch.Tensor, channels: int) -> torch.Tensor:
    """One zero-padded k×k convolution. img_f32: (C, H, W) fp32."""
    k = filt.shape[0]
    r = k // 2
    weight = filt.reshape(1, 1, k, k).expand(channels, 1, k, k).contiguous()
    x = img_f32.unsqueeze(0)
    y = F.conv2d(x, weight, padding=r, groups=channels)
    return y.squeeze(0)


def run_reference(spec: ProblemSpec, image: np.ndarray, iters: int | None = None) -> np.ndarray:
    """Iterate the convolution on a whole frame.

    image: (H, W*channels) storage layout (interleaved for RGB).
    Returns the same layout/dtype.
    """
    iters = spec.iters if iters is None else iters
    H, W, C = spec.height, spec.width, spec.channels
    assert image.shape == (H, W * C), image.shape
    # de-interleave to (C, H, W) fp32
    x = torch.from_numpy(np.ascontiguousarray(image)).reshape(H, W, C).permute(2, 0, 1)
    x = x.to(torch.float32)
    filt = torch.from_numpy(np.ascontiguousarray(spec.filter)).to(torch.float32)
    # conv2d cross-correlates; our contract is also cross-correlation style:
    # out[y][x] = sum_{dy,dx in [-r,r]} F[dy+r][dx+r] * in[y+dy][x+dx]
    for _ in range(iters):
        x = _quantize(conv_once(x, filt, C), spec.dtype)
    out = x.permute(1, 2, 0).reshape(H, W * C).contiguous()
    if spec.dtype == DType.U8:
        return out.to(torch.uint8).numpy()
    if spec.dtype == DType.F16:
        return out.to(torch.float16).numpy()
    return out.numpy()
