"""Process bootstrap: one process per GPU (or per CPU rank).

Replaces the reference's mpiexec/MPI_Init world with
torch.distributed. The process group itself is only the CONTROL plane
(geometry agreement, RCCL unique-id broadcast, final barrier); the data
plane on GPU is the C++ engine's own RCCL communicator over xGMI, and on
CPU the gloo p2p exchanger.

Always rendezvous via 127.0.0.1 — the container hostname may not resolve.
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist


def env_rank() -> int:
    return int(os.environ.get("RANK", "0"))


def env_world() -> int:
    return int(os.environ.get("WORLD_SIZE", "1"))


def env_local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", "0")))


def init_control_group(backend: str = "gloo", port: Optional[int] = None) -> None:
    """Initialize the control-plane process group from torchrun env vars
    (or as a trivial world of 1)."""
    if dist.is_initialized():
        return
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    # Under torchrun MASTER_PORT is always set; this fallback (world of 1,
    # or hand-launched ranks) prefers PCONV_MASTER_PORT over the default.
    os.environ.setdefault(
        "MASTER_PORT", str(port or os.environ.get("PCONV_MASTER_PORT") or 29531)
    )
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    dist.init_process_group(backend=backend)


def broadcast_bytes(payload: Optional[bytes], src: int = 0, length: int = 128) -> bytes:
    """Broadcast a fixed-length byte string (e.g. ncclUniqueId) over the
    control group."""
    t = torch.zeros(length, dtype=torch.uint8)
    if dist.get_rank() == src:
        assert payload is not None and len(payload) <= length
        t[: len(payload)] = torch.tensor(list(payload), dtype=torch.uint8)
    dist.broadcast(t, src=src)
    return bytes(t.tolist())


def finalize() -> None:
    if dist.is_initialized():
        dist.barrier()
        dist.destroy_process_group()
