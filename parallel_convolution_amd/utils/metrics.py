"""Run metrics: one JSON line per run.

Replaces the reference's printf-of-elapsed-time with a structured record:
config, ranks, timing, Gpixel-iters/s — the same shape bench.py emits, so
BASELINE.md tables can be assembled directly from run logs.
"""

from __future__ import annotations

import json
import sys
import time
from typing import Optional


def run_record(
    spec,
    n_ranks: int,
    iters: int,
    elapsed_s: float,
    backend: str,
    extra: Optional[dict] = None,
) -> dict:
    px_iters = float(spec.width) * spec.height * iters
    rec = {
        "ts": time.strftime("%Y-%m-%dT%H:%M:%S"),
        "metric": "Gpixel-iters/sec",
        "value": round(px_iters / elapsed_s / 1e9, 3) if elapsed_s > 0 else None,
        "elapsed_s": round(elapsed_s, 6),
        "iters": iters,
        "n_ranks": n_ranks,
        "backend": backend,
        "image": f"{spec.width}x{spec.height}x{spec.channels}",
        "dtype": spec.dtype.value,
        "k": spec.k,
    }
    if extra:
        rec.update(extra)
    return rec


def emit(rec: dict, path: Optional[str] = None) -> None:
    line = json.dumps(rec)
    print(line, file=sys.stderr, flush=True)
    if path:
        with open(path, "a") as f:
            f.write(line + "\n")
