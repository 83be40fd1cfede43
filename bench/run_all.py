"""Run the mandated BASELINE.json configs that fit this machine.

Config 1 (CPU plumbing) runs anywhere; configs 2-5 need a GPU. Multi-GPU
scaling points are launched by the driver (torchrun, one rank per GPU) —
this harness covers the single-process runs and emits one JSON line each
(bench.py contract) into bench/results.jsonl.

Usage: python bench/run_all.py [--steps N] [--out PATH]
"""

import argparse
import json
import os
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def run_config1(out):
    """1024² gray, 3×3, 10 iters, 1 rank on CPU (BASELINE.json:7)."""
    import numpy as np  # noqa: F401

    from parallel_convolution_amd.cpu_engine import CpuBlock, CpuEngine
    from parallel_convolution_amd.models.filters import GAUSSIAN_BLUR_3
    from parallel_convolution_amd.parallel.decomp import make_blocks
    from parallel_convolution_amd.spec import DType, ProblemSpec
    from parallel_convolution_amd.utils.rawio import synthetic_image

    spec = ProblemSpec(1024, 1024, 1, DType.U8, GAUSSIAN_BLUR_3, 10)
    img = synthetic_image(1024, 1024, 1, DType.U8)
    blk = CpuBlock(spec, make_blocks(1024, 1024, 1, 1)[0])
    blk.load(img)
    eng = CpuEngine(spec, [blk])
    t0 = time.perf_counter()
    eng.run()
    dt = time.perf_counter() - t0
    rec = {
        "metric": "Gpixel-iters/sec",
        "value": round(1024 * 1024 * 10 / dt / 1e9, 3),
        "unit": "Gpixel-iters/s",
        "n_gpus": 0,
        "config": {"preset": "config1-cpu", "image": "1024x1024x1", "iters": 10},
        "backend": "cpu",
    }
    print(json.dumps(rec), flush=True)
    out.write(json.dumps(rec) + "\n")


def run_gpu_preset(preset, steps, warmup, out):
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--preset", preset,
         "--steps", str(steps), "--warmup", str(warmup)],
        capture_output=True, text=True, timeout=1800,
    )
    line = r.stdout.strip().splitlines()[-1] if r.stdout.strip() else ""
    if r.returncode != 0 or not line.startswith("{"):
        print(f"[run_all] preset {preset} FAILED: {r.stderr[-500:]}", file=sys.stderr)
        return
    print(line, flush=True)
    out.write(line + "\n")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=100)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--out", default=os.path.join(REPO, "bench", "results.jsonl"))
    args = ap.parse_args()

    import torch

    with open(args.out, "a") as out:
        run_config1(out)
        if torch.cuda.is_available():
            for preset in ("single-8192", "headline", "rgb5", "fp16-65536"):
                steps = args.steps if preset != "fp16-65536" else min(args.steps, 20)
                run_gpu_preset(preset, steps, args.warmup, out)
        else:
            print("[run_all] no GPU: configs 2-5 skipped", file=sys.stderr)


if __name__ == "__main__":
    main()
