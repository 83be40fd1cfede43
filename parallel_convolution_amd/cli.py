"""Reference-compatible CLI (BASELINE.json:5 surface).

    python -m parallel_convolution_amd IMAGE W H FILTER ITERS [options]

Positional arguments mirror the MPI/OpenMP C reference: raw image path,
width, height, kernel/filter file (or builtin name like 'blur3'), iteration
count. Engine-only flags are strictly additive. The
output is written as raw to `<image>_out.raw` (reference-style naming,
SURVEY.md §2.1 R6) unless --output is given.

Multi-GPU: `--gpus N` re-launches under torch.distributed.run with one
rank per GPU (rendezvous on 127.0.0.1); each rank reads/writes only its
own block.
"""

from __future__ import annotations

import argparse
import json
import os
import signal
import subprocess
import sys
import time

from .models.filters import read_filter
from .parallel import bootstrap
from .parallel.decomp import halo_for_radius, make_blocks
from .spec import DType, ProblemSpec
from .utils import metrics, rawio


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(
        prog="parallel_convolution_amd",
        description="MI355X-native iterative 2D convolution",
    )
    p.add_argument("image", help="raw input image path, or 'synthetic'")
    p.add_argument("width", type=int)
    p.add_argument("height", type=int)
    p.add_argument("filter", help="filter file path or builtin name (blur3, blur5, sharpen5, ...)")
    p.add_argument("iters", type=int)
    p.add_argument("--rgb", action="store_true", help="packed-RGB image (3 B/px)")
    p.add_argument("--dtype", choices=["u8", "fp16", "fp32"], default="u8")
    p.add_argument("--output", default=None, help="output raw path (default <image>_out.raw)")
    p.add_argument("--gpus", type=int, default=1, help="GPUs (1 process per GPU)")
    p.add_argument("--backend", choices=["auto", "hip", "cpu"], default="auto")
    p.add_argument("--chunk", type=int, default=None, help="halo-deepening chunk iterations")
    p.add_argument("--grid", default=None, metavar="RxC",
                   help="process-grid override (default: near-square factorization)")
    p.add_argument("--seed", type=int, default=1234, help="seed for synthetic input")
    p.add_argument("--converge-every", type=int, default=0, metavar="N",
                   help="check max|Δ| every N iters, stop early below eps")
    p.add_argument("--converge-eps", type=float, default=0.0)
    p.add_argument("--checkpoint-every", type=int, default=0, metavar="N",
                   help="dump state every N iters (resumable)")
    p.add_argument("--resume", action="store_true", help="resume from checkpoint")
    p.add_argument("--watchdog", type=int, default=0, metavar="SEC",
                   help="abort if an engine segment exceeds SEC seconds")
    p.add_argument("--master-port", type=int, default=0, metavar="PORT",
                   help="rendezvous port for --gpus N (0 = pick a free one)")
    p.add_argument("--metrics-out", default=None, help="append JSON run record here")
    p.add_argument("--phase-timing", action="store_true",
                   help="per-phase (exchange/compute) hipEvent timing in the record")
    p.add_argument("--no-output", action="store_true", help="skip writing the output image")
    return p


def main(argv=None) -> int:
    args = build_parser().parse_args(argv)
    world_env = bootstrap.env_world()
    if args.gpus > 1 and world_env == 1:
        # re-exec under torchrun, one rank per GPU; rendezvous on a free
        # ephemeral port by default so concurrent runs on one host never
        # collide (--master-port pins it when reproducibility matters)
        port = args.master_port or _free_port()
        cmd = [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", f"--nproc-per-node={args.gpus}",
            "--master-addr=127.0.0.1", f"--master-port={port}",
            "-m", "parallel_convolution_amd",
        ] + (argv if argv is not None else sys.argv[1:])
        return subprocess.call(cmd)
    return run_rank(args)


def next_stop(done: int, iters: int, converge_every: int, checkpoint_every: int) -> int:
    """First iteration count > done at which a converge/checkpoint
    boundary (or the end) falls. Segments run to the NEXT upcoming
    boundary, so non-divisible --converge-every/--checkpoint-every
    combinations hit every point."""
    stops = [iters]
    for every in (converge_every, checkpoint_every):
        if every and every > 0:
            stops.append((done // every + 1) * every)
    return min(min(stops), iters)


def _free_port() -> int:
    import socket

    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _alarm_handler(signum, frame):
    raise TimeoutError("watchdog: engine segment exceeded time limit")


def run_rank(args) -> int:
    import torch
    import torch.distributed as dist

    rank = bootstrap.env_rank()
    world = bootstrap.env_world()
    local_rank = bootstrap.env_local_rank()

    spec = ProblemSpec(
        args.width,
        args.height,
        3 if args.rgb else 1,
        DType(args.dtype),
        read_filter(args.filter),
        args.iters,
    )
    backend = args.backend
    if backend == "auto":
        backend = "hip" if torch.cuda.is_available() else "cpu"

    if world > 1:
        bootstrap.init_control_group("gloo")
    if backend == "hip":
        torch.cuda.set_device(local_rank)

    grid = None
    if args.grid:
        gr, gc = args.grid.lower().split("x")
        grid = (int(gr), int(gc))
    blocks = make_blocks(
        spec.height,
        spec.width,
        world,
        spec.channels,
        halo=halo_for_radius(spec.radius),
        grid=grid,
    )
    my = blocks[rank]
    S = spec.channels

    out_path = args.output or (
        args.image + "_out.raw" if args.image != "synthetic" else "synthetic_out.raw"
    )
    ckpt_raw = out_path + ".ckpt.raw"
    ckpt_meta = out_path + ".ckpt.json"

    # --- input ---------------------------------------------------------
    iters_done = 0
    if args.resume and os.path.exists(ckpt_meta):
        with open(ckpt_meta) as f:
            iters_done = json.load(f)["iters_done"]
        img_block = rawio.read_block(
            ckpt_raw, spec.height, spec.width, S, spec.dtype, my.y0, my.x0, my.h, my.w
        )
    elif args.image == "synthetic":
        # world-size-invariant: each rank hashes its window of the same
        # seeded global image, so outputs match across GPU counts
        img_block = rawio.synthetic_block(
            spec.height, spec.width, S, spec.dtype,
            my.y0, my.x0, my.h, my.w, seed=args.seed,
        )
    else:
        img_block = rawio.read_block(
            args.image, spec.height, spec.width, S, spec.dtype, my.y0, my.x0, my.h, my.w
        )

    # --- engine --------------------------------------------------------
    if backend == "hip":
        from .hip_engine import HipEngine

        eng = HipEngine(spec, [my], world_blocks=blocks, chunk_iters=args.chunk)
        if world > 1:
            eng.init_distributed(rank, world)
        if args.phase_timing:
            eng.set_phase_timing(True)
        eng.load_block(0, img_block)

        def run_seg(n):
            eng.run(n)

        def diff():
            return eng.diff_max()

        def store():
            return eng.store_block(0)

    else:
        from .cpu_engine import CpuBlock, CpuEngine, GlooExchanger

        blk = CpuBlock(spec, my)
        blk.load(img_block)
        ex = GlooExchanger(blk) if world > 1 else None
        eng = CpuEngine(spec, [blk], exchanger=ex, chunk_iters=args.chunk)

        def run_seg(n):
            eng.run(n)

        def diff():
            return eng.diff_max()

        def store():
            return blk.store()

    # --- iterate with convergence/checkpoint segments (next_stop) ------
    if world > 1:
        dist.barrier()
    if backend == "hip":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    converged = False
    while iters_done < args.iters and not converged:
        n = next_stop(iters_done, args.iters, args.converge_every,
                      args.checkpoint_every) - iters_done
        if args.watchdog > 0:
            signal.signal(signal.SIGALRM, _alarm_handler)
            signal.alarm(args.watchdog)
        run_seg(n)
        if args.watchdog > 0:
            signal.alarm(0)
        iters_done += n
        if args.converge_every and iters_done % args.converge_every == 0:
            d = diff()
            if world > 1:
                t = torch.tensor([d])
                dist.all_reduce(t, op=dist.ReduceOp.MAX)
                d = float(t.item())
            if d <= args.converge_eps:
                converged = True
                if rank == 0:
                    print(f"converged at iter {iters_done}: max|Δ| = {d}", file=sys.stderr)
        if args.checkpoint_every and (
            iters_done % args.checkpoint_every == 0 or iters_done >= args.iters
        ):
            rawio.write_block(
                ckpt_raw, spec.height, spec.width, S, spec.dtype, my.y0, my.x0, store()
            )
            if world > 1:
                dist.barrier()
            if rank == 0:
                with open(ckpt_meta, "w") as f:
                    json.dump({"iters_done": iters_done}, f)
    if backend == "hip":
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    # --- output --------------------------------------------------------
    if not args.no_output:
        rawio.write_block(
            out_path, spec.height, spec.width, S, spec.dtype, my.y0, my.x0, store()
        )
    if rank == 0:
        rec = metrics.run_record(
            spec,
            world,
            iters_done,
            elapsed,
            backend,
            extra={
                "converged": converged,
                "grid": "x".join(map(str, my.grid)),
                "output": None if args.no_output else out_path,
                **(
                    {"phases": eng.phase_times()}
                    if args.phase_timing and backend == "hip"
                    else {}
                ),
            },
        )
        metrics.emit(rec, args.metrics_out)
    if world > 1:
        bootstrap.finalize()
    return 0


if __name__ == "__main__":
    sys.exit(main())
