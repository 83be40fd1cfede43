"""End-to-end CLI tests (CPU backend, reference CLI surface)."""

import json
import os
import subprocess
import sys

import numpy as np
import pytest

from parallel_convolution_amd.models.filters import GAUSSIAN_BLUR_3, write_filter
from parallel_convolution_amd.ops.reference import run_reference
from parallel_convolution_amd.spec import DType, ProblemSpec
from parallel_convolution_amd.utils import rawio

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_cli(args, **kw):
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    return subprocess.run(
        [sys.executable, "-m", "parallel_convolution_amd"] + args,
        cwd=REPO, env=env, capture_output=True, text=True, timeout=240, **kw
    )


@pytest.fixture
def image_and_filter(tmp_path):
    img = rawio.synthetic_image(48, 64, 1, DType.U8)
    ip = str(tmp_path / "img.raw")
    rawio.write_image(ip, img)
    fp = str(tmp_path / "filt.txt")
    write_filter(fp, GAUSSIAN_BLUR_3)
    return ip, fp, img


def test_cli_single_rank(image_and_filter, tmp_path):
    ip, fp, img = image_and_filter
    out = str(tmp_path / "out.raw")
    r = run_cli([ip, "64", "48", fp, "6", "--backend", "cpu", "--output", out])
    assert r.returncode == 0, r.stderr
    got = rawio.read_image(out, 48, 64, 1, DType.U8)
    spec = ProblemSpec(64, 48, 1, DType.U8, GAUSSIAN_BLUR_3, 6)
    np.testing.assert_array_equal(got, run_reference(spec, img))


def test_cli_default_output_name(image_and_filter):
    ip, fp, img = image_and_filter
    r = run_cli([ip, "64", "48", fp, "2", "--backend", "cpu"])
    assert r.returncode == 0, r.stderr
    assert os.path.exists(ip + "_out.raw")


def test_cli_builtin_filter_and_metrics(image_and_filter, tmp_path):
    ip, _, img = image_and_filter
    mpath = str(tmp_path / "m.jsonl")
    r = run_cli([ip, "64", "48", "blur3", "3", "--backend", "cpu",
                 "--no-output", "--metrics-out", mpath])
    assert r.returncode == 0, r.stderr
    rec = json.loads(open(mpath).read().strip())
    assert rec["metric"] == "Gpixel-iters/sec" and rec["iters"] == 3
    assert rec["n_ranks"] == 1 and rec["backend"] == "cpu"


def test_cli_two_ranks_cpu(image_and_filter, tmp_path):
    ip, fp, img = image_and_filter
    out = str(tmp_path / "out2.raw")
    r = run_cli([ip, "64", "48", fp, "7", "--backend", "cpu", "--gpus", "2",
                 "--output", out])
    assert r.returncode == 0, r.stderr + r.stdout
    got = rawio.read_image(out, 48, 64, 1, DType.U8)
    spec = ProblemSpec(64, 48, 1, DType.U8, GAUSSIAN_BLUR_3, 7)
    np.testing.assert_array_equal(got, run_reference(spec, img))


def test_cli_convergence_early_exit(image_and_filter, tmp_path):
    ip, _, img = image_and_filter
    out = str(tmp_path / "outc.raw")
    r = run_cli([ip, "64", "48", "identity3", "50", "--backend", "cpu",
                 "--converge-every", "5", "--converge-eps", "0.0",
                 "--output", out, "--metrics-out", str(tmp_path / "m.jsonl")])
    assert r.returncode == 0, r.stderr
    rec = json.loads(open(str(tmp_path / "m.jsonl")).read().strip())
    assert rec["converged"] is True
    assert rec["iters"] == 5  # stopped at the first check
    got = rawio.read_image(out, 48, 64, 1, DType.U8)
    np.testing.assert_array_equal(got, img)  # identity filter


def test_cli_checkpoint_resume(image_and_filter, tmp_path):
    ip, fp, img = image_and_filter
    out = str(tmp_path / "outr.raw")
    # phase 1: 4 iters with checkpointing
    r = run_cli([ip, "64", "48", fp, "4", "--backend", "cpu",
                 "--checkpoint-every", "4", "--output", out])
    assert r.returncode == 0, r.stderr
    assert os.path.exists(out + ".ckpt.json")
    # phase 2: resume to 10 total
    r = run_cli([ip, "64", "48", fp, "10", "--backend", "cpu",
                 "--checkpoint-every", "4", "--resume", "--output", out])
    assert r.returncode == 0, r.stderr
    got = rawio.read_image(out, 48, 64, 1, DType.U8)
    spec = ProblemSpec(64, 48, 1, DType.U8, GAUSSIAN_BLUR_3, 10)
    np.testing.assert_array_equal(got, run_reference(spec, img))


def test_cli_synthetic_rgb(tmp_path):
    out = str(tmp_path / "rgb.raw")
    r = run_cli(["synthetic", "32", "24", "sharpen5", "2", "--rgb",
                 "--backend", "cpu", "--output", out])
    assert r.returncode == 0, r.stderr
    got = rawio.read_image(out, 24, 32, 3, DType.U8)
    img = rawio.synthetic_image(24, 32, 3, DType.U8, seed=1234)
    from parallel_convolution_amd.models.filters import SHARPEN_5

    spec = ProblemSpec(32, 24, 3, DType.U8, SHARPEN_5, 2)
    np.testing.assert_array_equal(got, run_reference(spec, img))


def test_bench_cpu_world2(tmp_path):
    """bench.py's distributed flow (torchrun world 2) on the CPU/gloo
    backend — exercises the exact rank/barrier/allreduce path the 8-GPU
    driver run takes, minus RCCL."""
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=2", "--master-addr=127.0.0.1",
         "--master-port=29617", "bench.py", "--gpus", "2", "--steps", "3",
         "--warmup", "1", "--preset", "single-8192", "--backend", "cpu"],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=600,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    rec = json.loads(line)
    assert rec["n_gpus"] == 2 and rec["steps"] == 3
    assert rec["metric"] == "Gpixel-iters/sec"


def test_cli_grid_override(image_and_filter, tmp_path):
    ip, fp, img = image_and_filter
    out = str(tmp_path / "outg.raw")
    mpath = str(tmp_path / "mg.jsonl")
    r = run_cli([ip, "64", "48", fp, "5", "--backend", "cpu", "--gpus", "2",
                 "--grid", "2x1", "--output", out, "--metrics-out", mpath])
    assert r.returncode == 0, r.stderr + r.stdout
    rec = json.loads(open(mpath).read().strip())
    assert rec["grid"] == "2x1"  # override actually applied (default is 1x2)
    got = rawio.read_image(out, 48, 64, 1, DType.U8)
    spec = ProblemSpec(64, 48, 1, DType.U8, GAUSSIAN_BLUR_3, 5)
    np.testing.assert_array_equal(got, run_reference(spec, img))


def test_cli_wrong_image_size_fails_cleanly(image_and_filter, tmp_path):
    ip, fp, img = image_and_filter  # 64x48 image
    r = run_cli([ip, "128", "128", fp, "2", "--backend", "cpu",
                 "--output", str(tmp_path / "x.raw")])
    assert r.returncode != 0
    assert "short read" in r.stderr or "expected" in r.stderr


def test_cli_missing_filter_file_fails(image_and_filter, tmp_path):
    ip, _, _ = image_and_filter
    r = run_cli([ip, "64", "48", str(tmp_path / "nope.txt"), "2",
                 "--backend", "cpu", "--no-output"])
    assert r.returncode != 0


def test_cli_bad_filter_contents(image_and_filter, tmp_path):
    ip, _, _ = image_and_filter
    bad = tmp_path / "bad.txt"
    bad.write_text("4\n1 2 3 4\n")  # even k
    r = run_cli([ip, "64", "48", str(bad), "2", "--backend", "cpu",
                 "--no-output"])
    assert r.returncode != 0
    assert "odd" in r.stderr


def test_cli_watchdog_smoke(image_and_filter, tmp_path):
    ip, fp, img = image_and_filter
    # generous watchdog: must not fire on a normal run
    r = run_cli([ip, "64", "48", fp, "3", "--backend", "cpu",
                 "--watchdog", "120", "--no-output"])
    assert r.returncode == 0, r.stderr


def test_next_stop_boundaries():
    """Segments run to the NEXT boundary: --converge-every 2
    --checkpoint-every 3 over 10 iters stops at 2,3,4,6,8,9,10 — no
    boundary skipped (ADVICE.md r01 finding)."""
    from parallel_convolution_amd.cli import next_stop

    stops = []
    done = 0
    while done < 10:
        done = next_stop(done, 10, 2, 3)
        stops.append(done)
    assert stops == [2, 3, 4, 6, 8, 9, 10]
    # single boundary kind
    assert next_stop(0, 100, 0, 30) == 30
    assert next_stop(30, 100, 0, 30) == 60
    assert next_stop(95, 100, 0, 30) == 100
    # no boundaries: run to the end
    assert next_stop(0, 7, 0, 0) == 7
    # boundary beyond the end clamps
    assert next_stop(0, 5, 8, 0) == 5
