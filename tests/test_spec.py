"""ProblemSpec and metrics-record validation."""

import numpy as np
import pytest

from parallel_convolution_amd.models.filters import GAUSSIAN_BLUR_3
from parallel_convolution_amd.spec import DType, ProblemSpec
from parallel_convolution_amd.utils import metrics


def test_spec_validation_errors():
    with pytest.raises(ValueError):
        ProblemSpec(0, 10, 1, DType.U8, GAUSSIAN_BLUR_3, 1)
    with pytest.raises(ValueError):
        ProblemSpec(10, 10, 2, DType.U8, GAUSSIAN_BLUR_3, 1)  # channels
    with pytest.raises(ValueError):
        ProblemSpec(10, 10, 1, DType.U8, np.ones((2, 2), np.float32), 1)
    with pytest.raises(ValueError):
        ProblemSpec(10, 10, 1, DType.U8, np.ones((3, 4), np.float32), 1)
    with pytest.raises(ValueError):
        ProblemSpec(10, 10, 1, DType.U8, GAUSSIAN_BLUR_3, -1)


def test_spec_properties():
    s = ProblemSpec(64, 48, 3, DType.F16, GAUSSIAN_BLUR_3, 7)
    assert s.k == 3 and s.radius == 1
    assert s.elems_per_row == 192
    assert s.nbytes == 48 * 192 * 2
    assert s.dtype.itemsize == 2


def test_run_record_fields(tmp_path):
    s = ProblemSpec(100, 50, 1, DType.U8, GAUSSIAN_BLUR_3, 10)
    rec = metrics.run_record(s, 4, 10, 0.5, "hip", extra={"x": 1})
    assert rec["value"] == round(100 * 50 * 10 / 0.5 / 1e9, 3)
    assert rec["n_ranks"] == 4 and rec["x"] == 1
    p = tmp_path / "m.jsonl"
    metrics.emit(rec, str(p))
    import json

    assert json.loads(p.read_text())["backend"] == "hip"
