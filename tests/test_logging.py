"""Logging/metrics helpers: SmoothedValue stats, MetricLogger JSONL dump,
glog formatter (reference dinov3_jax/logging/)."""

import json
import logging

import torch

from dinov3_amd.logging import setup_logging
from dinov3_amd.logging.helpers import MetricLogger, SmoothedValue


def test_smoothed_value_stats():
    v = SmoothedValue(window_size=3)
    for x in (1.0, 2.0, 3.0, 4.0):
        v.update(x)
    assert v.value == 4.0
    assert v.max == 4.0
    assert abs(v.avg - 3.0) < 1e-9          # window (2,3,4)
    assert abs(v.median - 3.0) < 1e-9
    assert abs(v.global_avg - 2.5) < 1e-9   # all four
    v.update(10.0, num=6)
    assert abs(v.global_avg - (10.0 + 60.0) / 10) < 1e-9
    assert "(" in str(v)


def test_metric_logger_update_and_jsonl(tmp_path):
    out = tmp_path / "metrics.json"
    ml = MetricLogger(output_file=str(out))
    for i in range(12):
        ml.update(loss=float(i), lr=torch.tensor(0.1))
    header = "T"
    lines = list(ml.log_every(range(12), print_freq=5, header=header, n_iterations=12))
    assert len(lines) == 12
    assert abs(ml.loss.global_avg - sum(range(12)) / 12 / 2) < 20  # meters updated twice overall
    if out.exists():
        rows = [json.loads(l) for l in out.read_text().splitlines() if l.strip()]
        assert rows, "JSONL dump empty"
        assert "iteration" in rows[0]


def test_setup_logging_rank_files(tmp_path):
    setup_logging(output=str(tmp_path / "logs"), name="dinov3_test_logger")
    lg = logging.getLogger("dinov3_test_logger")
    lg.info("hello from test")
    for h in lg.handlers:
        try:
            h.flush()
        except Exception:
            pass
    import os

    files = os.listdir(tmp_path / "logs") if (tmp_path / "logs").exists() else []
    assert files, "no log files written"
