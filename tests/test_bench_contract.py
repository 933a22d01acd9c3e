"""Driver-contract test: `python bench.py` emits ONE JSON line from rank 0
with the exact field set and semantics the harness depends on."""

import json
import subprocess
import sys


def test_bench_json_contract():
    r = subprocess.run(
        [sys.executable, "bench.py", "--steps", "1", "--warmup", "0",
         "--batch-size", "2", "--arch", "vit_small", "--local-crops", "2",
         "--global-size", "64", "--local-size", "32"],
        capture_output=True, text=True, timeout=900,
    )
    assert r.returncode == 0, r.stderr[-800:]
    json_lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1, f"expected exactly one JSON line, got {json_lines}"
    out = json.loads(json_lines[0])
    for key, typ in [("metric", str), ("value", float), ("unit", str),
                     ("n_gpus", int), ("steps", int), ("warmup", int),
                     ("ms_per_step", float), ("higher_is_better", bool),
                     ("scaling", str), ("dtype", str), ("data", str),
                     ("config", dict)]:
        assert key in out, f"missing {key}"
        assert isinstance(out[key], typ), f"{key}: {type(out[key])}"
    assert out["metric"].startswith("images/sec")
    assert out["higher_is_better"] is True
    assert out["scaling"] == "weak"
    assert out["data"] == "synthetic"
    assert out["n_gpus"] == 1 and out["steps"] == 1 and out["warmup"] == 0
    assert out["value"] > 0 and out["ms_per_step"] > 0
    assert out["vs_baseline"] is None or out["vs_baseline"] > 0
    assert {"model", "global_batch", "parallelism"} <= set(out["config"])
    # whole-job value consistency: value == global_batch * steps / elapsed
    expect = out["config"]["global_batch"] * 1000.0 / out["ms_per_step"]
    assert abs(out["value"] - expect) / expect < 1e-6
