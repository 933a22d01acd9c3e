"""Driver-contract rehearsal: `bench.py --gpus 8` launched exactly the way the
driver launches it (torch.distributed.run, one rank per device) — on CPU/gloo
here, so the only untested piece at scale time is RCCL itself."""

import json
import os
import socket
import subprocess
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED_KEYS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.mark.timeout(540)
def test_bench_torchrun_cpu_world8():
    cmd = [
        sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
        "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
        "--master-port", str(_free_port()),
        "bench.py", "--gpus", "8", "--steps", "1", "--warmup", "0",
        "--batch-size", "2", "--arch", "vit_small", "--local-crops", "2",
        "--global-size", "64", "--local-size", "32",
    ]
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    proc = subprocess.run(cmd, cwd=REPO_ROOT, env=env, capture_output=True,
                          text=True, timeout=520)
    assert proc.returncode == 0, f"stdout:\n{proc.stdout[-2000:]}\nstderr:\n{proc.stderr[-3000:]}"
    json_lines = [ln for ln in proc.stdout.splitlines() if ln.startswith("{")]
    assert len(json_lines) == 1, f"expected exactly one JSON line from rank 0:\n{proc.stdout[-2000:]}"
    result = json.loads(json_lines[0])
    assert REQUIRED_KEYS <= set(result), REQUIRED_KEYS - set(result)
    assert result["n_gpus"] == 8
    assert result["value"] > 0
    assert result["config"]["global_batch"] == 16
    # world > 1 with the default SHARD_GRAD_OP strategy engages ShardedEngine
    assert result["config"]["parallelism"] == "fsdp8"
