"""Job launch helpers: SLURM submission (`submit.py`) and the `job_context`
context manager (reference dinov3_jax/run/init.py:19-37 — setup on entry,
teardown on exit)."""

from __future__ import annotations

import contextlib
from typing import Optional


@contextlib.contextmanager
def job_context(output_dir: Optional[str] = None, distributed_enabled: bool = True,
                logging_enabled: bool = True, seed: int = 0):
    from ..configs import setup_job
    from .. import parallel

    setup_job(output_dir=output_dir, seed=seed,
              distributed_enabled=distributed_enabled, logging_enabled=logging_enabled)
    try:
        yield
    finally:
        if distributed_enabled:
            parallel.destroy()
