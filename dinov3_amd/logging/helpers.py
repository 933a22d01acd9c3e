"""Training meters: SmoothedValue + MetricLogger with JSONL dump and ETA.

Parity with reference dinov3_jax/logging/helpers.py:24-197.
"""

from __future__ import annotations

import datetime
import json
import logging
import time
from collections import defaultdict, deque
from typing import Iterable, Optional

import torch

logger = logging.getLogger("dinov3")


class SmoothedValue:
    """Track a series of values; windowed median/avg + global avg.

    `synchronize_between_processes` all-reduces (count, total) — the RCCL/gloo
    equivalent of the reference's psum meter sync (helpers.py:43).
    """

    def __init__(self, window_size: int = 20, fmt: str = "{median:.4f} ({global_avg:.4f})"):
        self.deque: deque = deque(maxlen=window_size)
        self.total = 0.0
        self.count = 0
        self.fmt = fmt

    def update(self, value: float, num: int = 1) -> None:
        self.deque.append(value)
        self.count += num
        self.total += value * num

    def synchronize_between_processes(self) -> None:
        import torch.distributed as dist

        if not (dist.is_available() and dist.is_initialized()):
            return
        from .. import parallel

        # meters are per-student under multi-distillation: sync the subgroup
        t = torch.tensor([self.count, self.total], dtype=torch.float64)
        dist.all_reduce(t, group=parallel.subgroup())
        self.count = int(t[0].item())
        self.total = t[1].item()

    @property
    def median(self) -> float:
        return float(torch.tensor(list(self.deque)).median()) if self.deque else 0.0

    @property
    def avg(self) -> float:
        return float(torch.tensor(list(self.deque), dtype=torch.float64).mean()) if self.deque else 0.0

    @property
    def global_avg(self) -> float:
        return self.total / max(self.count, 1)

    @property
    def max(self) -> float:
        return max(self.deque) if self.deque else 0.0

    @property
    def value(self) -> float:
        return self.deque[-1] if self.deque else 0.0

    def __str__(self) -> str:
        return self.fmt.format(
            median=self.median, avg=self.avg, global_avg=self.global_avg, max=self.max, value=self.value
        )


class MetricLogger:
    def __init__(self, delimiter: str = "  ", output_file: Optional[str] = None):
        self.meters: defaultdict = defaultdict(SmoothedValue)
        self.delimiter = delimiter
        self.output_file = output_file

    def update(self, **kwargs) -> None:
        for k, v in kwargs.items():
            if isinstance(v, torch.Tensor):
                v = v.item()
            self.meters[k].update(float(v))

    def __getattr__(self, attr):
        if attr in self.meters:
            return self.meters[attr]
        raise AttributeError(attr)

    def __str__(self) -> str:
        return self.delimiter.join(f"{name}: {meter}" for name, meter in self.meters.items())

    def synchronize_between_processes(self) -> None:
        for meter in self.meters.values():
            meter.synchronize_between_processes()

    def add_meter(self, name: str, meter: SmoothedValue) -> None:
        self.meters[name] = meter

    def dump_in_output_file(self, iteration: int, iter_time: float, data_time: float) -> None:
        if self.output_file is None:
            from .. import parallel

            return
        from .. import parallel

        if not parallel.is_main_process():
            return
        entry = {"iteration": iteration, "iter_time": iter_time, "data_time": data_time}
        entry.update({k: v.median for k, v in self.meters.items()})
        with open(self.output_file, "a") as f:
            f.write(json.dumps(entry) + "\n")

    def log_every(self, iterable: Iterable, print_freq: int, header: str = "", n_iterations: Optional[int] = None,
                  start_iteration: int = 0):
        i = start_iteration
        if n_iterations is None:
            try:
                n_iterations = len(iterable)  # type: ignore[arg-type]
            except TypeError:
                n_iterations = None
        iter_time = SmoothedValue(fmt="{avg:.4f}")
        data_time = SmoothedValue(fmt="{avg:.4f}")
        start = time.time()
        end = time.time()
        for obj in iterable:
            data_time.update(time.time() - end)
            yield obj
            iter_time.update(time.time() - end)
            if i % print_freq == 0 or (n_iterations is not None and i == n_iterations - 1):
                self.dump_in_output_file(iteration=i, iter_time=iter_time.avg, data_time=data_time.avg)
                if n_iterations is not None:
                    eta = iter_time.global_avg * (n_iterations - i)
                    eta_str = str(datetime.timedelta(seconds=int(eta)))
                    logger.info(
                        "%s [%d/%d] eta: %s %s time: %s data: %s",
                        header, i, n_iterations, eta_str, str(self), str(iter_time), str(data_time),
                    )
                else:
                    logger.info("%s [%d] %s time: %s data: %s", header, i, str(self), str(iter_time), str(data_time))
            i += 1
            end = time.time()
            if n_iterations is not None and i >= n_iterations:
                break
        total_time = time.time() - start
        logger.info("%s Total time: %s", header, str(datetime.timedelta(seconds=int(total_time))))
