"""Side-stream batch prefetcher: overlaps the next batch's H2D copies (and
the host wait on the DataLoader) with the current step's compute.

The trainer's loop otherwise enqueues the pinned-memory copies on the default
stream between the optimizer step and the next forward — serialized with the
step-end param all-gather and the EMA update. Staging them on a dedicated
copy stream turns the H2D time into overlap (SURVEY §2.3 note on C1/C2
stream separation; the reference has no equivalent — JAX device_put blocks).
"""

from __future__ import annotations

from typing import Dict, Iterator, Optional

import torch


class CudaBatchPrefetcher:
    """Wraps an iterator of CPU (or device) batch dicts. `__next__` returns
    the pre-staged batch after a stream-event wait (no host sync) and kicks
    off the following batch's copies on the side stream."""

    def __init__(self, it: Iterator[Dict], device: torch.device):
        assert device.type == "cuda", "CudaBatchPrefetcher needs a CUDA device"
        self.it = it
        self.device = device
        self.stream = torch.cuda.Stream(device)
        self._next: Optional[Dict] = None
        self._event: Optional[torch.cuda.Event] = None
        self._preload()

    def _preload(self) -> None:
        try:
            cpu_batch = next(self.it)
        except StopIteration:
            self._next = None
            return
        with torch.cuda.stream(self.stream):
            self._next = {
                k: (v.to(self.device, non_blocking=True) if isinstance(v, torch.Tensor) else v)
                for k, v in cpu_batch.items()
            }
        self._event = torch.cuda.Event()
        self._event.record(self.stream)

    def __iter__(self):
        return self

    def __next__(self) -> Dict:
        if self._next is None:
            raise StopIteration
        torch.cuda.current_stream(self.device).wait_event(self._event)
        batch = self._next
        # the consumer stream now owns the tensors; keep the caching
        # allocator from recycling them while the copy stream still holds them
        for v in batch.values():
            if isinstance(v, torch.Tensor):
                v.record_stream(torch.cuda.current_stream(self.device))
        self._preload()
        return batch
