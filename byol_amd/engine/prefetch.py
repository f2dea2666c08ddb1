"""Device prefetcher: H2D upload of the NEXT minibatch on a dedicated HIP
copy stream, overlapped with the current step's compute (SURVEY.md K19:
"pinned-memory + hipMemcpyAsync on copy stream").

The DataLoader's pinned host tensors are copied on ``self.stream``; the
compute stream waits on a recorded event before the batch is yielded, and
``record_stream`` tells the caching allocator the tensors are consumed on
the compute stream.  Pass-through when batches are already on device (the
GPU-augmentation iterable) or when CUDA is unavailable.
"""

from collections import deque
from typing import Iterable, Iterator, Tuple

import torch

__all__ = ["DevicePrefetcher"]


class DevicePrefetcher:
    def __init__(self, loader: Iterable, device=None):
        self.loader = loader
        self.device = device or torch.device("cuda")
        self.stream = torch.cuda.Stream(device=self.device)
        # pinned host batches stay referenced for two iterations so an
        # in-flight async H2D never reads freed host memory (belt and
        # braces on top of torch's host-allocator event tracking)
        self._keepalive = deque(maxlen=2)

    def __len__(self):
        return len(self.loader)

    def _upload(self, batch) -> Tuple[torch.Tensor, ...]:
        with torch.cuda.stream(self.stream):
            return tuple(
                t.to(self.device, non_blocking=True)
                if torch.is_tensor(t) else t
                for t in batch)

    def __iter__(self) -> Iterator:
        pending = None
        event = None
        for batch in self.loader:
            if torch.is_tensor(batch[0]) and batch[0].is_cuda:
                # already device-resident (GPU augmentation path)
                if pending is not None:
                    self._release(pending, event)
                    yield pending
                    pending, event = None, None
                yield batch
                continue
            moved = self._upload(batch)
            self._keepalive.append(batch)
            ev = torch.cuda.Event()
            ev.record(self.stream)
            if pending is not None:
                self._release(pending, event)
                yield pending
            pending, event = moved, ev
        if pending is not None:
            self._release(pending, event)
            yield pending

    @staticmethod
    def _release(batch, event):
        cur = torch.cuda.current_stream()
        cur.wait_event(event)
        for t in batch:
            if torch.is_tensor(t):
                t.record_stream(cur)
