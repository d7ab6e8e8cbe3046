"""Host-pinned async data pipe (BASELINE config #5).

The reference streams datasets from EFS through the framework's decode
pipeline; on one MI355X node the analogue is: a background producer fills
host-pinned staging buffers, and H2D copies run on a dedicated HIP stream
overlapped with compute — the 288 GB HBM sizing means the device-side
queue can be deep (several batches in flight).
"""

from __future__ import annotations

import queue
import threading
from typing import Iterable, Iterator

import torch


class HostPinnedPrefetcher:
    """Wraps a host-batch source; yields device batches copied async.

    * producer thread: pulls CPU batches, pins them, queues them;
    * consumer (training loop): dequeues, issues ``.to(device,
      non_blocking=True)`` on a side stream, and hands the training
      stream an event-synchronised device batch one step ahead.
    On CPU (tests) it degrades to a plain pass-through iterator.
    """

    def __init__(self, source: Iterable, device: torch.device,
                 depth: int = 3):
        self.source = iter(source)
        self.device = device
        self.depth = depth
        self.cuda = (device.type == "cuda") and torch.cuda.is_available()
        self._q: "queue.Queue" = queue.Queue(maxsize=depth)
        self._stop = threading.Event()
        if self.cuda:
            self.copy_stream = torch.cuda.Stream()
            self._thread = threading.Thread(target=self._produce, daemon=True)
            self._thread.start()

    def _pin(self, batch):
        return tuple(t.pin_memory() if isinstance(t, torch.Tensor)
                     and not t.is_cuda and not t.is_pinned() else t
                     for t in batch)

    def _produce(self) -> None:
        try:
            while not self._stop.is_set():
                batch = self._pin(next(self.source))
                # bounded put so close() can always stop the thread (a
                # daemon thread killed mid-HIP-call aborts the process
                # at interpreter teardown)
                while not self._stop.is_set():
                    try:
                        self._q.put(batch, timeout=0.1)
                        break
                    except queue.Full:
                        continue
        except StopIteration:
            self._q.put(None)

    def __iter__(self) -> Iterator:
        if not self.cuda:
            yield from self.source
            return
        next_batch = self._stage(self._q.get())
        while next_batch is not None:
            cur, ev = next_batch
            staged = self._q.get()
            next_batch = self._stage(staged) if staged is not None else None
            torch.cuda.current_stream().wait_event(ev)
            yield cur

    def _stage(self, host_batch):
        if host_batch is None:
            return None
        with torch.cuda.stream(self.copy_stream):
            dev = tuple(t.to(self.device, non_blocking=True)
                        if isinstance(t, torch.Tensor) else t
                        for t in host_batch)
            ev = torch.cuda.Event()
            ev.record(self.copy_stream)
        return dev, ev

    def close(self) -> None:
        self._stop.set()
        while not self._q.empty():
            try:
                self._q.get_nowait()
            except queue.Empty:
                break
        if self.cuda and self._thread.is_alive():
            self._thread.join(timeout=2.0)
