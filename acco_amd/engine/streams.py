"""Thin HIP-stream/event abstraction that degrades to no-ops on CPU,
so the ACCO state machine runs identically under gloo tests."""

from __future__ import annotations

import contextlib

import torch


class Stream:
    def __init__(self, device: torch.device):
        self.cuda = device.type == "cuda"
        self._s = torch.cuda.Stream(device=device) if self.cuda else None

    def activate(self):
        if self.cuda:
            return torch.cuda.stream(self._s)
        return contextlib.nullcontext()

    def wait_default(self, device):
        if self.cuda:
            self._s.wait_stream(torch.cuda.default_stream(device))

    @property
    def torch_stream(self):
        return self._s


class Event:
    """A blocking event: record on the active stream, then `sync()` blocks
    the CPU until the GPU reaches it (reference trainer_decoupled.py:218-219
    uses torch.cuda.Event(blocking=True) the same way)."""

    def __init__(self, device: torch.device):
        self.cuda = device.type == "cuda"
        self._e = torch.cuda.Event(blocking=True) if self.cuda else None

    def record_and_sync(self, stream: Stream | None = None):
        if self.cuda:
            if stream is not None and stream.torch_stream is not None:
                self._e.record(stream.torch_stream)
            else:
                self._e.record()
            self._e.synchronize()
