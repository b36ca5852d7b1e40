"""Tracing hooks: rocTX ranges around the hot phases (SURVEY.md §5 —
the reference has no tracing; rocprofv3 --marker-trace picks these up).

Falls back to no-ops when the roctx module is unavailable (CPU CI)."""

from __future__ import annotations

import contextlib
import os

_roctx = None
if os.environ.get("ACCO_ROCTX", "1") != "0":
    try:
        from torch.cuda import nvtx as _roctx  # maps to roctx on ROCm builds
    except Exception:
        _roctx = None


@contextlib.contextmanager
def trace_range(name: str):
    if _roctx is None:
        yield
        return
    try:
        _roctx.range_push(name)
        yield
    finally:
        _roctx.range_pop()


def mark(name: str) -> None:
    if _roctx is not None:
        try:
            _roctx.mark(name)
        except Exception:
            pass
