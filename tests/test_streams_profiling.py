"""CPU degradation contracts of the stream/event abstraction and the
rocTX tracing hooks — these must be exact no-ops off-GPU so the ACCO
state machine and trainers run identically under gloo tests."""

import subprocess
import sys

import torch

from acco_amd.engine.streams import Event, Stream
from acco_amd.utils import profiling


def test_stream_cpu_noops():
    s = Stream(torch.device("cpu"))
    assert s.torch_stream is None
    with s.activate():           # nullcontext
        x = torch.ones(3) * 2
    s.wait_default(torch.device("cpu"))
    assert x.sum().item() == 6.0


def test_event_cpu_noop():
    e = Event(torch.device("cpu"))
    e.record_and_sync()                     # no stream
    e.record_and_sync(Stream(torch.device("cpu")))  # CPU stream


def test_trace_range_and_mark_cpu():
    with profiling.trace_range("phase"):
        pass
    profiling.mark("point")


def test_roctx_disable_env():
    """ACCO_ROCTX=0 must leave the module with no roctx binding (the env
    is read at import, so probe in a subprocess)."""
    code = ("import os; os.environ['ACCO_ROCTX']='0'; "
            "from acco_amd.utils import profiling; "
            "assert profiling._roctx is None; "
            "ctx = profiling.trace_range('x'); ctx.__enter__(); "
            "ctx.__exit__(None, None, None); print('ok')")
    out = subprocess.run([sys.executable, "-c", code], capture_output=True,
                         text=True, timeout=120)
    assert out.returncode == 0 and "ok" in out.stdout, out.stderr
