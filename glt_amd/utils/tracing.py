"""rocprof-friendly range annotations.

The reference has no built-in tracing; glt_amd annotates its
hot phases with roctx-style ranges so `rocprofv3 --marker-trace` attributes
kernel time to pipeline stages.  Falls back to no-ops when the ROCm tracer
is unavailable (CPU-only boxes).
"""
import os
from contextlib import contextmanager

_enabled = os.environ.get("GLT_TRACE", "1") != "0"
_push = _pop = None

if _enabled:
    try:
        import torch

        if torch.cuda.is_available():
            _push = torch.cuda.nvtx.range_push  # roctx under ROCm
            _pop = torch.cuda.nvtx.range_pop
    except Exception:
        pass


def range_push(name: str):
    if _push is not None:
        try:
            _push(name)
        except Exception:
            pass


def range_pop():
    if _pop is not None:
        try:
            _pop()
        except Exception:
            pass


@contextmanager
def trace_region(name: str):
    range_push(name)
    try:
        yield
    finally:
        range_pop()
