"""Lightweight phase tracing (SURVEY.md §5 tracing row).

Emits rocTX-style ranges via torch.cuda.nvtx (which maps onto the ROCm
marker API), visible in `rocprofv3 --marker-trace`.  No-ops on CPU or when
tracing is off (RLR_AMD_TRACE=0)."""

import contextlib
import os

import torch

_ENABLED = os.environ.get('RLR_AMD_TRACE', '0') == '1' \
    and torch.cuda.is_available()


@contextlib.contextmanager
def trace_range(name: str):
    if _ENABLED:
        torch.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield
