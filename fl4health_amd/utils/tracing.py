"""rocTX/NVTX range annotations around the hot phases (the
reference has no profiler hooks; here every K1-K18 phase is visible in
rocprofv3 --marker-trace / --sys-trace timelines)."""
from __future__ import annotations

from contextlib import contextmanager

import torch

_ENABLED = torch.cuda.is_available()


@contextmanager
def trace_range(name: str):
    if _ENABLED:
        torch.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield


def trace_mark(name: str) -> None:
    if _ENABLED:
        torch.cuda.nvtx.mark(name)
