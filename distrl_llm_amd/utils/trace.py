"""rocprof-compatible phase ranges (SURVEY.md §5.1 build note).

``torch.cuda.nvtx`` maps to roctx on ROCm builds, so these ranges show up
in ``rocprofv3 --marker-trace`` (and rocprof GUI timelines) and let a
profile be sliced by RL phase: generate / reward / advantage / update /
sync_weights / eval, plus engine-internal prefill and decode-wave spans.

No-ops on CPU-only hosts. Never used inside a hipGraph capture region
(markers are host-side; the captured step body stays marker-free).
"""

from __future__ import annotations

from contextlib import contextmanager

import torch

_ENABLED = None


def _enabled() -> bool:
    """Markers are pure observability — if roctx is unavailable in this
    torch build they self-disable rather than fail the compute path."""
    global _ENABLED
    if _ENABLED is None:
        if not torch.cuda.is_available():
            _ENABLED = False
        else:
            try:
                torch.cuda.nvtx.range_push("distrl/trace_probe")
                torch.cuda.nvtx.range_pop()
                _ENABLED = True
            except Exception:
                _ENABLED = False
    return _ENABLED


@contextmanager
def trace_range(name: str):
    """Context manager emitting a roctx range around the enclosed work."""
    if _enabled():
        try:
            torch.cuda.nvtx.range_push(name)
        except Exception:
            yield
            return
        try:
            yield
        finally:
            try:
                torch.cuda.nvtx.range_pop()
            except Exception:
                pass
    else:
        yield


def trace_mark(name: str) -> None:
    """Instantaneous roctx marker."""
    if _enabled():
        try:
            torch.cuda.nvtx.mark(name)
        except Exception:
            pass
