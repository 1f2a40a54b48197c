"""Opt-in roctx range markers for rocprof correlation (the
reference's convention is NVTX ranges in its harnesses, e.g.
tests/L1/common/main_amp.py:447-450; on ROCm ``torch.cuda.nvtx`` maps to
roctx and the ranges appear in ``rocprofv3 --sys-trace`` / marker traces).

Disabled by default — set ``APEX_TRACE=1`` to activate; with the knob unset
every helper is a no-op so hot paths (and hipGraph capture) see zero
overhead. roctx calls are host-side only: they annotate eager launches and
are NOT recorded into captured graphs (replays show the kernels, not the
ranges — trace the eager warmup iterations instead)."""

import contextlib
import os

import torch

_enabled_cache = None


def trace_enabled():
    global _enabled_cache
    if _enabled_cache is None:
        _enabled_cache = bool(os.environ.get("APEX_TRACE")) and torch.cuda.is_available()
    return _enabled_cache


@contextlib.contextmanager
def trace_range(name):
    """roctx push/pop bracket; no-op unless APEX_TRACE=1 and CUDA is up."""
    if trace_enabled():
        torch.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield


def trace_mark(name):
    """Instant roctx marker; same gating as trace_range."""
    if trace_enabled():
        torch.cuda.nvtx.mark(name)


def traced(name):
    """Decorator form of trace_range for method seams (optimizer steps)."""
    def deco(fn):
        import functools

        @functools.wraps(fn)
        def wrapper(*args, **kwargs):
            if not trace_enabled():
                return fn(*args, **kwargs)
            torch.cuda.nvtx.range_push(name)
            try:
                return fn(*args, **kwargs)
            finally:
                torch.cuda.nvtx.range_pop()
        return wrapper
    return deco
