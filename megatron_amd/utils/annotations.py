"""Profiler range annotations (rocTX / torch.profiler).

Capability analog of the reference's NVTX helpers (core/utils.py nvtx
ranges, training.py configure_nvtx_profiling): named ranges that show up in
rocprofv3 (--att / roctx) traces and in torch.profiler timelines.  Uses
torch.profiler's record_function (which emits roctx markers on ROCm when a
profiler is attached) so the same annotations serve both tools; no-ops with
zero overhead when disabled."""

from __future__ import annotations

import contextlib
import functools

import torch

_enabled = False


def enable_annotations(on: bool = True):
    global _enabled
    _enabled = on


@contextlib.contextmanager
def profile_range(name: str):
    if not _enabled:
        yield
        return
    with torch.profiler.record_function(name):
        yield


def annotated(name: str = None):
    """Decorator form: @annotated("fwd:attention")."""

    def wrap(fn):
        label = name or fn.__qualname__

        @functools.wraps(fn)
        def inner(*args, **kwargs):
            if not _enabled:
                return fn(*args, **kwargs)
            with torch.profiler.record_function(label):
                return fn(*args, **kwargs)

        return inner

    return wrap
