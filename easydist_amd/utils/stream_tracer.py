"""Kernel -> HIP-stream attribution via the torch profiler (kineto over
roctracer on ROCm).

Capability parity with the reference's CUPTI-side stream knowledge for
memory planning: the static memory plan records one stream id per
ALLOCATION (memory/csrc/profiling_allocator.cpp); this tracer recovers
which stream each KERNEL actually ran on, so lifetime analysis
(schedule/lifetime.py, which otherwise assumes single-stream order) can
be cross-checked against real execution.
"""
from __future__ import annotations

from collections import defaultdict
from typing import Callable, Dict, Set

import torch


def trace_kernel_streams(fn: Callable, *args,
                         **kwargs) -> Dict[str, Set[int]]:
    """Run ``fn`` under the profiler; return {op_name: {stream ids}} for
    every op that launched device kernels.

    On a GPU box the kineto device events carry the real HIP stream of
    each kernel launch; on CPU the map is empty (no device events)."""
    activities = [torch.profiler.ProfilerActivity.CPU]
    if torch.cuda.is_available():
        activities.append(torch.profiler.ProfilerActivity.CUDA)
    with torch.profiler.profile(activities=activities) as prof:
        fn(*args, **kwargs)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
    out: Dict[str, Set[int]] = defaultdict(set)
    try:
        raw = prof.profiler.kineto_results.events()
    except Exception:       # profiler internals moved — fall back below
        raw = []
    for ev in raw:
        try:
            if ev.device_type() == torch.autograd.DeviceType.CUDA:
                # kineto reports the HIP stream as the device resource row
                out[ev.name()].add(int(ev.device_resource_id()))
        except Exception:
            continue
    return dict(out)


def multi_stream_kernels(fn: Callable, *args, **kwargs) -> Set[str]:
    """Kernel names observed on MORE than one stream across the traced
    call — the set lifetime analysis must treat as cross-stream
    hazards."""
    m = trace_kernel_streams(fn, *args, **kwargs)
    return {name for name, streams in m.items() if len(streams) > 1}


def streams_used(fn: Callable, *args, **kwargs) -> Set[int]:
    """All device streams that executed at least one kernel."""
    m = trace_kernel_streams(fn, *args, **kwargs)
    return set().union(*m.values()) if m else set()
