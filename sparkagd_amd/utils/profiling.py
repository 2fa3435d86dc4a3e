"""Optional torch.profiler integration (SURVEY.md §5 'Tracing/profiling').

Framework-level tracing for users who want a Chrome/Perfetto timeline of the
optimizer loop; kernel-level analysis belongs to rocprofv3 (see profiles/).

Usage:
    from sparkagd_amd.utils.profiling import profile_run
    with profile_run("trace.json") as prof:
        run(shard, ...)
"""

from __future__ import annotations

import contextlib
from typing import Iterator, Optional

import torch


@contextlib.contextmanager
def profile_run(
    trace_path: Optional[str] = None,
    with_stack: bool = False,
    record_shapes: bool = False,
) -> Iterator[object]:
    """Context manager wrapping torch.profiler over CPU + device activity;
    exports a Chrome trace to ``trace_path`` on exit when given."""
    activities = [torch.profiler.ProfilerActivity.CPU]
    if torch.cuda.is_available():
        activities.append(torch.profiler.ProfilerActivity.CUDA)
    with torch.profiler.profile(
        activities=activities,
        with_stack=with_stack,
        record_shapes=record_shapes,
    ) as prof:
        yield prof
    if trace_path is not None:
        prof.export_chrome_trace(trace_path)
