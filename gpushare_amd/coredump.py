"""All-thread stack dumps (reference: pkg/gpu/nvidia/coredump.go).

The reference dumps all goroutine stacks to ``/etc/kubernetes/go_<ts>.txt``
on SIGQUIT (gpumanager.go:97-101).  Python equivalent: every thread's
traceback via ``sys._current_frames``."""

from __future__ import annotations

import os
import sys
import threading
import time
import traceback


def stack_trace_all() -> str:
    frames = sys._current_frames()
    names = {t.ident: t.name for t in threading.enumerate()}
    out = []
    for ident, frame in frames.items():
        out.append(f"--- thread {names.get(ident, '?')} ({ident}) ---")
        out.extend(traceback.format_stack(frame))
    return "\n".join(out)


def coredump(directory: str = "/etc/kubernetes") -> str:
    """Write all-thread stacks; returns the file path ('' on failure)."""
    path = os.path.join(directory, f"py_stacks_{int(time.time())}.txt")
    try:
        os.makedirs(directory, exist_ok=True)
        with open(path, "w") as f:
            f.write(stack_trace_all())
        return path
    except OSError:
        sys.stderr.write(stack_trace_all() + "\n")
        return ""
