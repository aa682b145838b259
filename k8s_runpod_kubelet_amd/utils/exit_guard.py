"""ROCm-safe process-exit guard.

ROCm's C++ static destructors can ``std::terminate`` at interpreter exit (a
joinable library thread torn down out of order), turning a fully green GPU
run into SIGABRT/rc=134 — observed intermittently on MI355X boxes *after*
the real work succeeded. The fix is to ``os._exit`` before those destructors
run — but a bare ``atexit.register(os._exit, status)`` also skips every
atexit hook registered *before* ours (atexit is LIFO), including any
instrumentation the host harness installed at process start, and drops
unflushed stdio.

``install(status)`` registers a guard that, when the interpreter begins
normal atexit processing:

1. flushes stdout/stderr,
2. drains the remaining atexit queue via ``atexit._run_exitfuncs()`` so
   earlier-registered hooks (harness instrumentation, coverage dumps, log
   handlers) still run — a reentrancy flag keeps our own callback from
   recursing when the drain reaches it again,
3. ``os._exit(status)`` before ROCm's static destructors get a chance.
"""

from __future__ import annotations

import atexit
import os
import sys

_installed = False
_entered = False


def install(status: int) -> None:
    """Install the exit guard once; later calls update nothing (first wins,
    matching 'the outcome decided when the guard was armed')."""
    global _installed
    if _installed:
        return
    _installed = True
    atexit.register(_guard, int(status))


def _guard(status: int) -> None:
    global _entered
    if _entered:  # reached again via our own _run_exitfuncs drain
        return
    _entered = True
    try:
        sys.stdout.flush()
        sys.stderr.flush()
    except Exception:
        pass
    try:
        # Run every atexit hook still queued (registered before this guard),
        # so harness-side instrumentation is not defeated by the early exit.
        atexit._run_exitfuncs()
    except Exception:
        pass
    try:
        sys.stdout.flush()
        sys.stderr.flush()
    except Exception:
        pass
    os._exit(status)
