"""utils/exit_guard.py: the ROCm-destructor exit guard must (a) exit with
the armed status, (b) still run atexit hooks registered *before* it (e.g.
harness instrumentation installed at process start), and (c) flush stdio.
Round-1 regression: a bare `atexit.register(os._exit, 0)` dropped both the
unflushed "smoke ok" line and the driver's native-so capture hook."""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

SCRIPT = r"""
import atexit, os, sys
sys.path.insert(0, {repo!r})

marker = sys.argv[1]

# Simulates harness instrumentation registered at process start — i.e.
# BEFORE the guard. With a bare os._exit atexit hook this never runs.
def harness_hook():
    with open(marker, "w") as fh:
        fh.write("hook-ran")
atexit.register(harness_hook)

from k8s_runpod_kubelet_amd.utils.exit_guard import install
print("work done", end="")  # deliberately unflushed, block-buffered via pipe
install(7)
# second install must not override the first status
install(0)
"""


def _run(tmp_path):
    marker = str(tmp_path / "hook.txt")
    proc = subprocess.run(
        [sys.executable, "-c", SCRIPT.format(repo=REPO), marker],
        capture_output=True, text=True, timeout=60,
    )
    return proc, marker


def test_exit_guard_status_hooks_and_flush(tmp_path):
    proc, marker = _run(tmp_path)
    assert proc.returncode == 7, proc.stderr
    # earlier-registered hook ran despite the _exit
    assert os.path.exists(marker)
    assert open(marker).read() == "hook-ran"
    # unflushed stdout made it out
    assert "work done" in proc.stdout
