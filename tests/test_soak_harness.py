"""The soak harness itself stays green: a short CPU churn run must complete
with ok=true (this is the tool behind profiles/soak*_gpu*.json)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_soak_short_cpu():
    proc = subprocess.run(
        [sys.executable, os.path.join(REPO, "scripts", "soak.py"),
         "--duration", "8", "--cpu", "--max-active", "4"],
        capture_output=True, text=True, timeout=120, cwd=REPO,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    out = json.loads(proc.stdout.strip().splitlines()[-1])
    assert out["ok"] is True
    assert out["timeouts"] == 0
    assert out["pods_done"] >= 4
    assert out["leaks"]["reservations"] == 0
