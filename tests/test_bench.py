"""bench.py contract tests (CPU mode; the driver runs the GPU variant)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_bench(args, timeout=240):
    proc = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py")] + args,
        capture_output=True, text=True, timeout=timeout, cwd=REPO,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    line = proc.stdout.strip().splitlines()[-1]
    return json.loads(line)


def check_contract(out, n_gpus, steps, warmup):
    assert out["metric"] == "gpu_pods_scheduled_per_sec"
    assert "p50 pod-Ready latency" in out["baseline_metric"]
    assert out["unit"] == "pods/s"
    assert out["n_gpus"] == n_gpus
    assert out["steps"] == steps
    assert out["warmup"] == warmup
    assert out["higher_is_better"] is True
    assert out["scaling"] == "weak"
    assert out["data"] == "synthetic"
    assert out["value"] > 0
    assert out["ms_per_step"] > 0
    assert out["p50_pod_ready_ms"] > 0
    assert out["config"]["pods_total"] == n_gpus * steps


def test_bench_single_gpu_cpu_mode():
    out = run_bench(["--gpus", "1", "--steps", "3", "--warmup", "1",
                     "--mode", "cpu"])
    check_contract(out, 1, 3, 1)


def test_bench_multi_gpu_cpu_mode():
    out = run_bench(["--gpus", "4", "--steps", "2", "--warmup", "0",
                     "--mode", "cpu"])
    check_contract(out, 4, 2, 0)


def test_bench_under_torchrun_world2():
    # The driver launches N>1 via torch.distributed.run; verify that shape
    # works on CPU (gloo, 127.0.0.1 rendezvous).
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29517", "bench.py",
         "--gpus", "2", "--steps", "2", "--warmup", "0", "--mode", "cpu"],
        capture_output=True, text=True, timeout=300, cwd=REPO,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    lines = [l for l in proc.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1  # only rank 0 prints
    out = json.loads(lines[0])
    check_contract(out, 2, 2, 0)
