"""CLI demo-mode smoke: the shipped binary boots a self-contained stack
(fake apiserver + synthetic GPUs), serves a pod over plain HTTP, and shuts
down cleanly on SIGTERM (reference main() shutdown path, main.go:344-350)."""

import json
import re
import signal
import subprocess
import sys
import tempfile
import time
import urllib.request

REPO = __file__.rsplit("/tests/", 1)[0]


def test_cli_fake_apiserver_demo():
    # isolated state dir: the CLI default (/var/lib/amd-virtual-kubelet)
    # accumulates adopted instances across runs (kubelet semantics) —
    # at ~100 leftovers the adoption/import work ate this test's startup
    # budget
    state = tempfile.mkdtemp(prefix="amdvk-clidemo-")
    p = subprocess.Popen(
        [sys.executable, "-m", "k8s_runpod_kubelet_amd.cli",
         "--gpu-count-override", "8", "--fake-apiserver",
         "--listen-port", "0", "--state-dir", state,
         "--health-server-address", "127.0.0.1:0"],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True, cwd=REPO,
    )
    try:
        url = None
        t0 = time.time()
        while time.time() - t0 < 30:
            line = p.stdout.readline()
            m = re.search(r"fake apiserver at (http://\S+)", line or "")
            if m:
                url = m.group(1)
                break
        assert url, "no apiserver url printed"
        time.sleep(1.5)
        pod = {"apiVersion": "v1", "kind": "Pod",
               "metadata": {"name": "demo", "namespace": "default"},
               "spec": {"nodeName": "virtual-runpod",
                        "containers": [{"name": "main", "image": "x",
                                        "command": ["podworker"],
                                        "args": ["--hold"]}]}}
        req = urllib.request.Request(
            f"{url}/api/v1/namespaces/default/pods",
            data=json.dumps(pod).encode(), method="POST",
            headers={"Content-Type": "application/json"})
        urllib.request.urlopen(req, timeout=10)
        deadline = time.time() + 20
        ready = False
        while time.time() < deadline:
            with urllib.request.urlopen(
                    f"{url}/api/v1/namespaces/default/pods/demo",
                    timeout=5) as r:
                obj = json.load(r)
            conds = {c["type"]: c["status"]
                     for c in obj.get("status", {}).get("conditions", [])}
            if conds.get("Ready") == "True":
                ready = True
                break
            time.sleep(0.1)
        assert ready, "demo pod never went Ready through the CLI stack"
    finally:
        p.send_signal(signal.SIGTERM)
        rc = p.wait(timeout=20)
    assert rc == 0  # clean shutdown on SIGTERM
