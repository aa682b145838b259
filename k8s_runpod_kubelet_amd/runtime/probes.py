"""Container liveness/readiness probes (tcpSocket / httpGet / exec).

The reference has no probe support at all — its backend is a cloud API and
readiness is inferred from port mappings (kubelet.go:566-605). Local pods
make real kubelet probe semantics cheap:

- readinessProbe: governs the container's Ready state (overriding the
  AMDVK_READY_FD pipe signal once defined),
- livenessProbe: failureThreshold consecutive failures kill the container;
  spec.restartPolicy then decides whether it restarts (CrashLoopBackOff
  machinery applies).

Host-process pods share the host network namespace, so tcpSocket/httpGet
probe 127.0.0.1:<port> directly.
"""

from __future__ import annotations

import http.client
import logging
import socket
import subprocess
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

log = logging.getLogger("runtime.probes")


@dataclass
class ProbeSpec:
    kind: str                 # "tcp" | "http" | "exec"
    port: int = 0
    path: str = "/"
    command: List[str] = field(default_factory=list)
    initial_delay_s: float = 0.0
    period_s: float = 10.0
    timeout_s: float = 1.0
    failure_threshold: int = 3
    success_threshold: int = 1

    @staticmethod
    def parse(raw: Optional[Dict[str, Any]]) -> Optional["ProbeSpec"]:
        """Parse a k8s probe object; None when absent or handler unknown."""
        if not raw:
            return None
        def num(key, default, cast=float):
            try:
                return cast(raw.get(key, default))
            except (TypeError, ValueError):
                return default

        common = dict(
            initial_delay_s=num("initialDelaySeconds", 0.0),
            period_s=max(1.0, num("periodSeconds", 10.0)),
            timeout_s=max(0.1, num("timeoutSeconds", 1.0)),
            failure_threshold=max(1, num("failureThreshold", 3, int)),
            success_threshold=max(1, num("successThreshold", 1, int)),
        )
        if raw.get("tcpSocket"):
            try:
                port = int(raw["tcpSocket"].get("port", 0))
            except (TypeError, ValueError):
                return None
            return ProbeSpec(kind="tcp", port=port, **common)
        if raw.get("httpGet"):
            hg = raw["httpGet"]
            try:
                port = int(hg.get("port", 80))
            except (TypeError, ValueError):
                return None
            return ProbeSpec(kind="http", port=port,
                             path=hg.get("path", "/"), **common)
        if raw.get("exec"):
            cmd = list(raw["exec"].get("command", []) or [])
            if cmd:
                return ProbeSpec(kind="exec", command=cmd, **common)
        if raw.get("sleep"):
            try:
                secs = float(raw["sleep"].get("seconds", 0))
            except (TypeError, ValueError):
                return None
            # lifecycle-hook sleep handler (k8s 1.29+): modeled as a probe
            # whose "timeout" is the sleep duration
            return ProbeSpec(kind="sleep", timeout_s=max(0.0, secs),
                             **{k: v for k, v in common.items()
                                if k != "timeout_s"})
        return None

    @staticmethod
    def parse_hook(raw) -> Optional["ProbeSpec"]:
        """Parse a lifecycle handler (postStart/preStop): same handler
        shapes as probes plus sleep; hook executions default to a 30 s
        bound instead of the probe's 1 s."""
        spec = ProbeSpec.parse(raw)
        if spec is not None and spec.kind != "sleep" \
                and "timeoutSeconds" not in (raw or {}):
            spec.timeout_s = 30.0
        return spec


@dataclass
class ProbeState:
    """Per-(container, probe-type) counters."""
    successes: int = 0
    failures: int = 0
    last_run: float = 0.0
    result: Optional[bool] = None  # thresholded outcome (None = undecided)


def run_probe(spec: ProbeSpec, env: Dict[str, str],
              exec_runner=None) -> bool:
    """Execute one probe attempt; True = success. Never raises.

    ``exec_runner(command, env, timeout_s) -> exit_code`` runs exec probes
    inside the container's confinement (cgroup/credentials via the native
    launcher — k8s runs probes *inside* the container, so they must never
    execute as the kubelet's root user when runAsUser is set). The
    subprocess fallback exists only for unit tests of this module."""
    try:
        if spec.kind == "tcp":
            with socket.create_connection(("127.0.0.1", spec.port),
                                          timeout=spec.timeout_s):
                return True
        if spec.kind == "http":
            conn = http.client.HTTPConnection("127.0.0.1", spec.port,
                                              timeout=spec.timeout_s)
            try:
                conn.request("GET", spec.path)
                status = conn.getresponse().status
                return 200 <= status < 400
            finally:
                conn.close()
        if spec.kind == "sleep":
            import time as _time

            _time.sleep(spec.timeout_s)
            return True
        if spec.kind == "exec":
            if exec_runner is not None:
                return exec_runner(spec.command, env, spec.timeout_s) == 0
            proc = subprocess.run(
                spec.command, env=env, timeout=spec.timeout_s,
                stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL,
            )
            return proc.returncode == 0
    except Exception:
        return False
    return False


def advance(state: ProbeState, spec: ProbeSpec, ok: bool) -> Optional[bool]:
    """Fold one attempt into the state; returns the thresholded outcome
    (True/False) or None while undecided. k8s semantics: successThreshold
    consecutive successes flip to healthy, failureThreshold consecutive
    failures flip to unhealthy."""
    if ok:
        state.successes += 1
        state.failures = 0
        if state.successes >= spec.success_threshold:
            state.result = True
    else:
        state.failures += 1
        state.successes = 0
        if state.failures >= spec.failure_threshold:
            state.result = False
    return state.result
