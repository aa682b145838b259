"""Health server: /healthz, /readyz, /metrics, /debug/threads.

Counterpart of the reference's HealthServer (pkg/virtual_kubelet/health.go:
11-74: /healthz = atomic liveness flag, /readyz = flag AND readyFunc wired to
provider.Ping, 5 s graceful stop) — plus the Prometheus endpoint and a
thread-dump debug endpoint the reference lacks (SURVEY §5.1/5.5 gaps)."""

from __future__ import annotations

import logging
import sys
import threading
import traceback
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

from ..utils.httpserver import QuietThreadingHTTPServer
from typing import Callable, Optional

from . import metrics

log = logging.getLogger("server.health")


class HealthServer:
    def __init__(self, address: str, ready_fn: Optional[Callable[[], None]] = None,
                 ledger=None, admin_token: str = ""):
        host, _, port = address.rpartition(":")
        self.host = host or "0.0.0.0"
        self.port = int(port)
        self.ready_fn = ready_fn
        self.ledger = ledger  # enables the cordon/uncordon admin endpoints
        # Admin (state-mutating) endpoints are restricted: loopback peers
        # always allowed; non-loopback needs this bearer token (set via
        # config/AMDVK_ADMIN_TOKEN). Empty token = loopback-only. The
        # health server binds 0.0.0.0 by default for probes — without this
        # gate any network peer could cordon every GPU.
        self.admin_token = admin_token
        self._alive = True
        self._server: Optional[ThreadingHTTPServer] = None
        self._thread: Optional[threading.Thread] = None

    def set_alive(self, value: bool) -> None:
        self._alive = value

    def start(self) -> None:
        outer = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, fmt, *args):  # route through logging
                log.debug(fmt % args)

            def _respond(self, code: int, body: bytes,
                         ctype: str = "text/plain") -> None:
                self.send_response(code)
                self.send_header("Content-Type", ctype)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def do_GET(self):
                if self.path == "/healthz":
                    if outer._alive:
                        self._respond(200, b"ok")
                    else:
                        self._respond(503, b"unhealthy")
                elif self.path == "/readyz":
                    if not outer._alive:
                        self._respond(503, b"not alive")
                        return
                    if outer.ready_fn is not None:
                        try:
                            outer.ready_fn()
                        except Exception as exc:
                            self._respond(503, f"not ready: {exc}".encode())
                            return
                    self._respond(200, b"ready")
                elif self.path == "/metrics":
                    self._respond(200, metrics.render(),
                                  "text/plain; version=0.0.4")
                elif self.path == "/debug/threads":
                    frames = sys._current_frames()
                    lines = []
                    for thread in threading.enumerate():
                        frame = frames.get(thread.ident)
                        lines.append(f"--- {thread.name} ({thread.ident}) ---")
                        if frame is not None:
                            lines.extend(
                                line.rstrip()
                                for line in traceback.format_stack(frame)
                            )
                    self._respond(200, "\n".join(lines).encode())
                else:
                    self._respond(404, b"not found")

            def _admin_authorized(self) -> bool:
                peer = self.client_address[0] if self.client_address else ""
                if peer in ("127.0.0.1", "::1", "::ffff:127.0.0.1"):
                    return True
                if outer.admin_token:
                    auth = self.headers.get("Authorization", "")
                    return auth == f"Bearer {outer.admin_token}"
                return False

            def do_POST(self):
                # Operator cordon/uncordon (kubectl-cordon analogue at GPU
                # granularity): POST /cordon/<idx> | /uncordon/<idx>.
                # Loopback or bearer-token only — these mutate scheduling
                # state and must not be open to arbitrary network peers.
                parts = [p for p in self.path.strip("/").split("/") if p]
                if (len(parts) == 2 and parts[0] in ("cordon", "uncordon")
                        and outer.ledger is not None):
                    if not self._admin_authorized():
                        self._respond(403, b"admin endpoint: loopback or "
                                           b"bearer token required")
                        return
                    try:
                        idx = int(parts[1])
                    except ValueError:
                        self._respond(400, b"bad gpu index")
                        return
                    cordoned = parts[0] == "cordon"
                    if not outer.ledger.set_cordoned(idx, cordoned):
                        self._respond(404, f"no GPU {idx}".encode())
                        return
                    log.info("gpu cordon state changed",
                             extra={"gpu": idx, "cordoned": cordoned})
                    self._respond(200, b"ok")
                else:
                    self._respond(404, b"not found")

        self._server = QuietThreadingHTTPServer((self.host, self.port), Handler)
        self.port = self._server.server_address[1]
        self._thread = threading.Thread(
            target=self._server.serve_forever, name="health-server", daemon=True
        )
        self._thread.start()
        log.info("health server listening", extra={"addr": f"{self.host}:{self.port}"})

    def stop(self) -> None:
        if self._server is not None:
            self._server.shutdown()
            self._server.server_close()
        if self._thread is not None:
            self._thread.join(timeout=5.0)
