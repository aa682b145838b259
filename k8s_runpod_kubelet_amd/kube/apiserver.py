"""Kubelet API server (:10250).

Counterpart of the reference's createAPIServer + api.AttachPodRoutes
(cmd/virtual_kubelet/main.go:217-248). The reference stubs RunInContainer and
GetContainerLogs with "not supported by RunPod" (kubelet.go:2027-2066);
containers here are local processes, so ``/containerLogs`` serves the real
log files (parity-plus per SURVEY §7 non-goals note) and ``/pods`` serves the
provider's tracked pods. Exec is non-interactive only (one-shot command in
the pod's environment) — interactive TTY streaming is not implemented.
"""

from __future__ import annotations

import json
import logging
import os
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

from ..utils.httpserver import QuietThreadingHTTPServer
from typing import Optional
from urllib.parse import parse_qs, unquote, urlparse

log = logging.getLogger("kube.apiserver")


class KubeletApiServer:
    def __init__(self, provider, internal_ip: str = "127.0.0.1",
                 port: int = 10250, token: str = ""):
        """token: bearer token required from non-loopback peers (the real
        kubelet authenticates :10250 via webhook/x509 — a static token +
        loopback allowance is the offline equivalent; empty token =
        loopback-only for every request)."""
        self.provider = provider
        self.internal_ip = internal_ip
        self.port = port
        self.token = token
        self._server: Optional[ThreadingHTTPServer] = None
        self._thread: Optional[threading.Thread] = None

    def start(self) -> None:
        outer = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, fmt, *args):
                log.debug(fmt % args)

            def _respond(self, code: int, body: bytes, ctype: str = "text/plain"):
                self.send_response(code)
                self.send_header("Content-Type", ctype)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def _stream_logs(self, namespace: str, pod: str, container: str,
                             tail: int) -> None:
                """`kubectl logs -f` (follow=true): chunked-encoding tail of
                the local log file, ending when the container terminates
                (the reference stubs logs entirely, kubelet.go:2047-2066)."""
                import time as _time

                path = outer.provider.get_container_log_path(
                    namespace, pod, container)
                if path is None:
                    self._respond(404, b"no log stream for this pod")
                    return
                self.send_response(200)
                self.send_header("Content-Type", "text/plain")
                self.send_header("Transfer-Encoding", "chunked")
                self.end_headers()

                def chunk(data: bytes) -> None:
                    self.wfile.write(b"%x\r\n" % len(data) + data + b"\r\n")
                    self.wfile.flush()

                try:
                    # one read for both backlog and offset: no gap between
                    # "what was sent" and "where the tail resumes"
                    try:
                        with open(path, "rb") as fh:
                            raw = fh.read()
                    except OSError:
                        raw = b""
                    offset = len(raw)
                    if tail > 0:
                        raw = b"".join(
                            raw.splitlines(keepends=True)[-tail:])
                    if raw:
                        chunk(raw)
                    while True:
                        finished = outer.provider.pod_log_finished(
                            namespace, pod)
                        try:
                            size = os.path.getsize(path)
                        except OSError:
                            size = offset
                        if size < offset:
                            # copytruncate rotation shrank the file:
                            # resume from the new beginning
                            offset = 0
                        if size > offset:
                            with open(path, "rb") as fh:
                                fh.seek(offset)
                                data = fh.read(size - offset)
                            offset = size
                            if data:
                                chunk(data)
                        elif finished:
                            break
                        else:
                            _time.sleep(0.1)
                    self.wfile.write(b"0\r\n\r\n")
                    self.wfile.flush()
                except (BrokenPipeError, ConnectionResetError):
                    pass  # client hung up (Ctrl-C on kubectl logs -f)

            def _authorized(self) -> bool:
                peer = self.client_address[0] if self.client_address else ""
                if peer in ("127.0.0.1", "::1", "::ffff:127.0.0.1"):
                    return True
                if outer.token:
                    return (self.headers.get("Authorization", "")
                            == f"Bearer {outer.token}")
                return False

            def do_GET(self):
                parsed = urlparse(self.path)
                if parsed.path not in ("/healthz",) and not self._authorized():
                    self._respond(401, b"unauthorized: kubelet API needs "
                                       b"loopback or bearer token")
                    return
                parts = [unquote(p) for p in parsed.path.strip("/").split("/") if p]
                query = parse_qs(parsed.query)
                if parts and parts[0] == "containerLogs" and len(parts) >= 3:
                    namespace, pod = parts[1], parts[2]
                    container = parts[3] if len(parts) > 3 else ""
                    tail = int(query.get("tailLines", ["-1"])[0])
                    limit_bytes = int(query.get("limitBytes", ["0"])[0])
                    follow = query.get(
                        "follow", ["false"])[0].lower() in ("true", "1")
                    previous = query.get(
                        "previous", ["false"])[0].lower() in ("true", "1")
                    if follow and not previous:
                        self._stream_logs(namespace, pod, container, tail)
                        return
                    text = outer.provider.get_container_logs(
                        namespace, pod, container, tail, previous=previous
                    )
                    body = text.encode()
                    if limit_bytes > 0:
                        body = body[:limit_bytes]  # kubectl logs --limit-bytes
                    self._respond(200, body)
                elif parts and parts[0] == "runningpods":
                    # kubelet debug endpoint: pods with live containers
                    # only (kubectl get --raw /api/v1/nodes/<n>/proxy/
                    # runningpods/ parity)
                    pods = outer.provider.get_running_pods()
                    body = json.dumps(
                        {"kind": "PodList", "apiVersion": "v1",
                         "items": pods}).encode()
                    self._respond(200, body, "application/json")
                elif parts and parts[0] == "pods":
                    pods = outer.provider.get_pods()
                    body = json.dumps(
                        {"kind": "PodList", "apiVersion": "v1", "items": pods}
                    ).encode()
                    self._respond(200, body, "application/json")
                elif parts and parts[0] == "stats" and len(parts) > 1 and parts[1] == "summary":
                    summary = outer.provider.get_stats_summary()
                    self._respond(200, json.dumps(summary).encode(),
                                  "application/json")
                elif parts and parts[0] == "configz":
                    # kubelet /configz parity: the live provider config
                    # (tokens redacted)
                    import dataclasses as _dc

                    cfg = outer.provider.config
                    data = _dc.asdict(cfg)
                    for k in list(data):
                        if "token" in k:
                            data[k] = "***" if data[k] else ""
                    self._respond(200, json.dumps(
                        {"kubeletconfig": data}).encode(),
                        "application/json")
                elif parts and parts[0] == "healthz":
                    self._respond(200, b"ok")
                else:
                    self._respond(404, b"not found")

            def do_POST(self):
                if not self._authorized():
                    self._respond(401, b"unauthorized: kubelet API needs "
                                       b"loopback or bearer token")
                    return
                parsed = urlparse(self.path)
                parts = [unquote(p) for p in parsed.path.strip("/").split("/") if p]
                query = parse_qs(parsed.query)
                if parts and parts[0] in ("exec", "run") and len(parts) >= 3:
                    commands = query.get("command", [])
                    if not commands:
                        self._respond(400, b"missing command")
                        return
                    namespace, pod = parts[1], parts[2]
                    container = parts[3] if len(parts) > 3 else ""
                    code, output = outer.provider.run_in_container(
                        namespace, pod, commands, container=container
                    )
                    body = json.dumps({"exitCode": code, "output": output}).encode()
                    self._respond(200 if code == 0 else 500, body,
                                  "application/json")
                else:
                    self._respond(404, b"not found")

        self._server = QuietThreadingHTTPServer((self.internal_ip, self.port), Handler)
        self.port = self._server.server_address[1]
        self._thread = threading.Thread(
            target=self._server.serve_forever, name="kubelet-api", daemon=True
        )
        self._thread.start()
        log.info("kubelet API listening",
                 extra={"addr": f"{self.internal_ip}:{self.port}"})

    def stop(self) -> None:
        if self._server is not None:
            self._server.shutdown()
            self._server.server_close()
        if self._thread is not None:
            self._thread.join(timeout=5.0)
