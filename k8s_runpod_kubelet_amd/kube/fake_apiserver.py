"""HTTP facade over :class:`FakeKube` — an in-process Kubernetes apiserver.

Serves the exact REST surface this kubelet consumes (pods CRUD + watch
streaming, pods/status PATCH, nodes, Leases, Secrets/ConfigMaps/Jobs,
Events, SelfSubjectReview) over real sockets, so the *production* client
path — ``HttpK8sClient`` → informer watch stream → controllers → provider —
can be integration-tested end to end without a cluster. This is the offline
stand-in for the reference's kind/real-cluster integration testing
(reference runpod_test.go:182-390 needs a live cluster AND a paid cloud
account; SURVEY §4 calls that gap out).

Also runnable standalone for manual poking:

    python -m k8s_runpod_kubelet_amd.kube.fake_apiserver --port 8001
    kubectl --server http://127.0.0.1:8001 get pods   # read paths work
"""

from __future__ import annotations

import json
import logging
import re
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

from ..utils.httpserver import QuietThreadingHTTPServer
from typing import Any, Dict, Optional, Tuple
from urllib.parse import parse_qs, urlparse

from .client import ConflictError, GoneError, NotFoundError
from .fake import FakeKube

log = logging.getLogger("kube.fake_apiserver")

_POD_RE = re.compile(r"^/api/v1/namespaces/([^/]+)/pods/([^/]+)$")
_POD_STATUS_RE = re.compile(r"^/api/v1/namespaces/([^/]+)/pods/([^/]+)/status$")
_PODS_NS_RE = re.compile(r"^/api/v1/namespaces/([^/]+)/pods$")
_NODE_RE = re.compile(r"^/api/v1/nodes/([^/]+)$")
_NODE_STATUS_RE = re.compile(r"^/api/v1/nodes/([^/]+)/status$")
_LEASE_RE = re.compile(
    r"^/apis/coordination\.k8s\.io/v1/namespaces/([^/]+)/leases/([^/]+)$")
_LEASES_NS_RE = re.compile(
    r"^/apis/coordination\.k8s\.io/v1/namespaces/([^/]+)/leases$")
_SECRET_RE = re.compile(r"^/api/v1/namespaces/([^/]+)/secrets/([^/]+)$")
_CONFIGMAP_RE = re.compile(r"^/api/v1/namespaces/([^/]+)/configmaps/([^/]+)$")
_JOB_RE = re.compile(r"^/apis/batch/v1/namespaces/([^/]+)/jobs/([^/]+)$")
_EVENTS_NS_RE = re.compile(r"^/api/v1/namespaces/([^/]+)/events$")
_EVENT_RE = re.compile(r"^/api/v1/namespaces/([^/]+)/events/([^/]+)$")


class FakeApiServer:
    """Threaded HTTP server delegating every route to a FakeKube."""

    def __init__(self, kube: Optional[FakeKube] = None,
                 host: str = "127.0.0.1", port: int = 0,
                 required_token: str = ""):
        self.kube = kube if kube is not None else FakeKube()
        self.host = host
        self.port = port
        # Bearer-token enforcement (mutable at runtime — tests rotate it
        # to exercise the client's 401-refresh path, like a real apiserver
        # after a bound SA token expires)
        self.required_token = required_token
        self._server: Optional[ThreadingHTTPServer] = None
        self._thread: Optional[threading.Thread] = None

    @property
    def url(self) -> str:
        return f"http://{self.host}:{self.port}"

    def start(self) -> "FakeApiServer":
        outer = self

        class Handler(BaseHTTPRequestHandler):
            protocol_version = "HTTP/1.1"

            def log_message(self, fmt, *args):
                log.debug(fmt % args)

            def _authorized(self) -> bool:
                if not outer.required_token:
                    return True
                return (self.headers.get("Authorization", "")
                        == f"Bearer {outer.required_token}")

            def parse_request(self):
                # enforce auth uniformly before routing
                ok = super().parse_request()
                if ok and not self._authorized():
                    body = json.dumps({"kind": "Status", "code": 401,
                                       "message": "Unauthorized"}).encode()
                    self.send_response(401)
                    self.send_header("Content-Type", "application/json")
                    self.send_header("Content-Length", str(len(body)))
                    self.end_headers()
                    self.wfile.write(body)
                    # drain request body if any so keep-alive stays sane
                    length = int(self.headers.get("Content-Length", 0) or 0)
                    if length:
                        self.rfile.read(length)
                    return False
                return ok

            # -- plumbing --

            def _json(self, code: int, obj: Any) -> None:
                body = json.dumps(obj).encode()
                self.send_response(code)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def _err(self, exc: Exception) -> None:
                if isinstance(exc, NotFoundError):
                    self._json(404, {"kind": "Status", "code": 404,
                                     "reason": "NotFound", "message": str(exc)})
                elif isinstance(exc, ConflictError):
                    self._json(409, {"kind": "Status", "code": 409,
                                     "reason": "Conflict", "message": str(exc)})
                else:
                    log.exception("fake apiserver handler error")
                    self._json(500, {"kind": "Status", "code": 500,
                                     "message": str(exc)})

            def _body(self) -> Dict[str, Any]:
                length = int(self.headers.get("Content-Length") or 0)
                raw = self.rfile.read(length) if length else b"{}"
                return json.loads(raw or b"{}")

            def _route(self) -> Tuple[str, Dict[str, str]]:
                parsed = urlparse(self.path)
                params = {k: v[0] for k, v in parse_qs(parsed.query).items()}
                return parsed.path, params

            # -- verbs --

            def do_GET(self):
                path, params = self._route()
                kube = outer.kube
                try:
                    if path == "/api/v1/pods" or _PODS_NS_RE.match(path):
                        ns_m = _PODS_NS_RE.match(path)
                        namespace = ns_m.group(1) if ns_m else None
                        if params.get("watch") == "true":
                            return self._stream_watch(namespace, params)
                        items, rv = kube.list_pods_with_rv(
                            namespace=namespace,
                            field_selector=params.get("fieldSelector", ""),
                            label_selector=params.get("labelSelector", ""),
                        )
                        return self._json(200, {
                            "kind": "PodList",
                            "metadata": {"resourceVersion": rv},
                            "items": items})
                    if m := _POD_RE.match(path):
                        return self._json(200, kube.get_pod(m.group(1), m.group(2)))
                    if m := _NODE_RE.match(path):
                        return self._json(200, kube.get_node(m.group(1)))
                    if path == "/apis/coordination.k8s.io/v1":
                        if not kube.leases_supported():
                            raise NotFoundError("coordination API disabled")
                        return self._json(200, {"kind": "APIResourceList"})
                    if m := _LEASE_RE.match(path):
                        return self._json(200, kube.get_lease(m.group(1), m.group(2)))
                    if m := _SECRET_RE.match(path):
                        return self._json(200, kube.get_secret(m.group(1), m.group(2)))
                    if m := _CONFIGMAP_RE.match(path):
                        return self._json(200, kube.get_configmap(m.group(1), m.group(2)))
                    if m := _JOB_RE.match(path):
                        return self._json(200, kube.get_job(m.group(1), m.group(2)))
                    raise NotFoundError(f"no route {path}")
                except Exception as exc:  # noqa: BLE001 — translated to HTTP
                    self._err(exc)

            def _stream_watch(self, namespace, params):
                kube = outer.kube
                timeout_s = float(params.get("timeoutSeconds", "30"))
                self.send_response(200)
                self.send_header("Content-Type", "application/json")
                self.send_header("Transfer-Encoding", "chunked")
                self.end_headers()

                def chunk(data: bytes) -> None:
                    self.wfile.write(f"{len(data):x}\r\n".encode())
                    self.wfile.write(data + b"\r\n")
                    self.wfile.flush()

                try:
                    try:
                        for ev_type, obj in kube.watch_pods(
                            namespace=namespace,
                            field_selector=params.get("fieldSelector", ""),
                            resource_version=params.get(
                                "resourceVersion", ""),
                            timeout_s=timeout_s,
                        ):
                            chunk(json.dumps(
                                {"type": ev_type,
                                 "object": obj}).encode() + b"\n")
                    except GoneError as exc:
                        # compacted RV: in-stream Status(410), the same
                        # shape a real apiserver emits mid-watch
                        chunk(json.dumps({
                            "type": "ERROR",
                            "object": {"kind": "Status", "code": 410,
                                       "reason": "Expired",
                                       "message": str(exc)},
                        }).encode() + b"\n")
                    chunk(b"")  # terminating chunk
                except (BrokenPipeError, ConnectionResetError):
                    pass  # client hung up — normal for watches

            def do_POST(self):
                path, _ = self._route()
                kube = outer.kube
                try:
                    body = self._body()
                    if m := _PODS_NS_RE.match(path):
                        return self._json(201, kube.create_pod(m.group(1), body))
                    if path == "/api/v1/nodes":
                        return self._json(201, kube.create_node(body))
                    if m := _LEASES_NS_RE.match(path):
                        return self._json(201, kube.create_lease(m.group(1), body))
                    if m := _EVENTS_NS_RE.match(path):
                        return self._json(201, kube.create_event(m.group(1), body))
                    if path == "/apis/authentication.k8s.io/v1/selfsubjectreviews":
                        return self._json(201, kube.self_subject_review())
                    raise NotFoundError(f"no route {path}")
                except Exception as exc:  # noqa: BLE001
                    self._err(exc)

            def do_PUT(self):
                path, _ = self._route()
                kube = outer.kube
                try:
                    body = self._body()
                    if m := _POD_RE.match(path):
                        return self._json(200, kube.update_pod(m.group(1), body))
                    if m := _NODE_RE.match(path):
                        return self._json(200, kube.update_node(body))
                    if m := _LEASE_RE.match(path):
                        return self._json(200, kube.update_lease(m.group(1), body))
                    if m := _EVENT_RE.match(path):
                        return self._json(200,
                                          kube.update_event(m.group(1), body))
                    raise NotFoundError(f"no route {path}")
                except Exception as exc:  # noqa: BLE001
                    self._err(exc)

            def do_PATCH(self):
                path, _ = self._route()
                kube = outer.kube
                try:
                    body = self._body()
                    if m := _POD_STATUS_RE.match(path):
                        return self._json(
                            200, kube.patch_pod_status(m.group(1), m.group(2), body))
                    if m := _POD_RE.match(path):
                        return self._json(
                            200, kube.patch_pod(m.group(1), m.group(2), body))
                    if m := _NODE_STATUS_RE.match(path):
                        return self._json(
                            200, kube.patch_node_status(m.group(1), body))
                    raise NotFoundError(f"no route {path}")
                except Exception as exc:  # noqa: BLE001
                    self._err(exc)

            def do_DELETE(self):
                path, params = self._route()
                kube = outer.kube
                try:
                    if m := _POD_RE.match(path):
                        grace = params.get("gracePeriodSeconds")
                        kube.delete_pod(m.group(1), m.group(2),
                                        grace_period_s=int(grace) if grace else None)
                        return self._json(200, {"kind": "Status", "status": "Success"})
                    if m := _NODE_RE.match(path):
                        kube.delete_node(m.group(1))
                        return self._json(200, {"kind": "Status", "status": "Success"})
                    raise NotFoundError(f"no route {path}")
                except Exception as exc:  # noqa: BLE001
                    self._err(exc)

        self._server = QuietThreadingHTTPServer((self.host, self.port), Handler)
        self.port = self._server.server_address[1]
        self._thread = threading.Thread(
            target=self._server.serve_forever, name="fake-apiserver", daemon=True)
        self._thread.start()
        log.info("fake apiserver listening", extra={"url": self.url})
        return self

    def stop(self) -> None:
        if self._server is not None:
            self._server.shutdown()
            self._server.server_close()
        if self._thread is not None:
            self._thread.join(timeout=5.0)


def main() -> None:
    import argparse
    import time as _time

    ap = argparse.ArgumentParser()
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8001)
    args = ap.parse_args()
    srv = FakeApiServer(host=args.host, port=args.port).start()
    print(f"fake apiserver at {srv.url} (Ctrl-C to stop)")
    try:
        while True:
            _time.sleep(3600)
    except KeyboardInterrupt:
        srv.stop()


if __name__ == "__main__":
    main()
