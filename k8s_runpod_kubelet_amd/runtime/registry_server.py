"""Minimal OCI Distribution (registry v2) server over an ImageStore.

Two jobs: (a) hermetic tests for RegistryClient.pull over real sockets,
(b) an operator tool — one node that received images (layout/archive
import) can serve them to the rest of the cluster:

    python -m k8s_runpod_kubelet_amd.runtime.registry_server --port 5000

Read-only API: /v2/ ping, /v2/<name>/manifests/<ref>, /v2/<name>/blobs/<d>.
Optional bearer token. Resolution: <name>:<tag> against the store's
reference annotations; manifests are also addressable by digest."""

from __future__ import annotations

import argparse
import logging
import re
import sys
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional

from .oci import ImageStore, _read_json, normalize_ref

log = logging.getLogger("runtime.registry_server")

_MANIFEST_RE = re.compile(r"^/v2/(.+)/manifests/([^/]+)$")
_BLOB_RE = re.compile(r"^/v2/(.+)/blobs/(sha256:[0-9a-f]+)$")


class RegistryServer:
    def __init__(self, store: ImageStore, host: str = "127.0.0.1",
                 port: int = 0, token: str = ""):
        self.store = store
        self.host = host
        self.port = port
        self.token = token
        self._server: Optional[ThreadingHTTPServer] = None
        self._thread: Optional[threading.Thread] = None

    @property
    def url(self) -> str:
        return f"http://{self.host}:{self.port}"

    def _find_layout(self, name: str, ref: str):
        """(layout_dir, manifest_descriptor) for name:ref (tag or digest).
        Tag lookup matches the repository path of the stored reference
        with its registry host stripped (the serving host differs from
        the original registry's)."""
        if not self.store.layouts_dir.is_dir():
            return None, None
        for d in sorted(self.store.layouts_dir.iterdir()):
            idx_file = d / "index.json"
            if not idx_file.exists():
                continue
            index = _read_json(idx_file)
            for m in index.get("manifests", []):
                if ref.startswith("sha256:"):
                    if m.get("digest") == ref:
                        return d, m
                    continue
                ann = (m.get("annotations") or {}).get(
                    "org.opencontainers.image.ref.name", "")
                if ann and self._repo_tag(normalize_ref(ann)) == (name, ref):
                    return d, m
        return None, None

    @staticmethod
    def _repo_tag(norm_ref: str):
        """'host/path/name:tag' -> ('path/name', 'tag')."""
        rest = norm_ref.partition("/")[2]
        repo, _, tag = rest.rpartition(":")
        return repo, tag

    def start(self) -> "RegistryServer":
        outer = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, fmt, *args):
                log.debug(fmt % args)

            def _respond(self, code, body=b"", ctype="application/json"):
                self.send_response(code)
                self.send_header("Content-Type", ctype)
                self.send_header("Content-Length", str(len(body)))
                if code == 401:
                    self.send_header("WWW-Authenticate", "Bearer")
                self.end_headers()
                if self.command != "HEAD":
                    self.wfile.write(body)

            def _authed(self) -> bool:
                if not outer.token:
                    return True
                return (self.headers.get("Authorization", "")
                        == f"Bearer {outer.token}")

            def do_GET(self):  # noqa: N802
                if not self._authed():
                    return self._respond(401, b'{"errors":[]}')
                if self.path == "/v2/" or self.path == "/v2":
                    return self._respond(200, b"{}")
                if m := _MANIFEST_RE.match(self.path):
                    name, ref = m.group(1), m.group(2)
                    layout, desc = outer._find_layout(name, ref)
                    if layout is None:
                        return self._respond(404, b'{"errors":[]}')
                    blob = (layout / "blobs" /
                            desc["digest"].replace(":", "/"))
                    data = blob.read_bytes()
                    return self._respond(
                        200, data,
                        desc.get("mediaType",
                                 "application/vnd.oci.image.manifest.v1+json"))
                if m := _BLOB_RE.match(self.path):
                    name, digest = m.group(1), m.group(2)
                    # search every layout for the blob (content-addressed)
                    if outer.store.layouts_dir.is_dir():
                        for d in outer.store.layouts_dir.iterdir():
                            blob = d / "blobs" / digest.replace(":", "/")
                            if blob.exists():
                                return self._respond(
                                    200, blob.read_bytes(),
                                    "application/octet-stream")
                    return self._respond(404, b'{"errors":[]}')
                return self._respond(404, b'{"errors":[]}')

            do_HEAD = do_GET  # noqa: N815

        self._server = ThreadingHTTPServer((self.host, self.port), Handler)
        self.port = self._server.server_address[1]
        self._thread = threading.Thread(target=self._server.serve_forever,
                                        name="registry-server", daemon=True)
        self._thread.start()
        log.info("registry serving", extra={"url": self.url})
        return self

    def stop(self) -> None:
        if self._server is not None:
            self._server.shutdown()
            self._server.server_close()
        if self._thread is not None:
            self._thread.join(timeout=5.0)


def main(argv=None) -> int:
    from ..config import Config

    ap = argparse.ArgumentParser(prog="amdvk-registry")
    ap.add_argument("--store", default="")
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, default=5000)
    ap.add_argument("--token", default="")
    args = ap.parse_args(argv)
    store = ImageStore(args.store or Config().resolved_image_store_dir())
    srv = RegistryServer(store, args.host, args.port, args.token).start()
    print(f"serving {len(store.list_refs())} image(s) at {srv.url}")
    try:
        threading.Event().wait()
    except KeyboardInterrupt:
        srv.stop()
    return 0


if __name__ == "__main__":
    sys.exit(main())
