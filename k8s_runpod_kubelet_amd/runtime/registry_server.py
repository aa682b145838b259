"""Minimal OCI Distribution (registry v2) server over an ImageStore.

Two jobs: (a) hermetic tests for RegistryClient.pull over real sockets,
(b) an operator tool — one node that received images (layout/archive
import) can serve them to the rest of the cluster:

    python -m k8s_runpod_kubelet_amd.runtime.registry_server --port 5000

Read-only API: /v2/ ping, /v2/<name>/manifests/<ref>, /v2/<name>/blobs/<d>.
With --allow-push, the standard push flow is accepted too (POST
/blobs/uploads/ + monolithic or chunked PUT, manifest PUT by tag) so
nodes can publish images to the cluster registry (`imagetool push`).
Optional bearer token. Resolution: <name>:<tag> against the store's
reference annotations; manifests are also addressable by digest."""

from __future__ import annotations

import argparse
import hashlib
import json
import logging
import re
import shutil
import sys
import tempfile
import threading
import uuid as uuidlib
from pathlib import Path
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

from ..utils.httpserver import QuietThreadingHTTPServer
from typing import Optional

from .oci import ImageStore, _read_json, normalize_ref

log = logging.getLogger("runtime.registry_server")

_MANIFEST_RE = re.compile(r"^/v2/(.+)/manifests/([^/]+)$")
_BLOB_RE = re.compile(r"^/v2/(.+)/blobs/(sha256:[0-9a-f]+)$")
_UPLOAD_START_RE = re.compile(r"^/v2/(.+)/blobs/uploads/?$")
_UPLOAD_RE = re.compile(r"^/v2/(.+)/blobs/uploads/([0-9a-f-]+)$")


class RegistryServer:
    def __init__(self, store: ImageStore, host: str = "127.0.0.1",
                 port: int = 0, token: str = "", allow_push: bool = False):
        self.store = store
        self.host = host
        self.port = port
        self.token = token
        self.allow_push = allow_push
        self._server: Optional[ThreadingHTTPServer] = None
        self._thread: Optional[threading.Thread] = None
        self._push_lock = threading.Lock()
        self._uploads: dict = {}  # uuid -> staging file path

    # ---- push support ----

    def _staging_dir(self) -> Path:
        d = Path(self.store.root) / "_push" / "sha256"
        d.mkdir(parents=True, exist_ok=True)
        return d

    def _find_blob_file(self, digest: str) -> Optional[Path]:
        """Content-addressed search: completed-push staging first, then
        every layout."""
        hexd = digest.partition(":")[2]
        staged = Path(self.store.root) / "_push" / "sha256" / hexd
        if staged.exists():
            return staged
        if self.store.layouts_dir.is_dir():
            for d in self.store.layouts_dir.iterdir():
                blob = d / "blobs" / digest.replace(":", "/")
                if blob.exists():
                    return blob
        return None

    def _purge_stale_uploads(self, staging: Path,
                             max_age_s: float = 3600.0) -> None:
        """Abandoned uploads (client died mid-push) are reaped after an
        hour so the staging dir cannot grow without bound."""
        import time as _time

        now = _time.time()
        with self._push_lock:
            for f in staging.iterdir():
                try:
                    if now - f.stat().st_mtime > max_age_s:
                        f.unlink()
                        self._uploads = {k: v for k, v in
                                         self._uploads.items() if v != f}
                except OSError:
                    pass

    def _commit_blob(self, digest: str, data: bytes) -> bool:
        if ("sha256:" + hashlib.sha256(data).hexdigest()) != digest:
            return False
        dst = self._staging_dir() / digest.partition(":")[2]
        tmp = dst.with_name(f".{dst.name}.{uuidlib.uuid4().hex[:8]}")
        tmp.write_bytes(data)
        tmp.rename(dst)
        return True

    def _commit_manifest(self, name: str, tag: str, body: bytes) -> str:
        """Validate + register a pushed manifest as a store layout; returns
        the manifest digest. Raises ValueError with an HTTP-able message."""
        try:
            manifest = json.loads(body)
        except ValueError:
            raise ValueError("MANIFEST_INVALID")
        descs = [manifest.get("config") or {}] + list(
            manifest.get("layers") or [])
        blob_files = {}
        for desc in descs:
            digest = desc.get("digest", "")
            f = self._find_blob_file(digest) if digest else None
            if f is None:
                raise ValueError(f"BLOB_UNKNOWN: {digest}")
            blob_files[digest] = f
        man_digest = "sha256:" + hashlib.sha256(body).hexdigest()
        ref = f"{name}:{tag}"
        with tempfile.TemporaryDirectory(prefix="amdvk-push-") as td:
            layout = Path(td)
            blobs = layout / "blobs" / "sha256"
            blobs.mkdir(parents=True)
            (blobs / man_digest.partition(":")[2]).write_bytes(body)
            for digest, f in blob_files.items():
                shutil.copy2(f, blobs / digest.partition(":")[2])
            (layout / "oci-layout").write_text(
                json.dumps({"imageLayoutVersion": "1.0.0"}))
            (layout / "index.json").write_text(json.dumps({
                "schemaVersion": 2,
                "manifests": [{
                    "mediaType": manifest.get(
                        "mediaType",
                        "application/vnd.oci.image.manifest.v1+json"),
                    "digest": man_digest, "size": len(body),
                    "annotations": {
                        "org.opencontainers.image.ref.name":
                            normalize_ref(ref)},
                }],
            }))
            with self._push_lock:
                self.store.add_layout(str(layout), ref)
        log.info("image pushed", extra={"ref": ref, "digest": man_digest})
        return man_digest

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
            # the store's registration ref (.amdvk-ref) is authoritative:
            # a layout re-registered under a new tag serves under THAT tag
            # even though its internal annotation still names the original
            stored_ref = ""
            try:
                stored_ref = (d / ".amdvk-ref").read_text().strip()
            except OSError:
                pass
            stored_match = bool(
                stored_ref
                and self._repo_tag(normalize_ref(stored_ref)) == (name, ref))
            for m in index.get("manifests", []):
                if ref.startswith("sha256:"):
                    if m.get("digest") == ref:
                        return d, m
                    continue
                if stored_match:
                    return d, m
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
                if self.path.split("?", 1)[0] == "/v2/_catalog":
                    repos = sorted({outer._repo_tag(r)[0]
                                    for r in outer.store.list_refs()})
                    return self._respond(200, json.dumps(
                        {"repositories": repos}).encode())
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
                    blob = outer._find_blob_file(digest)
                    if blob is None or not blob.exists():
                        return self._respond(404, b'{"errors":[]}')
                    # stream: layer blobs are GB-scale for ROCm images
                    size = blob.stat().st_size
                    self.send_response(200)
                    self.send_header("Content-Type",
                                     "application/octet-stream")
                    self.send_header("Content-Length", str(size))
                    self.send_header("Docker-Content-Digest", digest)
                    self.end_headers()
                    if self.command == "HEAD":
                        return
                    with open(blob, "rb") as fh:
                        while chunk := fh.read(1 << 20):
                            self.wfile.write(chunk)
                    return
                return self._respond(404, b'{"errors":[]}')

            do_HEAD = do_GET  # noqa: N815

            def _body(self):
                n = int(self.headers.get("Content-Length", "0") or 0)
                return self.rfile.read(n) if n else b""

            def _push_denied(self):
                if not self._authed():
                    self._respond(401, b'{"errors":[]}')
                    return True
                if not outer.allow_push:
                    self._respond(405, b'{"errors":[{"code":"DENIED"}]}')
                    return True
                return False

            def do_POST(self):  # noqa: N802
                if self._push_denied():
                    return
                from urllib.parse import parse_qs, urlparse

                parsed = urlparse(self.path)
                m = _UPLOAD_START_RE.match(parsed.path)
                if not m:
                    return self._respond(404, b'{"errors":[]}')
                name = m.group(1)
                digest = parse_qs(parsed.query).get("digest", [""])[0]
                if digest:  # single-POST monolithic upload
                    if not outer._commit_blob(digest, self._body()):
                        return self._respond(
                            400, b'{"errors":[{"code":"DIGEST_INVALID"}]}')
                    self.send_response(201)
                    self.send_header("Location",
                                     f"/v2/{name}/blobs/{digest}")
                    self.send_header("Docker-Content-Digest", digest)
                    self.send_header("Content-Length", "0")
                    self.end_headers()
                    return
                uid = str(uuidlib.uuid4())
                staging = Path(outer.store.root) / "_push" / "uploads"
                staging.mkdir(parents=True, exist_ok=True)
                outer._purge_stale_uploads(staging)
                with outer._push_lock:
                    outer._uploads[uid] = staging / uid
                (staging / uid).write_bytes(b"")
                self.send_response(202)
                self.send_header("Location",
                                 f"/v2/{name}/blobs/uploads/{uid}")
                self.send_header("Range", "0-0")
                self.send_header("Content-Length", "0")
                self.end_headers()

            def do_PATCH(self):  # noqa: N802
                if self._push_denied():
                    return
                m = _UPLOAD_RE.match(self.path.split("?", 1)[0])
                if not m:
                    return self._respond(404, b'{"errors":[]}')
                with outer._push_lock:
                    f = outer._uploads.get(m.group(2))
                if f is None:
                    return self._respond(404, b'{"errors":[]}')
                data = self._body()
                with open(f, "ab") as fh:
                    fh.write(data)
                size = f.stat().st_size
                self.send_response(202)
                self.send_header("Location", self.path)
                self.send_header("Range", f"0-{max(0, size - 1)}")
                self.send_header("Content-Length", "0")
                self.end_headers()

            def do_PUT(self):  # noqa: N802
                if self._push_denied():
                    return
                from urllib.parse import parse_qs, urlparse

                parsed = urlparse(self.path)
                if m := _UPLOAD_RE.match(parsed.path):
                    digest = parse_qs(parsed.query).get("digest", [""])[0]
                    uid = m.group(2)
                    with outer._push_lock:
                        f = outer._uploads.pop(uid, None)
                    if f is None:
                        return self._respond(404, b'{"errors":[]}')
                    data = f.read_bytes() + self._body()
                    f.unlink(missing_ok=True)
                    if not digest or not outer._commit_blob(digest, data):
                        return self._respond(
                            400, b'{"errors":[{"code":"DIGEST_INVALID"}]}')
                    self.send_response(201)
                    self.send_header("Location",
                                     f"/v2/{m.group(1)}/blobs/{digest}")
                    self.send_header("Docker-Content-Digest", digest)
                    self.send_header("Content-Length", "0")
                    self.end_headers()
                    return
                if m := _MANIFEST_RE.match(parsed.path):
                    name, ref = m.group(1), m.group(2)
                    if ref.startswith("sha256:"):
                        return self._respond(
                            400,
                            b'{"errors":[{"code":"TAG_INVALID",'
                            b'"message":"push manifests by tag"}]}')
                    try:
                        man_digest = outer._commit_manifest(
                            name, ref, self._body())
                    except ValueError as exc:
                        return self._respond(
                            400,
                            json.dumps({"errors": [
                                {"code": str(exc)}]}).encode())
                    self.send_response(201)
                    self.send_header(
                        "Location", f"/v2/{name}/manifests/{man_digest}")
                    self.send_header("Docker-Content-Digest", man_digest)
                    self.send_header("Content-Length", "0")
                    self.end_headers()
                    return
                return self._respond(404, b'{"errors":[]}')

        self._server = QuietThreadingHTTPServer((self.host, self.port), Handler)
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
    ap.add_argument("--allow-push", action="store_true",
                    help="accept image pushes into the store")
    args = ap.parse_args(argv)
    store = ImageStore(args.store or Config().resolved_image_store_dir())
    srv = RegistryServer(store, args.host, args.port, args.token,
                         allow_push=args.allow_push).start()
    print(f"serving {len(store.list_refs())} image(s) at {srv.url}")
    try:
        threading.Event().wait()
    except KeyboardInterrupt:
        srv.stop()
    return 0


if __name__ == "__main__":
    sys.exit(main())
