"""OCI Distribution (registry v2) client: pull images into the local store.

The reference's backend pulls `Containers[0].Image` from a registry
(RunPod's side of runpod_client.go:1304); this node normally has no
egress, so the store is fed from layouts/archives — but clusters commonly
run an in-network registry, and `RegistryClient.pull` speaks the standard
`/v2/` protocol (manifest negotiation incl. manifest lists, blob fetch,
sha256 verification) into an OCI layout registered in the ImageStore.
`registry_server.py` is the matching in-repo server, which also lets one
node serve its store to others."""

from __future__ import annotations

import hashlib
import json
import logging
import tempfile
from pathlib import Path
from typing import Dict, Optional, Tuple

import httpx

from .oci import (
    _INDEX_TYPES,
    _MANIFEST_TYPES,
    ImageError,
    ImageStore,
    normalize_ref,
)

log = logging.getLogger("runtime.registry")

_ACCEPT = ", ".join(sorted(_MANIFEST_TYPES | _INDEX_TYPES))


def parse_ref(ref: str) -> Tuple[str, str, str]:
    """normalized ref -> (registry_host, repository, tag_or_digest)."""
    norm = normalize_ref(ref)
    host, _, rest = norm.partition("/")
    if "@" in rest:
        name, _, digest = rest.partition("@")
        return host, name, digest
    name, _, tag = rest.rpartition(":")
    return host, name, tag or "latest"


class RegistryError(ImageError):
    pass


class RegistryClient:
    def __init__(self, base_url: str = "", token: str = "",
                 timeout_s: float = 60.0, verify: bool = True):
        """base_url overrides the registry host parsed from the ref (use
        for mirrors / plain-HTTP in-cluster registries: 'http://host:port')."""
        self.base_url = base_url.rstrip("/")
        headers = {"Accept": _ACCEPT}
        if token:
            headers["Authorization"] = f"Bearer {token}"
        self._http = httpx.Client(headers=headers, timeout=timeout_s,
                                  verify=verify, follow_redirects=True)

    def close(self) -> None:
        self._http.close()

    def _url(self, host: str, path: str) -> str:
        base = self.base_url or f"https://{host}"
        return f"{base}{path}"

    def _get(self, url: str, accept: Optional[str] = None) -> httpx.Response:
        headers = {"Accept": accept} if accept else {}
        try:
            resp = self._http.get(url, headers=headers)
        except httpx.HTTPError as exc:
            # uniform error surface: transport failures (refused, DNS,
            # timeout, TLS) become RegistryError like protocol failures
            raise RegistryError(f"registry unreachable: {url}: {exc}")
        if resp.status_code == 401:
            raise RegistryError(f"unauthorized: {url}")
        if resp.status_code == 404:
            raise RegistryError(f"not found: {url}")
        if resp.status_code >= 400:
            raise RegistryError(f"registry error {resp.status_code}: {url}")
        return resp

    def pull(self, ref: str, store: ImageStore) -> str:
        """Pull ref into the store as an OCI layout; returns the
        normalized reference. Every blob is sha256-verified against the
        digest that named it."""
        host, name, tagish = parse_ref(ref)
        manifest_url = self._url(host, f"/v2/{name}/manifests/{tagish}")
        resp = self._get(manifest_url, accept=_ACCEPT)
        body = resp.content
        mtype = resp.headers.get("Content-Type", "").split(";")[0]
        if mtype in _INDEX_TYPES:
            index = json.loads(body)
            desc = _pick_platform(index.get("manifests", []))
            if desc is None:
                raise RegistryError(f"{ref}: no linux/amd64 manifest")
            resp = self._get(
                self._url(host, f"/v2/{name}/manifests/{desc['digest']}"),
                accept=", ".join(sorted(_MANIFEST_TYPES)))
            body = resp.content
            _verify(desc["digest"], body)
            mtype = resp.headers.get("Content-Type", "").split(";")[0]
        if mtype not in _MANIFEST_TYPES:
            raise RegistryError(f"{ref}: unexpected manifest type {mtype!r}")
        manifest = json.loads(body)
        man_digest = "sha256:" + hashlib.sha256(body).hexdigest()

        with tempfile.TemporaryDirectory(prefix="amdvk-pull-") as td:
            layout = Path(td)
            blobs = layout / "blobs" / "sha256"
            blobs.mkdir(parents=True)
            _write_blob(blobs, man_digest, body)
            cfg_desc = manifest.get("config", {})
            self._fetch_blob(host, name, cfg_desc.get("digest", ""), blobs)
            for lay in manifest.get("layers", []):
                self._fetch_blob(host, name, lay.get("digest", ""), blobs)
            (layout / "oci-layout").write_text(
                json.dumps({"imageLayoutVersion": "1.0.0"}))
            (layout / "index.json").write_text(json.dumps({
                "schemaVersion": 2,
                "manifests": [{
                    "mediaType":
                        "application/vnd.oci.image.manifest.v1+json",
                    "digest": man_digest, "size": len(body),
                    "annotations": {
                        "org.opencontainers.image.ref.name":
                            normalize_ref(ref)},
                }],
            }))
            out = store.add_layout(str(layout), ref)
        log.info("image pulled", extra={"ref": out, "digest": man_digest})
        return out

    def push(self, ref: str, store: ImageStore) -> str:
        """Push a locally stored image to the registry (standard upload
        flow: blob existence HEAD, POST /blobs/uploads/ + monolithic PUT,
        manifest PUT by tag). Returns the manifest digest."""
        img = store.resolve(ref)
        if img is None:
            raise RegistryError(f"{ref}: not in the local store")
        host, name, tagish = parse_ref(ref)
        if tagish.startswith("sha256:"):
            raise RegistryError("push needs a tag reference")
        man_file = (img.layout_dir / "blobs" /
                    img.manifest_digest.replace(":", "/"))
        body = man_file.read_bytes()
        manifest = json.loads(body)
        for desc in [manifest.get("config") or {}] + list(
                manifest.get("layers") or []):
            digest = desc.get("digest", "")
            if not digest:
                raise RegistryError("manifest names a blob with no digest")
            blob = img.layout_dir / "blobs" / digest.replace(":", "/")
            self._push_blob(host, name, digest, blob)
        url = self._url(host, f"/v2/{name}/manifests/{tagish}")
        try:
            resp = self._http.put(url, content=body, headers={
                "Content-Type": manifest.get(
                    "mediaType",
                    "application/vnd.oci.image.manifest.v1+json")})
        except httpx.HTTPError as exc:
            raise RegistryError(f"registry unreachable: {url}: {exc}")
        if resp.status_code == 401:
            raise RegistryError(f"unauthorized: {url}")
        if resp.status_code not in (200, 201):
            raise RegistryError(
                f"manifest push failed {resp.status_code}: {resp.text}")
        log.info("image pushed", extra={"ref": normalize_ref(ref),
                                        "digest": img.manifest_digest})
        return img.manifest_digest

    def _push_blob(self, host: str, name: str, digest: str,
                   blob_file: Path) -> None:
        try:
            head = self._http.head(
                self._url(host, f"/v2/{name}/blobs/{digest}"))
            if head.status_code == 200:
                return  # registry already has it (content-addressed)
            start = self._http.post(
                self._url(host, f"/v2/{name}/blobs/uploads/"))
            if start.status_code == 401:
                raise RegistryError("unauthorized: blob upload")
            if start.status_code != 202:
                raise RegistryError(
                    f"upload start failed {start.status_code} "
                    f"(registry read-only? start it with --allow-push)")
            loc = start.headers.get("Location", "")
            if loc.startswith("/"):
                base = self.base_url or f"https://{host}"
                loc = f"{base}{loc}"
            sep = "&" if "?" in loc else "?"
            with open(blob_file, "rb") as fh:
                # file-object content: httpx streams it chunked from disk
                fin = self._http.put(f"{loc}{sep}digest={digest}",
                                     content=fh)
            if fin.status_code not in (200, 201):
                raise RegistryError(
                    f"blob upload failed {fin.status_code}: {fin.text}")
        except httpx.HTTPError as exc:
            raise RegistryError(f"registry unreachable: {exc}")

    def _fetch_blob(self, host: str, name: str, digest: str,
                    blobs: Path) -> None:
        if not digest:
            raise RegistryError("manifest names a blob with no digest")
        algo, _, hexd = digest.partition(":")
        if algo != "sha256":
            raise RegistryError(f"unsupported digest algorithm {algo!r}")
        url = self._url(host, f"/v2/{name}/blobs/{digest}")
        dst = blobs / hexd
        tmp = dst.with_name(f".{hexd}.part")
        h = hashlib.sha256()
        try:
            with self._http.stream("GET", url) as resp:
                if resp.status_code == 401:
                    raise RegistryError(f"unauthorized: {url}")
                if resp.status_code == 404:
                    raise RegistryError(f"not found: {url}")
                if resp.status_code >= 400:
                    raise RegistryError(
                        f"registry error {resp.status_code}: {url}")
                # stream to disk hashing on the fly: layer blobs are
                # GB-scale for ROCm images, never buffered whole
                with open(tmp, "wb") as fh:
                    for chunk in resp.iter_bytes(1 << 20):
                        h.update(chunk)
                        fh.write(chunk)
        except httpx.HTTPError as exc:
            tmp.unlink(missing_ok=True)
            raise RegistryError(f"registry unreachable: {url}: {exc}")
        except BaseException:
            tmp.unlink(missing_ok=True)
            raise
        if h.hexdigest() != hexd:
            tmp.unlink(missing_ok=True)
            raise RegistryError(f"digest mismatch for {digest}")
        tmp.rename(dst)


def _verify(digest: str, data: bytes) -> None:
    algo, _, hexd = digest.partition(":")
    if algo != "sha256":
        raise RegistryError(f"unsupported digest algorithm {algo!r}")
    actual = hashlib.sha256(data).hexdigest()
    if actual != hexd:
        raise RegistryError(
            f"digest mismatch: expected {digest}, got sha256:{actual}")


def _write_blob(blobs: Path, digest: str, data: bytes) -> None:
    (blobs / digest.partition(":")[2]).write_bytes(data)


def _pick_platform(manifests) -> Optional[Dict]:
    for m in manifests:
        p = m.get("platform") or {}
        if (p.get("os") in (None, "linux")
                and p.get("architecture") in (None, "amd64")):
            return m
    return manifests[0] if manifests else None
