"""Local OCI image store: layout resolution, layer unpacking, rootfs cache.

The reference's contract is that ``Containers[0].Image`` actually runs —
RunPod's backend pulls and executes it (reference
pkg/virtual_kubelet/runpod_client.go:1304 puts the image into the deploy
params; runpod_test.go:99 deploys a CUDA image with no command). This module
gives the local runtime the same contract without a registry (this node is
offline): images live in a local store of standard **OCI image layouts**
(the format written by ``skopeo copy``, ``podman save --format oci-dir``,
``umoci``), resolved by reference and unpacked into a content-addressed
rootfs cache that per-pod overlay/chroot roots are built from
(runtime/process_runtime.py).

Store layout:

    <store>/layouts/<enc(ref)>/      one OCI layout per image reference
        oci-layout                   {"imageLayoutVersion": "1.0.0"}
        index.json                   -> manifest (or nested index)
        blobs/sha256/<digest>        manifests, configs, layer tars
    <store>/rootfs/<manifest-digest>/  unpacked layer stack (shared, ro)

Also importable from tars: ``import_archive`` accepts an oci-archive
(``skopeo copy ... oci-archive:img.tar`` / ``podman save --format
oci-archive``) so operators can ship images to the node as files.
"""

from __future__ import annotations

import hashlib
import json
import logging
import os
import shutil
import tarfile
import tempfile
from dataclasses import dataclass, field
from pathlib import Path
from typing import Any, Dict, List, Optional, Tuple

log = logging.getLogger("runtime.oci")

REF_ANNOTATION = "org.opencontainers.image.ref.name"


class ImageError(Exception):
    """Malformed layout / unresolvable reference / unsafe layer content."""


@dataclass
class ImageConfig:
    """The runtime-relevant subset of an OCI image config."""
    env: List[str] = field(default_factory=list)
    entrypoint: List[str] = field(default_factory=list)
    cmd: List[str] = field(default_factory=list)
    working_dir: str = ""
    user: str = ""

    @property
    def path_env(self) -> str:
        for e in self.env:
            if e.startswith("PATH="):
                return e[5:]
        return "/usr/local/sbin:/usr/local/bin:/usr/sbin:/usr/bin:/sbin:/bin"


@dataclass
class ResolvedImage:
    ref: str
    manifest_digest: str            # content address of the rootfs
    config: ImageConfig
    layers: List[Path]              # layer tar blobs, base first
    layout_dir: Path


def normalize_ref(ref: str) -> str:
    """Docker-style reference normalization: default registry/library and
    :latest, so ``busybox`` == ``docker.io/library/busybox:latest``."""
    ref = ref.strip()
    if not ref:
        return ref
    # registry detection BEFORE tag defaulting (a ':tag' on a bare name
    # must not look like a registry port)
    if "/" in ref:
        head = ref.split("/", 1)[0]
        has_registry = "." in head or ":" in head or head == "localhost"
    else:
        has_registry = False
    if not has_registry:
        ref = ("docker.io/" + ref) if "/" in ref else (
            "docker.io/library/" + ref)
    elif ref.startswith("docker.io/") and ref.count("/") == 1:
        ref = "docker.io/library/" + ref.split("/", 1)[1]
    if "@" not in ref:
        # tag detection: a ':' after the last '/' is a tag separator
        last = ref.rsplit("/", 1)[-1]
        if ":" not in last:
            ref += ":latest"
    return ref


def _enc(ref: str) -> str:
    """Filesystem-safe directory name for a reference."""
    safe = ref.replace("/", "_").replace(":", "_").replace("@", "_")
    return f"{safe}-{hashlib.sha256(ref.encode()).hexdigest()[:12]}"


def _read_json(path: Path) -> Dict[str, Any]:
    try:
        return json.loads(path.read_text())
    except (OSError, json.JSONDecodeError) as exc:
        raise ImageError(f"bad JSON at {path}: {exc}") from exc


def _blob(layout: Path, digest: str) -> Path:
    algo, _, hexd = digest.partition(":")
    if not hexd or "/" in hexd or "/" in algo:
        raise ImageError(f"bad digest {digest!r}")
    p = layout / "blobs" / algo / hexd
    if not p.exists():
        raise ImageError(f"missing blob {digest} in {layout}")
    return p


_MANIFEST_TYPES = {
    "application/vnd.oci.image.manifest.v1+json",
    "application/vnd.docker.distribution.manifest.v2+json",
}
_INDEX_TYPES = {
    "application/vnd.oci.image.index.v1+json",
    "application/vnd.docker.distribution.manifest.list.v2+json",
}


class ImageStore:
    def __init__(self, root: str):
        self.root = Path(root)
        self.layouts_dir = self.root / "layouts"
        self.rootfs_dir = self.root / "rootfs"

    # ---- resolution ----

    def list_refs(self) -> List[str]:
        out = []
        if not self.layouts_dir.is_dir():
            return out
        for d in sorted(self.layouts_dir.iterdir()):
            ref_file = d / ".amdvk-ref"
            if ref_file.exists():
                out.append(ref_file.read_text().strip())
        return out

    def _layout_for(self, ref: str) -> Optional[Path]:
        norm = normalize_ref(ref)
        for candidate in (norm, ref):
            d = self.layouts_dir / _enc(candidate)
            if (d / "index.json").exists():
                return d
        return None

    def resolve(self, ref: str) -> Optional[ResolvedImage]:
        """Reference -> manifest/config/layers, or None when the image is
        not in the store (caller decides the fallback policy)."""
        layout = self._layout_for(ref)
        if layout is None:
            return None
        index = _read_json(layout / "index.json")
        manifests = index.get("manifests", [])
        if not manifests:
            raise ImageError(f"{ref}: empty index")
        desc = self._pick(manifests, normalize_ref(ref))
        # one level of nesting (index -> index) is common (buildkit)
        for _ in range(2):
            if desc.get("mediaType") in _INDEX_TYPES:
                nested = _read_json(_blob(layout, desc["digest"]))
                desc = self._pick(nested.get("manifests", []), "")
            else:
                break
        if desc.get("mediaType") not in _MANIFEST_TYPES:
            raise ImageError(
                f"{ref}: unsupported mediaType {desc.get('mediaType')!r}")
        manifest_digest = desc["digest"]
        manifest = _read_json(_blob(layout, manifest_digest))
        cfg_raw = _read_json(
            _blob(layout, manifest["config"]["digest"])).get("config", {}) or {}
        config = ImageConfig(
            env=list(cfg_raw.get("Env") or []),
            entrypoint=list(cfg_raw.get("Entrypoint") or []),
            cmd=list(cfg_raw.get("Cmd") or []),
            working_dir=cfg_raw.get("WorkingDir") or "",
            user=cfg_raw.get("User") or "",
        )
        layers = [_blob(layout, lay["digest"])
                  for lay in manifest.get("layers", [])]
        return ResolvedImage(ref=normalize_ref(ref),
                             manifest_digest=manifest_digest,
                             config=config, layers=layers, layout_dir=layout)

    @staticmethod
    def _pick(manifests: List[Dict[str, Any]], ref: str) -> Dict[str, Any]:
        """Choose the linux/amd64 (or unannotated) manifest; prefer an
        entry whose ref.name annotation matches."""
        def platform_ok(m):
            p = m.get("platform") or {}
            return (not p or (p.get("os") in (None, "linux")
                              and p.get("architecture") in (None, "amd64")))

        if ref:
            for m in manifests:
                ann = (m.get("annotations") or {}).get(REF_ANNOTATION, "")
                if ann and normalize_ref(ann) == ref and platform_ok(m):
                    return m
        for m in manifests:
            if platform_ok(m):
                return m
        return manifests[0]

    # ---- import ----

    def add_layout(self, src_dir: str, ref: str) -> str:
        """Register an existing OCI layout directory under a reference
        (copied into the store)."""
        norm = normalize_ref(ref)
        src = Path(src_dir)
        if not (src / "index.json").exists():
            raise ImageError(f"{src_dir}: not an OCI layout (no index.json)")
        dst = self.layouts_dir / _enc(norm)
        self.layouts_dir.mkdir(parents=True, exist_ok=True)
        tmp = Path(tempfile.mkdtemp(dir=self.layouts_dir, prefix=".import-"))
        try:
            for name in ("oci-layout", "index.json"):
                if (src / name).exists():
                    shutil.copy2(src / name, tmp / name)
            shutil.copytree(src / "blobs", tmp / "blobs", dirs_exist_ok=True)
            (tmp / ".amdvk-ref").write_text(norm + "\n")
            if dst.exists():
                shutil.rmtree(dst)
            tmp.rename(dst)
        except BaseException:
            shutil.rmtree(tmp, ignore_errors=True)
            raise
        log.info("image imported", extra={"ref": norm})
        return norm

    def import_archive(self, tar_path: str, ref: str = "") -> str:
        """Import an oci-archive tar (an OCI layout tarred at its root)."""
        with tempfile.TemporaryDirectory(prefix="amdvk-ociimp-") as td:
            with tarfile.open(tar_path) as tf:
                _safe_extract_plain(tf, Path(td))
            if not ref:
                index = _read_json(Path(td) / "index.json")
                for m in index.get("manifests", []):
                    ann = (m.get("annotations") or {}).get(REF_ANNOTATION, "")
                    if ann:
                        ref = ann
                        break
            if not ref:
                raise ImageError(
                    f"{tar_path}: no ref annotation; pass ref explicitly")
            return self.add_layout(td, ref)

    # ---- rootfs cache ----

    def rootfs_for(self, image: ResolvedImage) -> Path:
        """Unpacked (merged) rootfs for the image, shared read-only between
        pods; built once per manifest digest."""
        key = image.manifest_digest.replace(":", "-")
        dst = self.rootfs_dir / key
        done = dst / ".amdvk-unpacked"
        if done.exists():
            return dst
        self.rootfs_dir.mkdir(parents=True, exist_ok=True)
        tmp = Path(tempfile.mkdtemp(dir=self.rootfs_dir, prefix=".unpack-"))
        try:
            for layer in image.layers:
                _apply_layer(layer, tmp)
            (tmp / ".amdvk-unpacked").write_text(image.manifest_digest + "\n")
            if dst.exists():
                shutil.rmtree(dst)
            tmp.rename(dst)
        except BaseException:
            shutil.rmtree(tmp, ignore_errors=True)
            raise
        log.info("image unpacked",
                 extra={"ref": image.ref, "digest": image.manifest_digest})
        return dst

    def image_sizes(self) -> List[Tuple[str, int]]:
        """(ref, total blob bytes) per stored image — feeds
        node.status.images (kubectl describe node surface)."""
        out: List[Tuple[str, int]] = []
        if not self.layouts_dir.is_dir():
            return out
        for d in sorted(self.layouts_dir.iterdir()):
            ref_file = d / ".amdvk-ref"
            if not ref_file.exists():
                continue
            size = 0
            blobs = d / "blobs"
            if blobs.is_dir():
                for blob in blobs.rglob("*"):
                    if blob.is_file():
                        size += blob.stat().st_size
            out.append((ref_file.read_text().strip(), size))
        return out

    def remove(self, ref: str) -> bool:
        """Unregister an image reference (its layout); the shared rootfs
        cache is reclaimed by gc()."""
        layout = self._layout_for(ref)
        if layout is None:
            return False
        shutil.rmtree(layout, ignore_errors=True)
        return True

    def gc(self) -> List[str]:
        """Remove unpacked-rootfs cache entries whose manifest digest no
        longer belongs to any stored image (kubelet image-GC analogue).
        Returns the removed digests. Never touches per-pod container dirs
        — those belong to the runtime."""
        live = set()
        for ref in self.list_refs():
            try:
                img = self.resolve(ref)
            except ImageError:
                continue
            if img is not None:
                live.add(img.manifest_digest.replace(":", "-"))
        removed = []
        if self.rootfs_dir.is_dir():
            for d in self.rootfs_dir.iterdir():
                if d.name.startswith("."):
                    continue  # in-flight unpack
                if d.name not in live:
                    shutil.rmtree(d, ignore_errors=True)
                    removed.append(d.name)
        if removed:
            log.info("image cache gc", extra={"removed": len(removed)})
        return removed

    def resolve_user(self, rootfs: Path, user: str) -> Tuple[int, int]:
        """OCI config User ('uid', 'uid:gid', 'name', 'name:group') ->
        numeric (uid, gid); (-1, -1) when unset (inherit)."""
        if not user:
            return -1, -1
        uname, _, gname = user.partition(":")
        uid = _to_int(uname)
        gid = _to_int(gname) if gname else None
        if uid is None or (gname and gid is None):
            passwd = _parse_colon_file(rootfs / "etc" / "passwd")
            groups = _parse_colon_file(rootfs / "etc" / "group")
            if uid is None:
                row = passwd.get(uname)
                if row is None:
                    raise ImageError(f"image user {user!r} not in /etc/passwd")
                uid = int(row[1])
                if gid is None and not gname:
                    gid = int(row[2])
            if gname and gid is None:
                grow = groups.get(gname)
                if grow is None:
                    raise ImageError(f"image group {gname!r} not in /etc/group")
                gid = int(grow[1])
        if gid is None:
            # numeric uid with no explicit group: primary group from passwd
            passwd_by_uid = {int(v[1]): v for v in
                             _parse_colon_file(rootfs / "etc" / "passwd").values()
                             if _to_int(v[1]) is not None}
            row = passwd_by_uid.get(uid)
            gid = int(row[2]) if row else uid
        return uid, gid


def _to_int(s: str) -> Optional[int]:
    try:
        return int(s)
    except (TypeError, ValueError):
        return None


def _parse_colon_file(path: Path) -> Dict[str, List[str]]:
    out: Dict[str, List[str]] = {}
    try:
        for line in path.read_text().splitlines():
            parts = line.split(":")
            if len(parts) >= 3 and parts[0]:
                out[parts[0]] = parts[1:]
    except OSError:
        pass
    return out


# ---- layer application (OCI layer spec: whiteouts, opaque dirs) ----


def _clean_name(name: str) -> str:
    """Strip a leading './' prefix (NOT a character-set lstrip — that would
    eat '..' and defeat the traversal check)."""
    while name.startswith("./"):
        name = name[2:]
    return name.lstrip("/")


def _safe_dest(root: Path, name: str) -> Path:
    """Reject path traversal in layer members."""
    parts = [p for p in name.split("/") if p not in ("", ".")]
    if any(p == ".." for p in parts):
        raise ImageError(f"layer path escapes rootfs: {name!r}")
    return root.joinpath(*parts) if parts else root


def _safe_join(root: Path, name: str, depth: int = 0) -> Path:
    """Chroot-style join: resolve symlinked INTERMEDIATE components as if
    the filesystem root were `root` (absolute targets re-root at `root`;
    `..` clamps at `root`, like a real chroot). Without this, a malicious
    layer plants `etc -> /` and a later layer's `etc/passwd` write lands
    on the HOST filesystem — the classic unpacker escape."""
    if depth > 40:
        raise ImageError(f"symlink loop while resolving {name!r}")
    parts: List[str] = []
    for p in name.split("/"):
        if p in ("", "."):
            continue
        if p == "..":
            if parts:
                parts.pop()
            # else clamped at root (chroot semantics)
        else:
            parts.append(p)
    cur = root
    for i, part in enumerate(parts):
        cand = cur / part
        if i < len(parts) - 1 and cand.is_symlink():
            target = os.readlink(cand)
            rest = "/".join(parts[i + 1:])
            if target.startswith("/"):
                return _safe_join(root, target.lstrip("/") + "/" + rest,
                                  depth + 1)
            base = cur.relative_to(root)
            return _safe_join(root, f"{base}/{target}/{rest}", depth + 1)
        cur = cand
    return cur


def _apply_layer(layer_tar: Path, root: Path) -> None:
    """Apply one layer tar onto root, honoring OCI whiteouts:
    ``.wh.<name>`` deletes <name>; ``.wh..wh..opq`` empties the directory
    before this layer's contents apply. Every destination goes through the
    chroot-style _safe_join so symlinked parents cannot redirect writes or
    deletes outside the rootfs."""
    with tarfile.open(layer_tar) as tf:
        for member in tf:
            name = _clean_name(member.name)
            if not name:
                continue
            _safe_dest(root, name)  # loud rejection of '..' member names
            base = os.path.basename(name)
            parent = os.path.dirname(name)
            if base == ".wh..wh..opq":
                target = _safe_join(root, parent)
                if target.is_dir() and not target.is_symlink():
                    for child in target.iterdir():
                        _rm_rf(child)
                continue
            if base.startswith(".wh."):
                victim = _safe_join(root, os.path.join(parent, base[4:]))
                _rm_rf(victim)
                continue
            dest = _safe_join(root, name)
            _extract_member(tf, member, root, dest)


def _extract_member(tf: tarfile.TarFile, member: tarfile.TarInfo,
                    root: Path, dest: Path) -> None:
    # a changed file type replaces whatever a lower layer put there
    if dest.is_symlink() or (dest.exists() and not dest.is_dir()):
        if not (member.isdir() and dest.is_dir()):
            _rm_rf(dest)
    elif dest.is_dir() and not member.isdir():
        _rm_rf(dest)
    dest.parent.mkdir(parents=True, exist_ok=True)
    if member.isdir():
        dest.mkdir(exist_ok=True)
        _chmod_own(dest, member)
    elif member.issym():
        if dest.is_symlink() or dest.exists():
            _rm_rf(dest)
        # the link's TARGET string is stored verbatim (it resolves at
        # container runtime, inside the rootfs); only the link's own
        # location was safe-joined
        os.symlink(member.linkname, dest)
        _lchown(dest, member)
    elif member.islnk():
        # hardlink within the rootfs (chroot-style resolution)
        link_target = _safe_join(root, _clean_name(member.linkname))
        try:
            os.link(link_target, dest)
        except OSError:
            # cross-device / target missing: degrade to a copy
            if link_target.exists():
                shutil.copy2(link_target, dest)
    elif member.isfile():
        src_f = tf.extractfile(member)
        if src_f is None:
            return
        with open(dest, "wb") as out:
            shutil.copyfileobj(src_f, out)
        _chmod_own(dest, member)
    elif member.ischr() or member.isblk() or member.isfifo():
        # device nodes / fifos in layers: skip (devices are bind-mounted by
        # the runtime; CAP_MKNOD may be absent)
        return


def _chmod_own(path: Path, member: tarfile.TarInfo) -> None:
    try:
        os.chmod(path, member.mode & 0o7777)
    except OSError:
        pass
    _lchown(path, member)


def _lchown(path: Path, member: tarfile.TarInfo) -> None:
    if os.geteuid() != 0:
        return
    try:
        os.lchown(path, member.uid, member.gid)
    except OSError:
        pass


def _rm_rf(path: Path) -> None:
    try:
        if path.is_symlink() or not path.is_dir():
            path.unlink(missing_ok=True)
        else:
            shutil.rmtree(path, ignore_errors=True)
    except OSError:
        pass


def _safe_extract_plain(tf: tarfile.TarFile, root: Path) -> None:
    """Traversal- and symlink-safe extraction for archive imports (same
    chroot-style member handling as layers, minus whiteouts — a hostile
    oci-archive must not write through planted symlinks either)."""
    for member in tf:
        name = _clean_name(member.name)
        if not name:
            continue
        _safe_dest(root, name)  # raises on traversal
        dest = _safe_join(root, name)
        _extract_member(tf, member, root, dest)


# ---- building (test images + operator tooling) ----


def build_layout(dest_dir: str, ref: str, rootfs_dir: str,
                 entrypoint: Optional[List[str]] = None,
                 cmd: Optional[List[str]] = None,
                 env: Optional[List[str]] = None,
                 working_dir: str = "", user: str = "",
                 extra_layer_dirs: Optional[List[str]] = None,
                 whiteouts: Optional[List[str]] = None) -> str:
    """Build a minimal single-arch OCI layout from directory trees (one
    layer per dir; optional whiteout entries appended as a final layer).
    Used by tests and by the ``amdvk-image build`` operator tool — this
    node has no registry access, so images arrive as layouts/archives."""
    dest = Path(dest_dir)
    blobs = dest / "blobs" / "sha256"
    blobs.mkdir(parents=True, exist_ok=True)

    def put_blob(data: bytes) -> Tuple[str, int]:
        digest = hashlib.sha256(data).hexdigest()
        (blobs / digest).write_bytes(data)
        return f"sha256:{digest}", len(data)

    layer_descs = []
    diff_ids = []
    layer_dirs = [rootfs_dir] + list(extra_layer_dirs or [])
    for i, ldir in enumerate(layer_dirs):
        import io

        buf = io.BytesIO()
        with tarfile.open(fileobj=buf, mode="w") as tf:
            base = Path(ldir)
            for p in sorted(base.rglob("*")):
                tf.add(p, arcname=str(p.relative_to(base)), recursive=False)
            if i == len(layer_dirs) - 1:
                for wh in whiteouts or []:
                    parent, _, basen = wh.rpartition("/")
                    info = tarfile.TarInfo(
                        os.path.join(parent, f".wh.{basen}") if parent
                        else f".wh.{basen}")
                    info.size = 0
                    tf.addfile(info)
        data = buf.getvalue()
        digest, size = put_blob(data)
        diff_ids.append(digest)
        layer_descs.append({
            "mediaType": "application/vnd.oci.image.layer.v1.tar",
            "digest": digest, "size": size})

    config_json = json.dumps({
        "architecture": "amd64", "os": "linux",
        "config": {
            "Env": env or ["PATH=/usr/local/sbin:/usr/local/bin:/usr/sbin:"
                           "/usr/bin:/sbin:/bin"],
            "Entrypoint": entrypoint or [],
            "Cmd": cmd or [],
            "WorkingDir": working_dir,
            "User": user,
        },
        "rootfs": {"type": "layers", "diff_ids": diff_ids},
    }).encode()
    cfg_digest, cfg_size = put_blob(config_json)

    manifest_json = json.dumps({
        "schemaVersion": 2,
        "mediaType": "application/vnd.oci.image.manifest.v1+json",
        "config": {
            "mediaType": "application/vnd.oci.image.config.v1+json",
            "digest": cfg_digest, "size": cfg_size},
        "layers": layer_descs,
    }).encode()
    man_digest, man_size = put_blob(manifest_json)

    (dest / "oci-layout").write_text(
        json.dumps({"imageLayoutVersion": "1.0.0"}))
    (dest / "index.json").write_text(json.dumps({
        "schemaVersion": 2,
        "manifests": [{
            "mediaType": "application/vnd.oci.image.manifest.v1+json",
            "digest": man_digest, "size": man_size,
            "annotations": {REF_ANNOTATION: normalize_ref(ref)},
        }],
    }))
    return man_digest
