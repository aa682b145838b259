"""OCI registry pull path: RegistryClient against the in-repo registry
server over real sockets — manifest negotiation, blob fetch, sha256
verification, token auth. Gives `image:` pull capability a hermetic
offline test (the reference's backend pulls server-side)."""

import json

import pytest

from k8s_runpod_kubelet_amd.runtime.oci import ImageStore, build_layout
from k8s_runpod_kubelet_amd.runtime.registry import (
    RegistryClient,
    RegistryError,
    parse_ref,
)
from k8s_runpod_kubelet_amd.runtime.registry_server import RegistryServer


def test_parse_ref():
    assert parse_ref("busybox") == ("docker.io", "library/busybox", "latest")
    assert parse_ref("ghcr.io/org/app:v2") == ("ghcr.io", "org/app", "v2")
    assert parse_ref("example/app@sha256:" + "0" * 64) == (
        "docker.io", "example/app", "sha256:" + "0" * 64)


@pytest.fixture
def served_store(tmp_path):
    store = ImageStore(str(tmp_path / "src-store"))
    tree = tmp_path / "tree"
    (tree / "bin").mkdir(parents=True)
    (tree / "bin" / "tool").write_text("#!/bin/true\n")
    (tree / "etc").mkdir()
    (tree / "etc" / "release").write_text("pulled\n")
    layout = tmp_path / "layout"
    layout.mkdir()
    build_layout(str(layout), "example/pullme:v1", str(tree),
                 entrypoint=["/bin/tool"], env=["X=1"])
    store.add_layout(str(layout), "example/pullme:v1")
    return store


def test_pull_roundtrip(served_store, tmp_path):
    srv = RegistryServer(served_store).start()
    client = RegistryClient(base_url=srv.url)
    dst = ImageStore(str(tmp_path / "dst-store"))
    try:
        ref = client.pull("example/pullme:v1", dst)
        assert ref == "docker.io/example/pullme:v1"
        img = dst.resolve("example/pullme:v1")
        assert img is not None
        assert img.config.entrypoint == ["/bin/tool"]
        assert "X=1" in img.config.env
        # content round-trips bit-exact (same manifest digest)
        src_img = served_store.resolve("example/pullme:v1")
        assert img.manifest_digest == src_img.manifest_digest
        rootfs = dst.rootfs_for(img)
        assert (rootfs / "etc" / "release").read_text() == "pulled\n"
    finally:
        client.close()
        srv.stop()


def test_pull_missing_image_404(served_store, tmp_path):
    srv = RegistryServer(served_store).start()
    client = RegistryClient(base_url=srv.url)
    try:
        with pytest.raises(RegistryError, match="not found"):
            client.pull("example/ghost:v1",
                        ImageStore(str(tmp_path / "d2")))
    finally:
        client.close()
        srv.stop()


def test_pull_requires_token_when_set(served_store, tmp_path):
    srv = RegistryServer(served_store, token="hunter2").start()
    dst = ImageStore(str(tmp_path / "d3"))
    noauth = RegistryClient(base_url=srv.url)
    try:
        with pytest.raises(RegistryError, match="unauthorized"):
            noauth.pull("example/pullme:v1", dst)
    finally:
        noauth.close()
    authed = RegistryClient(base_url=srv.url, token="hunter2")
    try:
        assert authed.pull("example/pullme:v1", dst)
    finally:
        authed.close()
        srv.stop()


def test_pull_detects_corrupted_blob(served_store, tmp_path):
    """A blob whose bytes don't match its digest must be rejected."""
    # corrupt one layer blob in the served layout
    img = served_store.resolve("example/pullme:v1")
    layer = img.layers[0]
    layer.write_bytes(layer.read_bytes() + b"tamper")
    srv = RegistryServer(served_store).start()
    client = RegistryClient(base_url=srv.url)
    try:
        with pytest.raises(RegistryError, match="digest mismatch"):
            client.pull("example/pullme:v1",
                        ImageStore(str(tmp_path / "d4")))
    finally:
        client.close()
        srv.stop()


def test_pull_by_digest(served_store, tmp_path):
    img = served_store.resolve("example/pullme:v1")
    srv = RegistryServer(served_store).start()
    client = RegistryClient(base_url=srv.url)
    dst = ImageStore(str(tmp_path / "d5"))
    try:
        ref = client.pull(f"example/pullme@{img.manifest_digest}", dst)
        pulled = dst.resolve(ref)
        assert pulled.manifest_digest == img.manifest_digest
    finally:
        client.close()
        srv.stop()


def test_imagetool_pull_cli(served_store, tmp_path):
    from k8s_runpod_kubelet_amd.runtime.imagetool import main

    srv = RegistryServer(served_store).start()
    try:
        rc = main(["--store", str(tmp_path / "d6"), "pull",
                   "example/pullme:v1", "--registry", srv.url])
        assert rc == 0
        dst = ImageStore(str(tmp_path / "d6"))
        assert dst.resolve("example/pullme:v1") is not None
    finally:
        srv.stop()


def test_imagetool_build_list_rm_gc_cli(tmp_path):
    """Full imagetool CLI lifecycle: build from a rootfs dir, list with
    sizes, rm, gc."""
    import io
    from contextlib import redirect_stdout

    from k8s_runpod_kubelet_amd.runtime.imagetool import main

    tree = tmp_path / "clitree"
    (tree / "bin").mkdir(parents=True)
    (tree / "bin" / "x").write_text("#!/bin/true\n")
    store_dir = str(tmp_path / "clistore")

    assert main(["--store", store_dir, "build", str(tree),
                 "--ref", "cli/app:v1", "--entrypoint", "/bin/x"]) == 0
    out = io.StringIO()
    with redirect_stdout(out):
        assert main(["--store", store_dir, "list"]) == 0
    assert "docker.io/cli/app:v1" in out.getvalue()

    store = ImageStore(store_dir)
    img = store.resolve("cli/app:v1")
    assert img.config.entrypoint == ["/bin/x"]
    store.rootfs_for(img)  # populate cache so gc has work after rm

    assert main(["--store", store_dir, "rm", "cli/app:v1"]) == 0
    assert store.resolve("cli/app:v1") is None
    assert main(["--store", store_dir, "rm", "cli/app:v1"]) == 1  # gone
    assert main(["--store", store_dir, "gc"]) == 0
    assert not any(store.rootfs_dir.iterdir()) \
        if store.rootfs_dir.is_dir() else True


def test_registry_server_head_and_blob_404(served_store):
    """Distribution API details: HEAD mirrors GET headers (clients probe
    with HEAD), unknown blobs 404, /v2/ pings."""
    import urllib.error
    import urllib.request

    srv = RegistryServer(served_store).start()
    try:
        base = srv.url
        img = served_store.resolve("example/pullme:v1")
        # ping
        with urllib.request.urlopen(f"{base}/v2/", timeout=5) as r:
            assert r.status == 200
        # HEAD manifest
        req = urllib.request.Request(
            f"{base}/v2/example/pullme/manifests/v1", method="HEAD")
        with urllib.request.urlopen(req, timeout=5) as r:
            assert r.status == 200
            assert int(r.headers["Content-Length"]) > 0
            assert r.read() == b""  # HEAD: headers only
        # GET manifest by digest
        with urllib.request.urlopen(
                f"{base}/v2/example/pullme/manifests/"
                f"{img.manifest_digest}", timeout=5) as r:
            assert r.status == 200
        # unknown blob 404s
        try:
            urllib.request.urlopen(
                f"{base}/v2/example/pullme/blobs/sha256:" + "0" * 64,
                timeout=5)
            raise AssertionError("unknown blob served")
        except urllib.error.HTTPError as exc:
            assert exc.code == 404
        # unknown tag 404s
        try:
            urllib.request.urlopen(
                f"{base}/v2/example/pullme/manifests/nope", timeout=5)
            raise AssertionError("unknown tag served")
        except urllib.error.HTTPError as exc:
            assert exc.code == 404
    finally:
        srv.stop()


def test_imagetool_import_cli(tmp_path):
    import tarfile

    from k8s_runpod_kubelet_amd.runtime.imagetool import main
    from k8s_runpod_kubelet_amd.runtime.oci import build_layout

    tree = tmp_path / "itree"
    (tree / "f").parent.mkdir(parents=True, exist_ok=True)
    (tree / "f").write_text("data")
    layout = tmp_path / "ilayout"
    layout.mkdir()
    build_layout(str(layout), "cli/imported:v3", str(tree))
    tar_path = tmp_path / "img.tar"
    with tarfile.open(tar_path, "w") as tf:
        for p in sorted(layout.rglob("*")):
            tf.add(p, arcname=str(p.relative_to(layout)), recursive=False)
    store_dir = str(tmp_path / "istore")
    assert main(["--store", store_dir, "import", str(tar_path)]) == 0
    store = ImageStore(store_dir)
    img = store.resolve("cli/imported:v3")
    assert img is not None
    assert (store.rootfs_for(img) / "f").read_text() == "data"


def test_push_roundtrip(served_store, tmp_path):
    """Push flow: a node publishes a local image to a push-enabled
    registry; another store pulls it back bit-exact."""
    hub_store = ImageStore(str(tmp_path / "hub"))
    hub = RegistryServer(hub_store, allow_push=True).start()
    client = RegistryClient(base_url=hub.url)
    try:
        digest = client.push("example/pullme:v1", served_store)
        assert digest == served_store.resolve(
            "example/pullme:v1").manifest_digest
        # the hub can serve it now
        assert "docker.io/example/pullme:v1" in hub_store.list_refs()
        dst = ImageStore(str(tmp_path / "dst"))
        ref = client.pull("example/pullme:v1", dst)
        img = dst.resolve(ref)
        assert img.manifest_digest == digest
        rootfs = dst.rootfs_for(img)
        assert (rootfs / "etc" / "release").read_text() == "pulled\n"
        # idempotent re-push (blobs deduped via HEAD)
        assert client.push("example/pullme:v1", served_store) == digest
    finally:
        client.close()
        hub.stop()


def test_push_refused_without_allow_push(served_store, tmp_path):
    hub = RegistryServer(ImageStore(str(tmp_path / "ro-hub"))).start()
    client = RegistryClient(base_url=hub.url)
    try:
        with pytest.raises(RegistryError, match="read-only|405"):
            client.push("example/pullme:v1", served_store)
    finally:
        client.close()
        hub.stop()


def test_push_requires_token(served_store, tmp_path):
    hub = RegistryServer(ImageStore(str(tmp_path / "t-hub")),
                         token="sesame", allow_push=True).start()
    bad = RegistryClient(base_url=hub.url, token="wrong")
    good = RegistryClient(base_url=hub.url, token="sesame")
    try:
        with pytest.raises(RegistryError, match="unauthorized"):
            bad.push("example/pullme:v1", served_store)
        assert good.push("example/pullme:v1", served_store)
    finally:
        bad.close()
        good.close()
        hub.stop()


def test_push_manifest_missing_blob_rejected(served_store, tmp_path):
    """Manifest referencing an un-uploaded blob -> 400 BLOB_UNKNOWN (the
    server refuses to register a partial image)."""
    import hashlib
    import json

    import httpx

    hub = RegistryServer(ImageStore(str(tmp_path / "p-hub")),
                         allow_push=True).start()
    try:
        manifest = {"schemaVersion": 2,
                    "mediaType": "application/vnd.oci.image.manifest.v1+json",
                    "config": {"digest": "sha256:" + "0" * 64, "size": 2},
                    "layers": []}
        r = httpx.put(f"{hub.url}/v2/example/ghost/manifests/v1",
                      content=json.dumps(manifest).encode())
        assert r.status_code == 400
        assert "BLOB_UNKNOWN" in r.text
    finally:
        hub.stop()


def test_imagetool_push_cli(served_store, tmp_path, capsys):
    from k8s_runpod_kubelet_amd.runtime.imagetool import main as itool

    hub_store_dir = str(tmp_path / "cli-hub")
    hub = RegistryServer(ImageStore(hub_store_dir), allow_push=True).start()
    try:
        rc = itool(["--store", str(served_store.root), "push",
                    "example/pullme:v1", "--registry", hub.url])
        assert rc == 0
        assert "pushed" in capsys.readouterr().out
        assert ImageStore(hub_store_dir).resolve(
            "example/pullme:v1") is not None
    finally:
        hub.stop()


def test_catalog_endpoint(served_store):
    import httpx

    srv = RegistryServer(served_store).start()
    try:
        r = httpx.get(f"{srv.url}/v2/_catalog")
        assert r.status_code == 200
        assert r.json() == {"repositories": ["example/pullme"]}
    finally:
        srv.stop()


def test_chunked_blob_upload(tmp_path):
    """Docker-style chunked upload: POST -> PATCH chunks -> PUT?digest=
    finalize; blob then serves."""
    import hashlib

    import httpx

    hub = RegistryServer(ImageStore(str(tmp_path / "c-hub")),
                         allow_push=True).start()
    try:
        data = b"x" * 10_000 + b"y" * 10_000
        digest = "sha256:" + hashlib.sha256(data).hexdigest()
        start = httpx.post(f"{hub.url}/v2/example/chunky/blobs/uploads/")
        assert start.status_code == 202
        loc = hub.url + start.headers["Location"]
        r1 = httpx.patch(loc, content=data[:10_000])
        assert r1.status_code == 202
        assert r1.headers["Range"] == "0-9999"
        r2 = httpx.patch(loc, content=data[10_000:])
        assert r2.status_code == 202
        fin = httpx.put(f"{loc}?digest={digest}")
        assert fin.status_code == 201
        assert fin.headers["Docker-Content-Digest"] == digest
        # server finds it content-addressed for manifest validation
        assert hub._find_blob_file(digest).read_bytes() == data
        # wrong-digest finalize is refused
        start2 = httpx.post(f"{hub.url}/v2/example/chunky/blobs/uploads/")
        loc2 = hub.url + start2.headers["Location"]
        httpx.patch(loc2, content=b"garbage")
        bad = httpx.put(f"{loc2}?digest={digest}")
        assert bad.status_code == 400
    finally:
        hub.stop()


def test_concurrent_pushes_and_pulls(served_store, tmp_path):
    """Thread-safety: N clients pushing distinct tags while others pull —
    the hub ends with every tag resolvable and bit-exact manifests."""
    import threading

    hub_store = ImageStore(str(tmp_path / "mt-hub"))
    hub = RegistryServer(hub_store, allow_push=True).start()
    src_digest = served_store.resolve("example/pullme:v1").manifest_digest
    errors = []

    def push_one(i):
        c = RegistryClient(base_url=hub.url)
        try:
            # same content under a unique tag (content-addressed dedup)
            layout = served_store._layout_for("example/pullme:v1")
            dst = ImageStore(str(tmp_path / f"src-{i}"))
            dst.add_layout(str(layout), f"example/mt-{i}:v1")
            c.push(f"example/mt-{i}:v1", dst)
        except Exception as exc:  # noqa: BLE001
            errors.append(f"push{i}: {exc}")
        finally:
            c.close()

    threads = [threading.Thread(target=push_one, args=(i,))
               for i in range(6)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=30)
    assert not errors, errors
    for i in range(6):
        img = hub_store.resolve(f"example/mt-{i}:v1")
        assert img is not None
        assert img.manifest_digest == src_digest

    def pull_one(i):
        c = RegistryClient(base_url=hub.url)
        try:
            dst = ImageStore(str(tmp_path / f"pl-{i}"))
            c.pull(f"example/mt-{i}:v1", dst)
            assert dst.resolve(
                f"example/mt-{i}:v1").manifest_digest == src_digest
        except Exception as exc:  # noqa: BLE001
            errors.append(f"pull{i}: {exc}")
        finally:
            c.close()

    threads = [threading.Thread(target=pull_one, args=(i,))
               for i in range(6)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=30)
    assert not errors, errors
    hub.stop()


def test_retagged_layout_served_under_store_ref(served_store, tmp_path):
    """A layout re-registered under a new tag (add_layout) serves under
    that tag: the store's .amdvk-ref is authoritative over the layout's
    internal annotation (bug found by the img-pull soak mode: every
    re-tagged pull 404'd)."""
    layout = served_store._layout_for("example/pullme:v1")
    served_store.add_layout(str(layout), "example/retag:v7")
    srv = RegistryServer(served_store).start()
    client = RegistryClient(base_url=srv.url)
    try:
        dst = ImageStore(str(tmp_path / "rt-dst"))
        ref = client.pull("example/retag:v7", dst)
        assert ref == "docker.io/example/retag:v7"
        assert dst.resolve("example/retag:v7") is not None
    finally:
        client.close()
        srv.stop()


def test_large_blob_streams_roundtrip(tmp_path):
    """GB-scale layer posture: a 48 MiB layer pushes and pulls bit-exact
    through the streaming paths (server chunked GET, client stream-to-disk
    with on-the-fly hashing, file-object upload)."""
    import hashlib
    import io
    import tarfile

    # build an image whose single layer contains a 48 MiB file
    tree = tmp_path / "big-tree"
    tree.mkdir()
    payload = bytes(range(256)) * (48 * 1024 * 4)  # 48 MiB patterned
    (tree / "blob.bin").write_bytes(payload)
    layout = tmp_path / "big-layout"
    layout.mkdir()
    build_layout(str(layout), "example/big:v1", str(tree))
    src = ImageStore(str(tmp_path / "big-src"))
    src.add_layout(str(layout), "example/big:v1")

    hub_store = ImageStore(str(tmp_path / "big-hub"))
    hub = RegistryServer(hub_store, allow_push=True).start()
    client = RegistryClient(base_url=hub.url)
    try:
        digest = client.push("example/big:v1", src)
        dst = ImageStore(str(tmp_path / "big-dst"))
        client.pull("example/big:v1", dst)
        img = dst.resolve("example/big:v1")
        assert img.manifest_digest == digest
        rootfs = dst.rootfs_for(img)
        assert hashlib.sha256(
            (rootfs / "blob.bin").read_bytes()).digest() == \
            hashlib.sha256(payload).digest()
    finally:
        client.close()
        hub.stop()


def test_imagetool_inspect_cli(served_store, capsys):
    import json as _json

    from k8s_runpod_kubelet_amd.runtime.imagetool import main as itool

    rc = itool(["--store", str(served_store.root), "inspect",
                "example/pullme:v1"])
    assert rc == 0
    out = _json.loads(capsys.readouterr().out)
    assert out["entrypoint"] == ["/bin/tool"]
    assert out["digest"].startswith("sha256:")
    assert out["layers"] and out["layers"][0]["digest"].startswith("sha256:")
    assert itool(["--store", str(served_store.root), "inspect",
                  "nope:v9"]) == 1


def test_server_survives_hostile_paths(served_store):
    """Robustness: arbitrary request paths never crash the server or leak
    files — everything unexpected is a clean 4xx."""
    import httpx

    srv = RegistryServer(served_store).start()
    try:
        hostile = [
            "/", "/v2/../../etc/passwd", "/v2//blobs/sha256:zz",
            "/v2/a/manifests/", "/v2/a/blobs/sha256:" + "g" * 64,
            "/v2/a/blobs/uploads/../../x", "/v2/%2e%2e/manifests/v1",
            "/v2/a/manifests/" + "A" * 5000,
        ]
        for path in hostile:
            r = httpx.get(srv.url + path)
            assert 400 <= r.status_code < 500, (path, r.status_code)
        # push endpoints refused cleanly when push disabled
        assert httpx.post(
            srv.url + "/v2/a/blobs/uploads/").status_code == 405
        assert httpx.put(
            srv.url + "/v2/a/manifests/v1", content=b"{}").status_code == 405
        # server still serves normally afterwards
        assert httpx.get(srv.url + "/v2/").status_code == 200
    finally:
        srv.stop()
