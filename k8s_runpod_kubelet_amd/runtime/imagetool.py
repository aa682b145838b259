"""Operator CLI for the node-local OCI image store (no registry on this
node — images arrive as OCI layouts / oci-archive tars, e.g. produced by
``skopeo copy docker://rocm/pytorch oci-archive:img.tar``).

    python -m k8s_runpod_kubelet_amd.runtime.imagetool list
    python -m k8s_runpod_kubelet_amd.runtime.imagetool import img.tar [--ref R]
    python -m k8s_runpod_kubelet_amd.runtime.imagetool add-layout DIR --ref R
    python -m k8s_runpod_kubelet_amd.runtime.imagetool build DIR --ref R \
        [--entrypoint CMD...] [--env K=V...]
"""

from __future__ import annotations

import argparse
import sys
import tempfile

from ..config import Config
from .oci import ImageStore, build_layout


def main(argv=None) -> int:
    parser = argparse.ArgumentParser(prog="amdvk-image")
    parser.add_argument("--store", default="",
                        help="image store dir (default: config state_dir/images)")
    sub = parser.add_subparsers(dest="cmd", required=True)

    sub.add_parser("list", help="list stored image references")

    p_rm = sub.add_parser("rm", help="remove an image reference")
    p_rm.add_argument("ref")

    sub.add_parser("gc", help="reclaim unpacked-rootfs cache space for "
                              "removed images")

    p_imp = sub.add_parser("import", help="import an oci-archive tar")
    p_imp.add_argument("tar")
    p_imp.add_argument("--ref", default="",
                       help="reference (default: from the archive annotation)")

    p_add = sub.add_parser("add-layout", help="register an OCI layout dir")
    p_add.add_argument("dir")
    p_add.add_argument("--ref", required=True)

    p_pull = sub.add_parser("pull", help="pull from an OCI registry")
    p_pull.add_argument("ref")
    p_pull.add_argument("--registry", default="",
                        help="registry base URL override, e.g. http://host:5000")
    p_pull.add_argument("--token", default="")
    p_pull.add_argument("--insecure", action="store_true")

    p_i = sub.add_parser("inspect", help="print an image's config and "
                         "layer digests as JSON")
    p_i.add_argument("ref")

    p_push = sub.add_parser("push", help="push a stored image to a "
                            "registry (server needs --allow-push)")
    p_push.add_argument("ref")
    p_push.add_argument("--registry", default="",
                        help="registry base URL override, e.g. http://host:5000")
    p_push.add_argument("--token", default="")
    p_push.add_argument("--insecure", action="store_true")

    p_b = sub.add_parser("build", help="build an image from a rootfs dir")
    p_b.add_argument("rootfs")
    p_b.add_argument("--ref", required=True)
    p_b.add_argument("--entrypoint", nargs="+", default=[])
    p_b.add_argument("--cmd", dest="image_cmd", nargs="+",
                     default=[])
    p_b.add_argument("--env", nargs="*", default=[])
    p_b.add_argument("--workdir", default="")
    p_b.add_argument("--user", default="")

    args = parser.parse_args(argv)
    store = ImageStore(args.store or Config().resolved_image_store_dir())

    if args.cmd == "list":
        for ref, size in store.image_sizes():
            print(f"{ref}\t{size >> 20} MiB")
        return 0
    if args.cmd == "rm":
        if not store.remove(args.ref):
            print(f"not found: {args.ref}", file=sys.stderr)
            return 1
        removed = store.gc()
        print(f"removed {args.ref} (+{len(removed)} cache entries)")
        return 0
    if args.cmd == "gc":
        removed = store.gc()
        print(f"removed {len(removed)} unreferenced cache entries")
        return 0
    if args.cmd == "import":
        ref = store.import_archive(args.tar, args.ref)
        print(f"imported {ref}")
        return 0
    if args.cmd == "add-layout":
        ref = store.add_layout(args.dir, args.ref)
        print(f"added {ref}")
        return 0
    if args.cmd == "pull":
        from .registry import RegistryClient

        client = RegistryClient(base_url=args.registry, token=args.token,
                                verify=not args.insecure)
        try:
            ref = client.pull(args.ref, store)
        finally:
            client.close()
        print(f"pulled {ref}")
        return 0
    if args.cmd == "inspect":
        import json as _json

        img = store.resolve(args.ref)
        if img is None:
            print(f"not found: {args.ref}", file=sys.stderr)
            return 1
        manifest = _json.loads(
            (img.layout_dir / "blobs" /
             img.manifest_digest.replace(":", "/")).read_bytes())
        print(_json.dumps({
            "ref": img.ref,
            "digest": img.manifest_digest,
            "entrypoint": img.config.entrypoint,
            "cmd": img.config.cmd,
            "env": img.config.env,
            "user": img.config.user,
            "workingDir": img.config.working_dir,
            "layers": [{"digest": l.get("digest", ""),
                        "size": l.get("size", 0)}
                       for l in manifest.get("layers", [])],
        }, indent=2))
        return 0
    if args.cmd == "push":
        from .registry import RegistryClient

        client = RegistryClient(base_url=args.registry, token=args.token,
                                verify=not args.insecure)
        try:
            digest = client.push(args.ref, store)
        finally:
            client.close()
        print(f"pushed {args.ref} ({digest})")
        return 0
    if args.cmd == "build":
        with tempfile.TemporaryDirectory(prefix="amdvk-build-") as td:
            build_layout(td, args.ref, args.rootfs,
                         entrypoint=args.entrypoint, cmd=args.image_cmd,
                         env=args.env or None, working_dir=args.workdir,
                         user=args.user)
            ref = store.add_layout(td, args.ref)
        print(f"built {ref}")
        return 0
    return 2


if __name__ == "__main__":
    sys.exit(main())
