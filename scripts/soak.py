#!/usr/bin/env python3
"""Continuous-churn robustness soak for the full stack.

Runs the production path — FakeApiServer (HTTP) ← HttpK8sClient ← informer
← PodController ← Provider ← ProcessRuntime — under randomized pod churn
for --duration seconds: a rolling population of 1-GPU (or CPU) pods in
mixed modes (hold-then-delete, run-to-completion success/failure,
out-of-band SIGKILL), then checks for leaks and prints one JSON summary.

    python scripts/soak.py --duration 240          # GPU box
    python scripts/soak.py --duration 20 --cpu     # dev container
"""

from __future__ import annotations

import argparse
import json
import os
import random
import signal
import sys
import tempfile
import time

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO_ROOT)


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--duration", type=float, default=240.0)
    ap.add_argument("--max-active", type=int, default=6)
    ap.add_argument("--cpu", action="store_true",
                    help="synthetic inventory + CPU podworker")
    ap.add_argument("--seed", type=int, default=12345)
    ap.add_argument("--restart-kubelet-at", type=float, default=0.0,
                    help="simulate a kubelet crash+restart T seconds in "
                         "(pods untouched; adoption must resume everything)")
    ap.add_argument("--reap-orphans", action="store_true",
                    help="at startup, SIGKILL podworker processes leaked "
                         "by previous soaks that were killed outright "
                         "(exact exe match only)")
    ap.add_argument("--images", action="store_true",
                    help="mix OCI image pods (overlay+pivot_root) into the "
                         "churn (needs mount-ns capability and gcc)")
    args = ap.parse_args()
    rng = random.Random(args.seed)

    if args.reap_orphans:
        from k8s_runpod_kubelet_amd.ops import podworker_binary

    target = None
    if args.reap_orphans:
        target = os.path.realpath(podworker_binary())
        reaped = 0
        for pid in os.listdir("/proc"):
            if not pid.isdigit() or int(pid) == os.getpid():
                continue
            try:
                exe = os.readlink(f"/proc/{pid}/exe")
            except OSError:
                continue
            if exe in (target, target + " (deleted)"):
                try:
                    os.kill(int(pid), signal.SIGKILL)
                    reaped += 1
                except OSError:
                    pass
        if reaped:
            print(f"reaped {reaped} leaked podworker(s)", file=sys.stderr)

    from k8s_runpod_kubelet_amd.app import build_stack
    from k8s_runpod_kubelet_amd.config import Config
    from k8s_runpod_kubelet_amd.kube.client import NotFoundError
    from k8s_runpod_kubelet_amd.kube.fake_apiserver import FakeApiServer
    from k8s_runpod_kubelet_amd.kube.real import ClusterConfig, HttpK8sClient
    from k8s_runpod_kubelet_amd.provider import annotations as ann
    from k8s_runpod_kubelet_amd.server import metrics as m

    # an outer `timeout` kills with SIGTERM: a flag (not an exception —
    # raising from a handler can interrupt lock-held teardown and
    # deadlock) makes the main loop fall through to the finally-block
    # reap instead of orphaning pod processes
    stop_requested = []
    signal.signal(signal.SIGTERM,
                  lambda *_: stop_requested.append(True))

    srv = FakeApiServer().start()
    client = HttpK8sClient(ClusterConfig(server=srv.url))
    cfg = Config(
        state_dir=tempfile.mkdtemp(prefix="amdvk-soak-"),
        gpu_count_override=(8 if args.cpu else -1),
        pending_retry_interval_s=0.5,
        notify_interval_s=0,
        pod_controller_workers=8,
    )
    image_ref = ""
    if args.images:
        import shutil
        import subprocess

        from k8s_runpod_kubelet_amd.ops import load_native
        from k8s_runpod_kubelet_amd.runtime.oci import ImageStore, build_layout

        if not load_native().probe_mount_namespace():
            print("--images needs mount-namespace capability", file=sys.stderr)
            return 2
        bdir = tempfile.mkdtemp(prefix="amdvk-soakimg-")
        with open(os.path.join(bdir, "app.c"), "w") as fh:
            fh.write(
                '#include <unistd.h>\n#include <stdio.h>\n'
                '#include <string.h>\n'
                'int main(int c, char** v) {\n'
                '  printf("img-ready\\n"); fflush(stdout);\n'
                '  if (c > 1 && !strcmp(v[1], "hold")) pause();\n'
                '  return 0;\n}\n')
        subprocess.run(["gcc", "-static", "-O1", "-o",
                        os.path.join(bdir, "app"),
                        os.path.join(bdir, "app.c")], check=True)
        tree = os.path.join(bdir, "tree")
        os.makedirs(os.path.join(tree, "bin"))
        shutil.copy2(os.path.join(bdir, "app"),
                     os.path.join(tree, "bin", "app"))
        layout = os.path.join(bdir, "layout")
        os.makedirs(layout)
        build_layout(layout, "soak/img:v1", tree, entrypoint=["/bin/app"])
        ImageStore(cfg.resolved_image_store_dir()).add_layout(
            layout, "soak/img:v1")
        image_ref = "soak/img:v1"
        # registry pull-on-miss under churn: a hub serves the same content
        # under many unique tags; img-pull pods each force a live pull
        # into the kubelet's store (concurrent deploy workers pulling)
        from k8s_runpod_kubelet_amd.runtime.registry_server import (
            RegistryServer,
        )

        hub_store = ImageStore(os.path.join(bdir, "hub-store"))
        for i in range(64):
            hub_store.add_layout(layout, f"soak/pull-{i}:v1")
        hub = RegistryServer(hub_store).start()
        cfg.image_registry = hub.url

    stack = build_stack(cfg, client=client)
    stack.start(serve_http=False)

    gpus = 0 if args.cpu else 1
    pw_args = (["--expect-gpus", "1"] if gpus else [])

    def make(name, mode):
        if mode == "img-pull":
            # unique tag -> guaranteed store miss -> in-kubelet pull
            tag = f"soak/pull-{rng.randrange(64)}:v1"
            return {
                "apiVersion": "v1", "kind": "Pod",
                "metadata": {"name": name, "namespace": "default"},
                "spec": {"nodeName": cfg.node_name,
                         "restartPolicy": "Never",
                         "containers": [{"name": "main", "image": tag}]},
            }
        if mode == "img-ok":
            # image pod, no command: the image entrypoint runs in its rootfs
            return {
                "apiVersion": "v1", "kind": "Pod",
                "metadata": {"name": name, "namespace": "default"},
                "spec": {"nodeName": cfg.node_name,
                         "restartPolicy": "Never",
                         "containers": [{"name": "main",
                                         "image": image_ref}]},
            }
        if mode == "img-hold":
            # pid-1-in-namespace ignores default-action TERM: the delete
            # path exercises the grace->SIGKILL ladder AND the
            # object-visible-until-dead semantics every time; short grace
            # keeps the churn brisk
            return {
                "apiVersion": "v1", "kind": "Pod",
                "metadata": {"name": name, "namespace": "default"},
                "spec": {"nodeName": cfg.node_name,
                         "restartPolicy": "Never",
                         "terminationGracePeriodSeconds": 2,
                         "containers": [{"name": "main", "image": image_ref,
                                         "command": ["/bin/app"],
                                         "args": ["hold"]}]},
            }
        if mode in ("hold", "crash", "probed", "started", "hooked"):
            a = pw_args + ["--hold"]
        elif mode == "ok":
            a = pw_args + ["--run-for", f"{rng.uniform(0.05, 0.4):.2f}"]
        elif mode == "restarting":
            a = pw_args + ["--run-for", "0.05", "--exit-code", "9"]
        else:  # fail
            a = pw_args + ["--run-for", "0.05", "--exit-code", "7"]
        container = {
            "name": "main", "image": "amdvk/podworker:soak",
            "command": ["podworker"], "args": a,
            **({"resources": {"limits": {"amd.com/gpu": str(gpus)}}}
               if gpus else {}),
        }
        spec = {"nodeName": cfg.node_name, "containers": [container],
                "restartPolicy": "Never"}
        if mode == "probed":
            container["readinessProbe"] = {
                "exec": {"command": ["/bin/true"]},
                "periodSeconds": 1, "failureThreshold": 1,
            }
        elif mode == "started":
            container["startupProbe"] = {
                "exec": {"command": ["/bin/true"]},
                "periodSeconds": 1, "failureThreshold": 3,
            }
        elif mode == "hooked":
            container["lifecycle"] = {
                "postStart": {"exec": {"command": ["/bin/true"]}},
                "preStop": {"sleep": {"seconds": 0.1}},
            }
        elif mode == "restarting":
            spec["restartPolicy"] = "OnFailure"
        return {
            "apiVersion": "v1", "kind": "Pod",
            "metadata": {"name": name, "namespace": "default"},
            "spec": spec,
        }

    def pod_state(name):
        try:
            p = client.get_pod("default", name)
        except NotFoundError:
            return "Gone", None
        conds = {c["type"]: c["status"]
                 for c in p.get("status", {}).get("conditions", [])}
        phase = p.get("status", {}).get("phase", "")
        if conds.get("Ready") == "True":
            return "Ready", p
        return phase or "Pending", p

    def fd_count():
        return len(os.listdir("/proc/self/fd"))

    fd0 = fd_count()
    active = {}  # name -> dict(mode, created, state)
    counters = {"created": 0, "succeeded": 0, "failed": 0, "crashed": 0,
                "deleted_holds": 0, "restarted": 0, "timeouts": 0}
    seq = 0
    start_ts = time.time()
    deadline = start_ts + args.duration
    kubelet_restarts = 0
    try:
        while (time.time() < deadline or active) and not stop_requested:
            now = time.time()
            if (args.restart_kubelet_at > 0 and kubelet_restarts == 0
                    and now - start_ts >= args.restart_kubelet_at):
                # Simulated kubelet crash: stop every control-plane thread,
                # leave pod processes running, then build a fresh stack on
                # the same state dir — adoption must pick everything up.
                print("KUBELET RESTART", file=sys.stderr)
                stack.pod_controller.stop()
                stack.node_controller.stop()
                stack.provider.stop()
                stack.runtime.close()
                old = stack
                stack = build_stack(cfg, client=client)
                stack.start(serve_http=False)
                kubelet_restarts += 1
                # A real crash ends the process and the kernel reaps its
                # fds; emulate that for the fd-leak check by collecting the
                # dead stack's objects (EventLoop epfd/pidfds close on GC).
                del old
                import gc
                gc.collect()
            # top up population (only while inside the window)
            while now < deadline and len(active) < args.max_active:
                modes = ["hold", "ok", "fail", "crash", "probed",
                         "restarting", "started", "hooked"]
                weights = [3, 4, 2, 1, 2, 1, 1, 1]
                if image_ref:
                    modes += ["img-ok", "img-hold", "img-pull"]
                    weights += [3, 2, 2]
                mode = rng.choices(modes, weights=weights)[0]
                name = f"soak-{seq:05d}"
                seq += 1
                client.create_pod("default", make(name, mode))
                active[name] = {"mode": mode, "created": now,
                                "dwell": rng.uniform(0.2, 1.5),
                                "killed": False, "deleted": False}
                counters["created"] += 1
            for name, st in list(active.items()):
                state, pod = pod_state(name)
                age = now - st["created"]
                if state == "Gone":
                    del active[name]
                    continue
                if age > 120:
                    counters["timeouts"] += 1
                    print(f"TIMEOUT: {name} mode={st['mode']} state={state} "
                          f"pod={json.dumps((pod or {}).get('status', {}))}",
                          file=sys.stderr)
                    for inst in stack.runtime.list_instances():
                        print(f"  inst {inst.id} {inst.desired_status} "
                              f"{inst.namespace}/{inst.name}", file=sys.stderr)
                    print(f"  reservations={list(stack.ledger.reservations)}",
                          file=sys.stderr)
                    try:
                        client.delete_pod("default", name)
                    except NotFoundError:
                        pass
                    del active[name]
                    continue
                if st["mode"] in ("ok", "fail", "img-ok", "img-pull"):
                    want = ("Failed" if st["mode"] == "fail"
                            else "Succeeded")
                    if state == want and not st["deleted"]:
                        counters["succeeded" if want == "Succeeded"
                                 else "failed"] += 1
                        st["deleted"] = True
                        client.delete_pod("default", name)
                elif st["mode"] == "crash":
                    if state == "Ready" and not st["killed"]:
                        iid = pod["metadata"]["annotations"].get(ann.POD_ID)
                        det = stack.runtime.get_detailed_status(iid)
                        if det.containers:
                            os.kill(det.containers[0].pid, 9)
                            st["killed"] = True
                    elif state == "Failed" and not st["deleted"]:
                        counters["crashed"] += 1
                        st["deleted"] = True
                        client.delete_pod("default", name)
                elif st["mode"] == "restarting":
                    css = (pod or {}).get("status", {}).get(
                        "containerStatuses", [])
                    if css and css[0].get("restartCount", 0) >= 1 \
                            and not st["deleted"]:
                        counters["restarted"] += 1
                        st["deleted"] = True
                        client.delete_pod("default", name)
                else:  # hold / probed / started / hooked / img-hold
                    if (st["mode"] == "hooked" and state == "Ready"
                            and not st.get("debugged")):
                        # kubectl-debug under churn: attach an ephemeral
                        # container to a fraction of hooked pods
                        st["debugged"] = True
                        if rng.random() < 0.5:
                            try:
                                p2 = client.get_pod("default", name)
                                p2["spec"]["ephemeralContainers"] = [{
                                    "name": "dbg",
                                    "image": "amdvk/debug:soak",
                                    "command": ["/bin/sh"],
                                    "args": ["-c", "echo dbg; exit 0"],
                                }]
                                client.update_pod("default", p2)
                            except NotFoundError:
                                pass
                    if state == "Ready" and age > st["dwell"] and not st["deleted"]:
                        counters["deleted_holds"] += 1
                        st["deleted"] = True
                        client.delete_pod("default", name)
            time.sleep(0.02)

        stack.provider.cleanup_deleted_pods()
        leftovers = stack.runtime.list_instances()
        summary = {
            "duration_s": args.duration,
            "mode": "cpu" if args.cpu else "gpu",
            "kubelet_restarts": kubelet_restarts,
            **counters,
            "pods_done": counters["succeeded"] + counters["failed"]
            + counters["crashed"] + counters["deleted_holds"]
            + counters["restarted"],
            "mean_ready_ms": round(m.hist_mean_ms(m.pod_ready_seconds), 1),
            "leaks": {
                "reservations": len(stack.ledger.reservations),
                "tracked_processes": stack.runtime.tracked_process_count(),
                "instances": len(leftovers),
                "fds_delta": fd_count() - fd0,
            },
        }
        ok = (counters["timeouts"] == 0
              and summary["leaks"]["reservations"] == 0
              and summary["leaks"]["tracked_processes"] == 0
              and summary["leaks"]["instances"] == 0
              and summary["leaks"]["fds_delta"] < 20)
        summary["ok"] = ok
        print(json.dumps(summary))
        return 0 if ok else 1
    finally:
        # Reap every pod process before exiting: pods are DESIGNED to
        # outlive a kubelet (adoption), but a finished soak must not
        # leave its workloads running on the host (leaked holds from
        # timeout-killed soaks once stole a fixed test port).
        try:
            for inst in stack.runtime.list_instances():
                try:
                    stack.runtime.terminate(inst.id, grace_override_s=0.0)
                except Exception:
                    pass
        except Exception:
            pass
        stack.stop()
        client.close()
        srv.stop()


if __name__ == "__main__":
    sys.exit(main())
