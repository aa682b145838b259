"""Property-based tests (hypothesis) for the pure-logic layers.

The reference has no property testing at all (integration-first,
no hermetic suite); these pin down invariants of the strategic-merge patch
engine, env escaping, duration parsing and port gating that example-based
tests under-cover."""

import string

from hypothesis import given, settings
from hypothesis import strategies as st

from k8s_runpod_kubelet_amd.kube.patch import json_merge, strategic_merge
from k8s_runpod_kubelet_amd.provider.envvars import _escape, is_k8s_auto_injected
from k8s_runpod_kubelet_amd.provider.ports import check_ports_exposed
from k8s_runpod_kubelet_amd.utils.backoff import parse_duration_s

# -- strategies --------------------------------------------------------------

scalars = st.one_of(
    st.integers(-1000, 1000),
    st.text(string.ascii_letters, max_size=8),
    st.booleans(),
)
json_maps = st.recursive(
    st.dictionaries(st.text(string.ascii_lowercase, min_size=1, max_size=6),
                    scalars, max_size=4),
    lambda children: st.dictionaries(
        st.text(string.ascii_lowercase, min_size=1, max_size=6),
        st.one_of(scalars, children), max_size=4),
    max_leaves=12,
)


# -- strategic merge ---------------------------------------------------------

@settings(max_examples=200)
@given(original=json_maps, patch=json_maps)
def test_patch_values_win(original, patch):
    """Every key present in the patch ends up with the patch's value (or is
    deleted when the patch value is None); keys absent from the patch keep
    their original value."""
    merged = strategic_merge(original, patch)
    for key, value in patch.items():
        if value is None:
            assert key not in merged
        elif isinstance(value, dict) and isinstance(original.get(key), dict):
            inner = strategic_merge(original[key], value)
            assert merged[key] == inner
        else:
            assert merged[key] == value
    for key, value in original.items():
        if key not in patch:
            assert merged[key] == value


@settings(max_examples=100)
@given(original=json_maps, patch=json_maps)
def test_merge_is_idempotent(original, patch):
    once = strategic_merge(original, patch)
    twice = strategic_merge(once, patch)
    assert once == twice


@settings(max_examples=100)
@given(original=json_maps, patch=json_maps)
def test_merge_does_not_mutate_inputs(original, patch):
    import copy

    o2, p2 = copy.deepcopy(original), copy.deepcopy(patch)
    strategic_merge(original, patch)
    assert original == o2 and patch == p2


@settings(max_examples=100)
@given(original=json_maps, patch=json_maps)
def test_json_merge_agrees_on_flat_scalars(original, patch):
    """For scalar-only patches RFC 7386 json-merge and SMP agree."""
    flat_patch = {k: v for k, v in patch.items() if not isinstance(v, dict)}
    a = strategic_merge(original, flat_patch)
    b = json_merge(original, flat_patch)
    for key in flat_patch:
        assert a.get(key) == b.get(key)


conditions = st.lists(
    st.fixed_dictionaries({
        "type": st.sampled_from(["Ready", "PodScheduled", "Initialized"]),
        "status": st.sampled_from(["True", "False", "Unknown"]),
    }),
    max_size=4,
    unique_by=lambda c: c["type"],
)


@settings(max_examples=200)
@given(original=conditions, patch=conditions)
def test_conditions_merge_by_type(original, patch):
    merged = strategic_merge({"conditions": original},
                             {"conditions": patch})["conditions"]
    by_type = {c["type"]: c for c in merged}
    # no duplicate types after merge
    assert len(by_type) == len(merged)
    for cond in patch:  # patched conditions win
        assert by_type[cond["type"]]["status"] == cond["status"]
    for cond in original:  # untouched conditions survive
        if cond["type"] not in {c["type"] for c in patch}:
            assert by_type[cond["type"]]["status"] == cond["status"]


# -- env escaping ------------------------------------------------------------

@settings(max_examples=200)
@given(st.text(max_size=64))
def test_env_escape_removes_raw_newlines(value):
    escaped = _escape(value)
    assert "\n" not in escaped.replace("\\n", "")


@given(st.text(string.ascii_uppercase + "_", min_size=1, max_size=20))
def test_auto_injected_filter_matches_reference_patterns(name):
    """isK8sAutoInjectedVar semantics (reference runpod_client.go:886-904):
    substring patterns, not prefixes."""
    flagged = is_k8s_auto_injected(name)
    expected = name.startswith("KUBERNETES_") or any(
        p in name for p in ("_PORT_", "_TCP_", "_SERVICE_PORT_", "_SERVICE_HOST")
    ) or name.endswith("_PORT") or name.endswith("_SERVICE_HOST")
    # our filter must never let a k8s-injected pattern through
    if expected:
        assert flagged
    if not flagged:
        assert not expected


# -- duration parsing --------------------------------------------------------

@given(st.integers(0, 10**6))
def test_parse_duration_seconds(n):
    assert parse_duration_s(f"{n}s", -1) == float(n)
    assert parse_duration_s(str(n), -1) == float(n)


@given(st.integers(0, 10**4))
def test_parse_duration_minutes_hours(n):
    assert parse_duration_s(f"{n}m", -1) == float(n * 60)
    assert parse_duration_s(f"{n}h", -1) == float(n * 3600)


@given(st.text(max_size=12))
def test_parse_duration_never_raises(junk):
    """Malformed durations fall back to the default instead of crashing the
    CLI (bug found by this property: float('abc') used to propagate)."""
    out = parse_duration_s(junk, 42.0)
    assert isinstance(out, float)


# -- port gating -------------------------------------------------------------

ports = st.integers(1, 65535)


@settings(max_examples=200)
@given(requested=st.lists(ports, max_size=6, unique=True),
       listening=st.sets(ports, max_size=8))
def test_tcp_ports_gate_readiness(requested, listening):
    """TCP ports must all be exposed; HTTP ports are assumed proxied
    (reference kubelet.go:566-605 semantics)."""
    reqs = [f"{p}/tcp" for p in requested]
    mappings = {p: p for p in listening}
    ok = check_ports_exposed(reqs, mappings)
    assert ok == all(p in listening for p in requested)


@settings(max_examples=50)
@given(requested=st.lists(ports, max_size=6, unique=True))
def test_http_ports_always_ready(requested):
    reqs = [f"{p}/http" for p in requested]
    assert check_ports_exposed(reqs, {})


# -- spec-translation fuzz ---------------------------------------------------

_names = st.text(string.ascii_lowercase + "-", min_size=1, max_size=10)
_maybe_int = st.one_of(st.none(), st.integers(-5, 500),
                       st.text(string.digits + "GiB", max_size=6))
_container = st.fixed_dictionaries({}, optional={
    "name": _names,
    "image": st.text(max_size=12),
    "command": st.lists(st.text(max_size=8), max_size=3),
    "args": st.lists(st.text(max_size=8), max_size=3),
    "workingDir": st.text(max_size=10),
    "ports": st.lists(st.fixed_dictionaries({}, optional={
        "containerPort": _maybe_int,
        "protocol": st.sampled_from(["TCP", "UDP", "SCTP", ""]),
    }), max_size=3),
    "env": st.lists(st.fixed_dictionaries({}, optional={
        "name": st.text(max_size=10),
        "value": st.text(max_size=10),
        "valueFrom": st.fixed_dictionaries({}, optional={
            "secretKeyRef": st.fixed_dictionaries(
                {}, optional={"name": _names, "key": st.text(max_size=6)}),
            "configMapKeyRef": st.fixed_dictionaries(
                {}, optional={"name": _names, "key": st.text(max_size=6)}),
            "fieldRef": st.fixed_dictionaries(
                {}, optional={"fieldPath": st.text(max_size=20)}),
        }),
    }), max_size=3),
    "resources": st.fixed_dictionaries({}, optional={
        "limits": st.dictionaries(
            st.sampled_from(["amd.com/gpu", "cpu", "memory", "junk"]),
            st.one_of(st.text(max_size=6), st.integers(0, 16)), max_size=3),
    }),
    "securityContext": st.fixed_dictionaries({}, optional={
        "runAsUser": _maybe_int, "runAsGroup": _maybe_int}),
    "livenessProbe": st.fixed_dictionaries({}, optional={
        "tcpSocket": st.fixed_dictionaries({}, optional={"port": _maybe_int}),
        "exec": st.fixed_dictionaries({}, optional={
            "command": st.lists(st.text(max_size=6), max_size=2)}),
        "periodSeconds": _maybe_int,
    }),
})
_podspec = st.fixed_dictionaries({
    "containers": st.lists(_container, max_size=3),
}, optional={
    "initContainers": st.lists(_container, max_size=2),
    "restartPolicy": st.sampled_from(["Always", "OnFailure", "Never", "??"]),
    "hostname": st.text(max_size=10),
    "terminationGracePeriodSeconds": _maybe_int,
    "activeDeadlineSeconds": _maybe_int,
    "securityContext": st.fixed_dictionaries({}, optional={
        "runAsUser": _maybe_int}),
})
_fuzz_pod = st.fixed_dictionaries({
    "metadata": st.fixed_dictionaries({
        "name": _names,
        "namespace": _names,
    }, optional={
        "annotations": st.dictionaries(
            st.sampled_from(["runpod.io/required-gpu-memory",
                             "runpod.io/gpu-memory", "runpod.io/cloud-type",
                             "runpod.io/ports", "runpod.io/datacenter-ids",
                             "other"]),
            st.text(max_size=10), max_size=3),
    }),
    "spec": _podspec,
})


@settings(max_examples=150, deadline=None)
@given(pod=_fuzz_pod)
def test_prepare_deploy_params_never_crashes(pod):
    """Fuzz: arbitrary (semi-structured) pod specs either translate into
    DeployParams or raise ValidationError — never TypeError/ValueError/
    KeyError from deep inside the translation pipeline."""
    from k8s_runpod_kubelet_amd.config import Config
    from k8s_runpod_kubelet_amd.kube.fake import FakeKube
    from k8s_runpod_kubelet_amd.provider.translate import (
        ValidationError, prepare_deploy_params)
    from k8s_runpod_kubelet_amd.runtime.types import DeployParams

    try:
        out = prepare_deploy_params(pod, FakeKube(), Config())
    except ValidationError:
        return
    assert isinstance(out, DeployParams)
    assert out.gpu_count >= 0


# -- status translation invariants -------------------------------------------

_statuses = st.sampled_from(
    ["RUNNING", "STARTING", "EXITED", "TERMINATING", "TERMINATED", "NOT_FOUND"])


@settings(max_examples=200, deadline=None)
@given(status=_statuses, ports_exposed=st.booleans(),
       exit_code=st.one_of(st.none(), st.integers(0, 255)),
       ready=st.booleans(), n_containers=st.integers(1, 3))
def test_translate_status_invariants(status, ports_exposed, exit_code,
                                     ready, n_containers):
    """Every translation yields a kubectl-valid status: a known phase, all
    four conditions exactly once, one containerStatus per spec container,
    each with exactly one state key."""
    from k8s_runpod_kubelet_amd.provider.instance import InstanceInfo
    from k8s_runpod_kubelet_amd.provider.status import translate_status
    from k8s_runpod_kubelet_amd.runtime.types import (
        ContainerRuntimeInfo, DetailedStatus)

    pod = {"metadata": {"name": "p", "namespace": "d"},
           "spec": {"containers": [{"name": f"c{i}", "image": "x"}
                                   for i in range(n_containers)]}}
    info = InstanceInfo(instance_id="i1", status=status,
                        ports_exposed=ports_exposed)
    detailed = DetailedStatus(
        id="i1", desired_status=status,
        containers=[ContainerRuntimeInfo(name=f"c{i}", pid=100 + i,
                                         exit_code=exit_code, ready=ready)
                    for i in range(n_containers)])
    out = translate_status(pod, info, detailed)
    assert out["phase"] in ("Pending", "Running", "Succeeded", "Failed",
                            "Unknown")
    conds = [c["type"] for c in out["conditions"]]
    assert sorted(conds) == sorted(
        ["PodScheduled", "Initialized", "Ready", "ContainersReady"])
    css = out["containerStatuses"]
    assert [c["name"] for c in css] == [f"c{i}" for i in range(n_containers)]
    for cs in css:
        assert len(cs["state"]) == 1
        assert next(iter(cs["state"])) in ("running", "waiting", "terminated")
    # Ready condition True only when the pod is actually Running+exposed
    ready_cond = {c["type"]: c["status"] for c in out["conditions"]}["Ready"]
    if ready_cond == "True":
        assert status == "RUNNING" and ports_exposed


@given(st.text(alphabet="abcdefghij./-_:@", min_size=0, max_size=60))
@settings(max_examples=300, deadline=None)
def test_normalize_ref_idempotent_and_parseable(ref):
    """normalize_ref is idempotent, and parse_ref round-trips every
    normalized non-empty reference without raising."""
    from k8s_runpod_kubelet_amd.runtime.oci import normalize_ref
    from k8s_runpod_kubelet_amd.runtime.registry import parse_ref

    norm = normalize_ref(ref)
    assert normalize_ref(norm) == norm
    if norm:
        host, name, tagish = parse_ref(norm)
        assert host
        assert isinstance(name, str)
        assert isinstance(tagish, str)


@given(st.lists(
    st.sampled_from(["a", "b", "..", ".", "", "c.d", "..."]),
    min_size=0, max_size=8).map("/".join))
@settings(max_examples=300, deadline=None)
def test_safe_join_never_escapes(path):
    """_safe_join stays inside the root for arbitrary member paths (the
    unpacker-escape guard, lexical half; the symlink half is covered by
    the attack regression tests in test_oci.py)."""
    import tempfile
    from pathlib import Path

    from k8s_runpod_kubelet_amd.runtime.oci import _safe_join

    root = Path(tempfile.gettempdir()) / "amdvk-prop-root"
    out = _safe_join(root, path)
    # result is root or strictly under it
    assert out == root or root in out.parents, (path, out)


@given(st.text(alphabet="ab/.", min_size=0, max_size=40))
@settings(max_examples=300, deadline=None)
def test_container_rel_never_escapes(path):
    """_container_rel (pod-spec mountPath/workingDir clamp) never yields
    a path with '..' or an absolute component."""
    from k8s_runpod_kubelet_amd.runtime.rootfs import _container_rel

    rel = _container_rel(path)
    assert not rel.startswith("/")
    assert ".." not in rel.split("/")


@given(st.lists(
    st.tuples(
        st.sampled_from(["besteffort", "burstable", "guaranteed"]),
        st.integers(min_value=-10, max_value=2_100_000_000),
        st.integers(min_value=0, max_value=1 << 40),
    ), max_size=12))
@settings(max_examples=80, deadline=None)
def test_eviction_ranking_invariants(specs):
    """rank_victims: never returns critical-priority pods; output is a
    subset of input; ordering is stable w.r.t. the documented key (qos
    class rank, then priority, then -usage)."""
    from k8s_runpod_kubelet_amd.provider.eviction import (
        CRITICAL_PRIORITY,
        rank_victims,
    )

    def mk(i, qos, prio):
        c = {"name": "c"}
        if qos == "guaranteed":
            c["resources"] = {"requests": {"cpu": "1", "memory": "1Gi"},
                              "limits": {"cpu": "1", "memory": "1Gi"}}
        elif qos == "burstable":
            c["resources"] = {"requests": {"memory": "1Mi"}}
        return {"metadata": {"name": f"p{i}", "namespace": "default"},
                "spec": {"priority": prio, "containers": [c]}}

    cands = [(f"k{i}", mk(i, qos, prio), usage)
             for i, (qos, prio, usage) in enumerate(specs)]
    ranked = rank_victims(cands)
    keys_in = {k for k, _, _ in cands}
    assert all(k in keys_in for k, _, _ in ranked)
    assert all(int(p["spec"]["priority"]) < CRITICAL_PRIORITY
               for _, p, _ in ranked)
    rank_of = {"besteffort": 0, "burstable": 1, "guaranteed": 2}
    seq = [(rank_of[specs[int(k[1:])][0]], specs[int(k[1:])][1],
            -specs[int(k[1:])][2]) for k, _, _ in ranked]
    assert seq == sorted(seq)
