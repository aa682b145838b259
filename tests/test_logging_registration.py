import json
import logging

from k8s_runpod_kubelet_amd.logging_setup import (
    JSONFormatter,
    KVFormatter,
    MultiHandler,
    initialize_logger,
)
from k8s_runpod_kubelet_amd.provider.registration import Registrar, build_payload


def test_multihandler_fans_out():
    records_a, records_b = [], []

    class Sink(logging.Handler):
        def __init__(self, store, level=logging.INFO):
            super().__init__(level)
            self.store = store

        def emit(self, record):
            self.store.append(record.getMessage())

    mh = MultiHandler([Sink(records_a), Sink(records_b, level=logging.ERROR)])
    logger = logging.Logger("t")
    logger.addHandler(mh)
    logger.info("hello")
    logger.error("bad")
    assert records_a == ["hello", "bad"]
    assert records_b == ["bad"]  # per-handler level respected


def test_log_level_applied(tmp_path):
    # reference quirk: --log-level parsed but never applied (main.go:69);
    # here it must take effect.
    root = initialize_logger("error")
    assert root.level == logging.ERROR
    root = initialize_logger("debug")
    assert root.level == logging.DEBUG
    initialize_logger("info")


def test_kv_and_json_formatters():
    record = logging.LogRecord("x", logging.INFO, "f.py", 1, "msg here", (), None)
    record.pod = "ns/p"
    text = KVFormatter().format(record)
    assert 'msg="msg here"' in text and "pod='ns/p'" in text
    payload = json.loads(JSONFormatter().format(record))
    assert payload["msg"] == "msg here"
    assert payload["pod"] == "ns/p"


def test_json_file_sink(tmp_path):
    path = tmp_path / "log.json"
    initialize_logger("info", str(path))
    logging.getLogger("t2").info("to file", extra={"k": 1})
    initialize_logger("info")  # reset handlers / flush
    lines = [json.loads(l) for l in path.read_text().splitlines()]
    assert any(l["msg"] == "to file" and l["k"] == 1 for l in lines)


def test_registration_payload_shape():
    p = build_payload("node1", "kube-system", 8)
    # reference RegistrationPayload fields (kubelet.go:88-127)
    assert set(p) == {"clusterName", "namespace", "nodeName", "version",
                      "capabilities", "metadata"}
    assert p["metadata"]["gpuCount"] == 8


def test_registrar_disabled_by_default():
    r = Registrar("", "n", "ns", 8)
    assert r.register() is True  # no endpoint: registration is a no-op
    r.start_heartbeat()
    assert r._ticker is None
    r.stop_heartbeat()


def test_registrar_failure_not_fatal(monkeypatch):
    # endpoint set but unreachable: register() returns False, no raise
    # (unlike the reference, which fails provider construction,
    # kubelet.go:369-371)
    r = Registrar("http://127.0.0.1:1", "n", "ns", 8, heartbeat_interval_s=0)
    assert r.register() is False
    r.stop_heartbeat()


def test_event_recorder_window_expiry(fake_kube, monkeypatch):
    """Aggregation window: repeats inside the window bump one Event;
    after the window a fresh Event object starts at count 1."""
    import k8s_runpod_kubelet_amd.kube.events as ev_mod

    clock = [1000.0]
    monkeypatch.setattr(ev_mod.time, "monotonic", lambda: clock[0])
    rec = ev_mod.EventRecorder(fake_kube)
    obj = {"kind": "Pod", "metadata": {"name": "w", "namespace": "default",
                                       "uid": "u"}}
    rec.event(obj, "Warning", "BackOff", "m")
    clock[0] += 100
    rec.event(obj, "Warning", "BackOff", "m")
    clock[0] += rec.AGGREGATION_WINDOW_S + 1
    rec.event(obj, "Warning", "BackOff", "m")
    with fake_kube._lock:
        evs = [e for e in fake_kube.events.objects.values()
               if e.get("reason") == "BackOff"]
    assert sorted(e["count"] for e in evs) == [1, 2]
    assert len(evs) == 2


def test_event_update_falls_back_to_create(fake_kube):
    """If the aggregated Event was TTL-GC'd by the apiserver, the update
    404s — the recorder recreates instead of dropping the signal."""
    from k8s_runpod_kubelet_amd.kube.events import EventRecorder

    rec = EventRecorder(fake_kube)
    obj = {"kind": "Pod", "metadata": {"name": "g", "namespace": "default",
                                       "uid": "u"}}
    rec.event(obj, "Warning", "BackOff", "m")

    def gone(_ns, _ev):
        raise RuntimeError("404 not found")

    fake_kube.update_event = gone
    rec.event(obj, "Warning", "BackOff", "m")  # update fails -> recreate
    with fake_kube._lock:
        evs = [e for e in fake_kube.events.objects.values()
               if e.get("reason") == "BackOff"]
    assert len(evs) == 2  # old + recreated series
    assert all(e["count"] == 1 for e in evs)
