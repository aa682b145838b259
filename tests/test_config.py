import pytest

from k8s_runpod_kubelet_amd.config import Config, load_config
from k8s_runpod_kubelet_amd.utils.backoff import parse_duration_s


def test_defaults_match_reference_contract():
    cfg = Config()
    # reference flag defaults (main.go:59-73)
    assert cfg.node_name == "virtual-runpod"
    assert cfg.listen_port == 10250
    assert cfg.namespace == "kube-system"
    assert cfg.max_gpu_price == 0.5  # DefaultMaxPrice (runpod_client.go:49)
    assert cfg.gpu_memory_default_gb == 16  # runpod_client.go:1189
    assert cfg.heartbeat_interval_s == 300.0
    # ladder thresholds ≙ 5/10/15 min (kubelet.go:1333/:1285/:1350, :788)
    assert cfg.stuck_reterminate_after_s == 300.0
    assert cfg.stuck_statuserr_force_after_s == 600.0
    assert cfg.stuck_force_after_s == 900.0
    assert cfg.pending_pod_timeout_s == 900.0


def test_load_yaml_overlay(tmp_path):
    p = tmp_path / "cfg.yaml"
    p.write_text("node_name: n1\nmax_gpu_price: 0.9\ndatacenter-ids: 'a,b'\n")
    cfg = load_config(str(p))
    assert cfg.node_name == "n1"
    assert cfg.max_gpu_price == 0.9
    assert cfg.datacenter_ids == ["a", "b"]


def test_load_yaml_rejects_unknown_keys(tmp_path):
    p = tmp_path / "cfg.yaml"
    p.write_text("not_a_real_key: 1\n")
    with pytest.raises(ValueError, match="unknown key"):
        load_config(str(p))


def test_load_none_returns_defaults():
    assert load_config(None).node_name == "virtual-runpod"


def test_parse_duration():
    assert parse_duration_s("30s", 0) == 30.0
    assert parse_duration_s("5m", 0) == 300.0
    assert parse_duration_s("500ms", 0) == 0.5
    assert parse_duration_s("2h", 0) == 7200.0
    assert parse_duration_s(None, 7.0) == 7.0
    assert parse_duration_s(12, 0) == 12.0
