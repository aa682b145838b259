import pytest

from k8s_runpod_kubelet_amd.gpu.binder import (
    Binder,
    BindRequest,
    PlacementError,
    device_env,
)
from k8s_runpod_kubelet_amd.gpu.inventory import Inventory
from k8s_runpod_kubelet_amd.gpu.ledger import Ledger

GIB = 1024**3


def make_ledger(count=8, vram_gb=288):
    inv = Inventory(synthetic_count=count, synthetic_vram_gb=vram_gb)
    inv.discover()
    ledger = Ledger(inv)
    ledger.sync_inventory()
    return inv, ledger


def test_reserve_release_headroom():
    _, ledger = make_ledger()
    ledger.reserve("p1", [0], 100 * GIB)
    state = ledger.states[0]
    assert state.reserved_bytes == 100 * GIB
    assert state.headroom_bytes == 188 * GIB
    assert ledger.get_reservation("p1").gpu_indices == [0]
    ledger.release("p1")
    assert ledger.states[0].reserved_bytes == 0
    assert ledger.get_reservation("p1") is None


def test_double_reserve_rejected():
    _, ledger = make_ledger()
    ledger.reserve("p1", [0], GIB)
    with pytest.raises(ValueError, match="already holds"):
        ledger.reserve("p1", [1], GIB)


def test_exclusive_gpu_claims():
    # amd.com/gpu is exclusive: a bound GPU is ineligible for other pods.
    _, ledger = make_ledger(count=3)
    binder = Binder(ledger)
    first = binder.bind(BindRequest("p1", 1, GIB))
    second = binder.bind(BindRequest("p2", 1, GIB))
    assert set(first).isdisjoint(second)
    with pytest.raises(ValueError, match="already bound"):
        ledger.reserve("p3", first, GIB)


def test_best_fit_prefers_smallest_sufficient_headroom():
    inv, ledger = make_ledger(count=3)
    # Live VRAM use (other tenants/system) differentiates headroom.
    inv.gpus[0].vram_used_bytes = 200 * GIB  # headroom 88
    inv.gpus[1].vram_used_bytes = 100 * GIB  # headroom 188
    ledger.sync_inventory()
    binder = Binder(ledger)
    # 50 GiB fits on GPU 0's 88 GiB → best-fit picks 0
    chosen = binder.select(BindRequest("p", 1, 50 * GIB, max_cost=1.0))
    assert chosen == [0]
    # 100 GiB doesn't fit on 0 → picks 1 (headroom 188 < GPU2's 288)
    chosen = binder.select(BindRequest("p", 1, 100 * GIB, max_cost=1.0))
    assert chosen == [1]


def test_cost_gating_respects_max_gpu_price():
    inv, ledger = make_ledger(count=2)
    inv.gpus[0].busy_percent = 60  # cost 0.3
    ledger.sync_inventory()
    binder = Binder(ledger)
    chosen = binder.select(BindRequest("p", 1, GIB, max_cost=0.2))
    assert chosen == [1]  # GPU 0 too "expensive"
    # Both eligible at default 0.5; equal headroom → lower cost wins
    chosen = binder.select(BindRequest("p", 1, GIB, max_cost=0.5))
    assert chosen == [1]


def test_multi_gpu_set_is_xgmi_connected():
    inv, ledger = make_ledger(count=8)
    # Break links: GPU 7 only links to 6.
    for g in inv.gpus:
        if g.index == 7:
            g.xgmi_peers = {6: 1}
        else:
            g.xgmi_peers = {j: 1 for j in range(7) if j != g.index}
            if g.index == 6:
                g.xgmi_peers[7] = 1
    ledger.sync_inventory()
    binder = Binder(ledger)
    chosen = binder.bind(BindRequest("p", 4, 256 * GIB))
    assert len(chosen) == 4
    assert 7 not in chosen  # poorly-connected GPU avoided
    # reservation split across the set
    res = ledger.get_reservation("p")
    assert res.bytes_per_gpu == 64 * GIB


def test_placement_error_when_full():
    _, ledger = make_ledger(count=2)
    binder = Binder(ledger)
    binder.bind(BindRequest("p1", 2, 0))
    with pytest.raises(PlacementError):
        binder.bind(BindRequest("p2", 1, GIB))
    binder.unbind("p1")
    assert len(binder.bind(BindRequest("p3", 2, 2 * 280 * GIB))) == 2


def test_unhealthy_gpu_not_schedulable():
    inv, ledger = make_ledger(count=2)
    inv.gpus[0].healthy = False
    ledger.sync_inventory()
    binder = Binder(ledger)
    assert binder.select(BindRequest("p", 1, GIB)) == [1]
    assert ledger.schedulable_count() == 1
    with pytest.raises(PlacementError):
        binder.select(BindRequest("p", 2, GIB))


def test_adopt_rebuilds_without_checks():
    inv, ledger = make_ledger(count=2)
    inv.gpus[0].healthy = False
    ledger.sync_inventory()
    ledger.adopt("p1", [0], 10 * GIB)  # adoption ignores schedulability
    assert ledger.states[0].reserved_bytes == 10 * GIB


def test_device_env():
    env = device_env([2, 5])
    assert env["ROCR_VISIBLE_DEVICES"] == "2,5"
    assert env["HIP_VISIBLE_DEVICES"] == "2,5"
    assert device_env([])["ROCR_VISIBLE_DEVICES"] == ""


def test_device_env_exports_xgmi_topology():
    inv, _ = make_ledger(count=4)
    env = device_env([1, 3], inv)
    # Pod-local adjacency: devices 0 (=GPU 1) and 1 (=GPU 3) are peers.
    assert env["AMDVK_XGMI_PEERS"] == "0:1@1;1:0@1"
    # Single-GPU pods get no topology map.
    assert "AMDVK_XGMI_PEERS" not in device_env([2], inv)


def test_binder_prefers_settled_gpu():
    """A GPU freed < 1 s ago still runs the previous pod's KFD teardown
    (measured ~140 ms HIP-init penalty, profiles/pw_timing.txt): with a
    settled alternative available, the binder must pick the alternative."""
    _, ledger = make_ledger(count=2)
    binder = Binder(ledger)
    assert binder.bind(BindRequest("p1", 1, GIB)) == [0]
    binder.unbind("p1")  # GPU 0 now settling
    assert binder.select(BindRequest("p2", 1, GIB)) == [1]
    # When every eligible GPU is settling, placement still proceeds.
    assert binder.bind(BindRequest("p3", 1, GIB)) == [1]
    binder.unbind("p3")
    chosen = binder.select(BindRequest("p4", 2, 2 * GIB))
    assert sorted(chosen) == [0, 1]
