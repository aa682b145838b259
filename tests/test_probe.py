"""Native KFD/DRM probe tests against a fixture sysfs tree (no GPU needed)."""

import os

import pytest

from k8s_runpod_kubelet_amd.ops import load_native


def write(path, content):
    os.makedirs(os.path.dirname(path), exist_ok=True)
    with open(path, "w") as fh:
        fh.write(content)


def make_fake_sysfs(root, gpus=2, cpu_nodes=1, vram_gb=288, with_ras_error=()):
    """Build a KFD topology + DRM tree like an MI355X node's."""
    topo = os.path.join(root, "class/kfd/kfd/topology/nodes")
    node_id = 0
    for _ in range(cpu_nodes):
        write(f"{topo}/{node_id}/properties", "simd_count 0\ncpu_cores_count 64\n")
        node_id += 1
    first_gpu_node = node_id
    for g in range(gpus):
        ndir = f"{topo}/{node_id}"
        minor = 128 + g
        write(
            f"{ndir}/properties",
            f"simd_count 1024\nsimd_per_cu 4\ngfx_target_version 90500\n"
            f"drm_render_minor {minor}\ndevice_id 29857\nlocation_id {g}\n"
            f"max_engine_clk_fcompute 2400\n",
        )
        write(f"{ndir}/gpu_id", f"{4000 + g}")
        write(
            f"{ndir}/mem_banks/0/properties",
            f"heap_type 1\nsize_in_bytes {vram_gb * 1024**3}\n",
        )
        # xGMI links to every other GPU node (type 11), plus one PCIe link
        # to the CPU node (type 2) that must be ignored.
        link = 0
        write(f"{ndir}/io_links/{link}/properties",
              "type 2\nnode_to 0\nweight 20\n")
        link += 1
        for peer in range(gpus):
            if peer == g:
                continue
            write(
                f"{ndir}/io_links/{link}/properties",
                f"type 11\nnode_to {first_gpu_node + peer}\nweight 15\n"
                f"min_bandwidth 153000\nmax_bandwidth 153000\n",
            )
            link += 1
        # DRM side
        dev = os.path.join(root, f"class/drm/renderD{minor}/device")
        write(f"{dev}/mem_info_vram_total", str(vram_gb * 1024**3))
        write(f"{dev}/mem_info_vram_used", str((g + 1) * 1024**3))
        write(f"{dev}/gpu_busy_percent", str(5 * g))
        write(f"{dev}/unique_id", f"0xabc{g}")
        write(f"{dev}/hwmon/hwmon{g}/temp1_input", "45000")
        if g in with_ras_error:
            write(f"{dev}/ras/umc_err_count", "ue: 3\nce: 10\n")
        else:
            write(f"{dev}/ras/umc_err_count", "ue: 0\nce: 0\n")
        node_id += 1
    return root


def test_enumerate_fake_tree(tmp_path):
    native = load_native()
    root = make_fake_sysfs(str(tmp_path), gpus=4)
    gpus = native.enumerate_gpus(root)
    assert len(gpus) == 4  # CPU node skipped
    g0 = gpus[0]
    assert g0.index == 0
    assert g0.render_minor == 128
    assert g0.gfx_target_version == 90500
    assert g0.vram_total_bytes == 288 * 1024**3
    assert g0.vram_used_bytes == 1 * 1024**3
    assert g0.cu_count == 256
    assert g0.temperature_mc == 45000
    assert g0.unique_id == "0xabc0"
    # xGMI links: 3 peers, remapped to dense GPU indices, PCIe link ignored
    peers = sorted(l.peer_gpu_index for l in g0.xgmi_links)
    assert peers == [1, 2, 3]
    assert all(l.weight == 15 for l in g0.xgmi_links)


def test_ras_health_gate(tmp_path):
    native = load_native()
    root = make_fake_sysfs(str(tmp_path), gpus=2, with_ras_error=(1,))
    gpus = native.enumerate_gpus(root)
    assert gpus[0].healthy
    assert not gpus[1].healthy
    assert gpus[1].ras_uncorrectable == 3


def test_read_gpu_dynamic(tmp_path):
    native = load_native()
    root = make_fake_sysfs(str(tmp_path), gpus=1)
    d = native.read_gpu_dynamic(root, 128)
    assert d.vram_used_bytes == 1024**3
    assert d.temperature_mc == 45000


def test_empty_tree(tmp_path):
    native = load_native()
    assert native.enumerate_gpus(str(tmp_path)) == []


def test_inventory_wraps_probe(tmp_path):
    from k8s_runpod_kubelet_amd.gpu.inventory import Inventory

    root = make_fake_sysfs(str(tmp_path), gpus=3)
    inv = Inventory(sysfs_root=root, allow_synthetic=False)
    gpus = inv.discover()
    assert len(gpus) == 3
    assert not inv.synthetic
    assert gpus[0].arch == "gfx950"
    assert gpus[0].xgmi_peers == {1: 15, 2: 15}
    inv.refresh_dynamic()
    assert gpus[2].vram_used_bytes == 3 * 1024**3


def test_inventory_synthetic_fallback():
    from k8s_runpod_kubelet_amd.gpu.inventory import Inventory

    inv = Inventory(sysfs_root="/nonexistent", allow_synthetic=True)
    gpus = inv.discover()
    assert len(gpus) == 8
    assert inv.synthetic
    assert gpus[0].vram_total_bytes == 288 * 1024**3
    assert len(gpus[0].xgmi_peers) == 7  # all-to-all
