"""GPU inventory — local-node replacement for the reference's cloud GPU
catalog (reference pkg/virtual_kubelet/runpod_client.go:429-520 GetGPUTypes).

Wraps the native KFD/DRM probe (ops/csrc/probe.cpp) and presents typed GPU
records. On a box without GPUs (or in hermetic tests with a fixture sysfs
tree) the inventory is whatever the probe finds under ``sysfs_root`` — plus a
synthetic fallback for CPU-only development, sized like one 8×MI355X node
(288 GB HBM3E per GPU, 7 xGMI links per GPU pair-wise — all-to-all on one
node).
"""

from __future__ import annotations

import logging
from dataclasses import dataclass, field
from typing import Dict, List, Optional

log = logging.getLogger("gpu.inventory")

MI355X_VRAM_BYTES = 288 * 1024**3
MI355X_GFX_TARGET = 90500  # gfx950
XGMI_LINK_GBS = 153  # per point-to-point link, one direction


@dataclass
class Gpu:
    index: int
    render_minor: int = -1
    kfd_node: int = -1
    gpu_id: str = ""
    unique_id: str = ""
    gfx_target_version: int = MI355X_GFX_TARGET
    cu_count: int = 256
    vram_total_bytes: int = MI355X_VRAM_BYTES
    vram_used_bytes: int = 0
    busy_percent: int = 0
    temperature_mc: int = -1
    ras_uncorrectable: int = 0
    healthy: bool = True
    # peer GPU index -> xGMI link weight/bandwidth (MB/s); absent = no link
    xgmi_peers: Dict[int, int] = field(default_factory=dict)

    @property
    def vram_free_bytes(self) -> int:
        return max(0, self.vram_total_bytes - self.vram_used_bytes)

    @property
    def arch(self) -> str:
        v = self.gfx_target_version
        if v:
            major, minor, step = v // 10000, (v // 100) % 100, v % 100
            return f"gfx{major}{minor:x}{step:x}" if minor > 9 else f"gfx{major}{minor}{step:x}"
        return "unknown"


class Inventory:
    """Enumerates and refreshes the node's GPUs.

    ``synthetic_count > 0`` forces a synthetic inventory (CPU-only dev/tests);
    otherwise the native probe runs against ``sysfs_root`` and a synthetic
    fallback only kicks in if *no* GPUs are found and ``allow_synthetic``.
    """

    def __init__(
        self,
        sysfs_root: str = "/sys",
        synthetic_count: int = -1,
        synthetic_vram_gb: int = 288,
        allow_synthetic: bool = True,
    ):
        self.sysfs_root = sysfs_root
        self.synthetic_count = synthetic_count
        self.synthetic_vram_gb = synthetic_vram_gb
        self.allow_synthetic = allow_synthetic
        self.synthetic = False
        self._native = None
        self.gpus: List[Gpu] = []

    def _load_native(self):
        if self._native is None:
            from ..ops import load_native

            self._native = load_native()
        return self._native

    def discover(self) -> List[Gpu]:
        if self.synthetic_count and self.synthetic_count > 0:
            self.gpus = self._make_synthetic(self.synthetic_count)
            self.synthetic = True
            return self.gpus

        native = self._load_native()
        raw = native.enumerate_gpus(self.sysfs_root)
        gpus: List[Gpu] = []
        for r in raw:
            g = Gpu(
                index=r.index,
                render_minor=r.render_minor,
                kfd_node=r.kfd_node,
                gpu_id=r.gpu_id,
                unique_id=r.unique_id,
                gfx_target_version=int(r.gfx_target_version),
                cu_count=int(r.cu_count) or 256,
                vram_total_bytes=int(r.vram_total_bytes) or MI355X_VRAM_BYTES,
                vram_used_bytes=int(r.vram_used_bytes),
                busy_percent=int(r.busy_percent),
                temperature_mc=int(r.temperature_mc),
                ras_uncorrectable=int(r.ras_uncorrectable),
                healthy=bool(r.healthy),
            )
            for link in r.xgmi_links:
                if link.peer_gpu_index >= 0:
                    g.xgmi_peers[link.peer_gpu_index] = int(link.weight) or 1
            gpus.append(g)

        if not gpus and self.allow_synthetic:
            log.warning(
                "no GPUs found under %s; using synthetic 8xMI355X inventory",
                self.sysfs_root,
            )
            gpus = self._make_synthetic(8)
            self.synthetic = True
        else:
            self.synthetic = False
        self.gpus = gpus
        return gpus

    def refresh_dynamic(self) -> None:
        """Per-tick refresh of live VRAM/busy/temp counters AND health: RAS
        uncorrectable errors appearing at runtime must cordon the GPU (the
        reference re-probes backend health every tick, kubelet.go:320-331 —
        here that check is per-GPU)."""
        if self.synthetic:
            return
        native = self._load_native()
        by_index = {}
        try:
            by_index = {r.index: r for r in native.enumerate_gpus(self.sysfs_root)}
        except Exception:
            log.exception("health re-probe failed; refreshing counters only")
        for g in self.gpus:
            r = by_index.get(g.index)
            if r is not None:
                g.vram_total_bytes = int(r.vram_total_bytes) or g.vram_total_bytes
                g.vram_used_bytes = int(r.vram_used_bytes)
                g.busy_percent = int(r.busy_percent)
                g.temperature_mc = int(r.temperature_mc)
                was_healthy = g.healthy
                g.ras_uncorrectable = int(r.ras_uncorrectable)
                g.healthy = bool(r.healthy)
                if was_healthy and not g.healthy:
                    log.warning("GPU became unhealthy (RAS)",
                                extra={"gpu": g.index,
                                       "uncorrectable": g.ras_uncorrectable})
                continue
            if g.render_minor < 0:
                continue
            d = native.read_gpu_dynamic(self.sysfs_root, g.render_minor)
            if d.vram_total_bytes:
                g.vram_total_bytes = d.vram_total_bytes
            g.vram_used_bytes = d.vram_used_bytes
            g.busy_percent = d.busy_percent
            g.temperature_mc = d.temperature_mc

    def _make_synthetic(self, count: int) -> List[Gpu]:
        vram = self.synthetic_vram_gb * 1024**3
        gpus = []
        for i in range(count):
            g = Gpu(index=i, render_minor=128 + i, vram_total_bytes=vram)
            # One MI355X node is xGMI all-to-all: 7 p2p links per GPU.
            g.xgmi_peers = {j: 1 for j in range(count) if j != i}
            gpus.append(g)
        return gpus

    def get(self, index: int) -> Optional[Gpu]:
        for g in self.gpus:
            if g.index == index:
                return g
        return None
