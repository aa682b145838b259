"""Prometheus metrics.

The reference exports no metrics at all (prometheus is only an
indirect dep; the kubelet API stats hooks are stubbed, main.go:233-235).
These cover the BASELINE protocol directly: pod-Ready latency histogram,
scheduling counters, per-GPU HBM/busy gauges from the DRM probe."""

from __future__ import annotations

from typing import List

from prometheus_client import (
    CollectorRegistry,
    Counter,
    Gauge,
    Histogram,
    generate_latest,
)

registry = CollectorRegistry()

pods_created = Counter("amdvk_pods_created_total", "Pods accepted by CreatePod",
                       registry=registry)
pods_evicted = Counter("amdvk_pods_evicted_total",
                       "Pods evicted under node memory pressure",
                       registry=registry)
pods_deleted = Counter("amdvk_pods_deleted_total", "Pods removed by DeletePod",
                       registry=registry)

pod_ready_seconds = Histogram(
    "amdvk_pod_ready_seconds",
    "Pod creation to Ready latency (the BASELINE headline metric)",
    buckets=(0.005, 0.01, 0.025, 0.05, 0.1, 0.25, 0.5, 1, 2.5, 5, 10, 30, 60, 300),
    registry=registry,
)
_PHASE_BUCKETS = (0.0005, 0.001, 0.0025, 0.005, 0.01, 0.025, 0.05, 0.1,
                  0.25, 0.5, 1, 2.5, 5)
deploy_seconds = Histogram(
    "amdvk_deploy_seconds",
    "Spec translation + bind + launch latency",
    buckets=_PHASE_BUCKETS,
    registry=registry,
)
# Per-phase latency histograms (the reference has no tracing at
# all; the north-star metric is latency, so every deploy phase is measured).
translate_seconds = Histogram(
    "amdvk_translate_seconds", "Spec translation (annotations/env/ports)",
    buckets=_PHASE_BUCKETS, registry=registry)
bind_seconds = Histogram(
    "amdvk_bind_seconds", "GPU set selection + ledger reservation",
    buckets=_PHASE_BUCKETS, registry=registry)
launch_seconds = Histogram(
    "amdvk_launch_seconds", "posix_spawn of containers (native clock)",
    buckets=_PHASE_BUCKETS, registry=registry)
cgroup_migrate_seconds = Histogram(
    "amdvk_cgroup_migrate_seconds",
    "cgroup.procs migration write (native clock; serializes on cgroup_mutex)",
    buckets=_PHASE_BUCKETS, registry=registry)


def hist_mean_ms(h) -> float:
    count = sum(b.get() for b in h._buckets)
    return (h._sum.get() / count * 1000.0) if count else 0.0

gpu_hbm_free = Gauge("amdvk_gpu_hbm_free_bytes", "Free HBM headroom per GPU",
                     ["gpu"], registry=registry)
gpu_hbm_total = Gauge("amdvk_gpu_hbm_total_bytes", "Total HBM per GPU",
                      ["gpu"], registry=registry)
gpu_busy = Gauge("amdvk_gpu_busy_percent", "GPU busy percent", ["gpu"],
                 registry=registry)
gpu_temperature = Gauge("amdvk_gpu_temperature_celsius", "GPU temperature",
                        ["gpu"], registry=registry)
gpu_schedulable = Gauge("amdvk_gpu_schedulable", "1 if the GPU accepts pods",
                        ["gpu"], registry=registry)


def observe_gpus(states: List) -> None:
    for state in states:
        label = str(state.gpu.index)
        gpu_hbm_free.labels(label).set(state.headroom_bytes)
        gpu_hbm_total.labels(label).set(state.gpu.vram_total_bytes)
        if state.gpu.busy_percent >= 0:
            gpu_busy.labels(label).set(state.gpu.busy_percent)
        if state.gpu.temperature_mc >= 0:
            gpu_temperature.labels(label).set(state.gpu.temperature_mc / 1000.0)
        gpu_schedulable.labels(label).set(1 if state.schedulable else 0)


def render() -> bytes:
    return generate_latest(registry)
