"""Process wiring: build and run the full kubelet stack.

Counterpart of the reference's main() orchestration
(cmd/virtual_kubelet/main.go:333-431): logger → config → k8s client → env
check → provider → informers → controllers → API server → health server →
shutdown handlers → LoadRunning → block. Used by the CLI, by bench.py and by
tests (with fake client/runtime injected)."""

from __future__ import annotations

import logging
import os
import signal
import threading
from dataclasses import dataclass, field
from typing import Optional

from .config import Config
from .gpu.inventory import Inventory
from .gpu.ledger import Ledger
from .kube.apiserver import KubeletApiServer
from .kube.client import K8sClient
from .kube.events import EventRecorder
from .kube.informer import PodInformer
from .kube.nodecontroller import NodeController
from .kube.podcontroller import PodController
from .provider.provider import Provider
from .runtime.base import Runtime
from .server.health import HealthServer

log = logging.getLogger("app")


@dataclass
class Stack:
    config: Config
    client: K8sClient
    runtime: Runtime
    provider: Provider
    informer: PodInformer
    pod_controller: PodController
    node_controller: NodeController
    health: Optional[HealthServer] = None
    api_server: Optional[KubeletApiServer] = None
    inventory: Optional[Inventory] = None
    ledger: Optional[Ledger] = None
    recorder: Optional[EventRecorder] = None
    _stopped: bool = field(default=False, repr=False)

    def start(self, serve_http: bool = True) -> None:
        self.provider.start()
        self.node_controller.start()
        self.pod_controller.start()
        if serve_http:
            self.health = HealthServer(
                self.config.health_server_address, self.provider.ping,
                ledger=self.ledger,
                admin_token=self.config.admin_token
                or os.environ.get("AMDVK_ADMIN_TOKEN", ""),
            )
            self.health.start()
            self.api_server = KubeletApiServer(
                self.provider, self.config.internal_ip,
                self.config.listen_port,
                token=self.config.kubelet_api_token
                or os.environ.get("AMDVK_KUBELET_TOKEN", ""),
            )
            self.api_server.start()
        # Startup reconciliation (reference main.go:426 → LoadRunning).
        self.provider.load_running()

    def stop(self) -> None:
        if self._stopped:
            return
        self._stopped = True
        if self.api_server is not None:
            self.api_server.stop()
        if self.health is not None:
            self.health.stop()
        self.pod_controller.stop()
        self.node_controller.stop()
        self.provider.stop()
        self.runtime.close()
        self.client.close()


def build_stack(
    config: Config,
    client: Optional[K8sClient] = None,
    runtime: Optional[Runtime] = None,
) -> Stack:
    if client is None:
        from .kube.real import create_k8s_client

        client = create_k8s_client(config.kubeconfig)

    inventory: Optional[Inventory] = None
    ledger: Optional[Ledger] = None
    if runtime is None:
        inventory = Inventory(
            sysfs_root=config.sysfs_root,
            synthetic_count=config.gpu_count_override,
            synthetic_vram_gb=(
                config.gpu_vram_gb_override if config.gpu_vram_gb_override > 0 else 288
            ),
        )
        inventory.discover()
        ledger = Ledger(inventory)
        ledger.sync_inventory()
        if config.runtime == "fake":
            from .runtime.fake import FakeRuntime

            runtime = FakeRuntime(gpu_count=ledger.total_gpus())
        else:
            from .runtime.oci import ImageStore
            from .runtime.process_runtime import ProcessRuntime

            runtime = ProcessRuntime(
                ledger,
                state_dir=config.state_dir,
                cgroup_root=config.cgroup_root,
                cgroup_parent=config.cgroup_parent,
                pod_namespaces=config.pod_namespaces,
                log_max_bytes=config.pod_log_max_bytes,
                image_store=ImageStore(config.resolved_image_store_dir()),
                image_isolation=config.image_isolation,
                image_gpu_binds=config.image_gpu_binds,
                image_extra_binds=config.image_extra_binds,
                image_registry=config.image_registry,
                image_registry_token=config.image_registry_token
                or os.environ.get("AMDVK_REGISTRY_TOKEN", ""),
            )

    provider = Provider(client, config, runtime, ledger=ledger, inventory=inventory)
    recorder = EventRecorder(client)
    provider.recorder = recorder
    informer = PodInformer(
        client, config.node_name, resync_interval_s=config.reconcile_interval_s
    )
    pod_controller = PodController(
        client, informer, provider, workers=config.pod_controller_workers
    )
    node_controller = NodeController(
        client, provider, status_interval_s=config.reconcile_interval_s
    )
    return Stack(
        config=config,
        client=client,
        runtime=runtime,
        provider=provider,
        informer=informer,
        pod_controller=pod_controller,
        node_controller=node_controller,
        inventory=inventory,
        ledger=ledger,
        recorder=recorder,
    )


def run_forever(stack: Stack) -> None:
    stop_event = threading.Event()

    def handle(signum, frame):  # reference main.go:344-350
        log.info("signal received; shutting down", extra={"signal": signum})
        stop_event.set()

    signal.signal(signal.SIGINT, handle)
    signal.signal(signal.SIGTERM, handle)
    stack.start()
    try:
        stop_event.wait()
    finally:
        stack.stop()
