"""Kubernetes client interface.

The reference leans on client-go (reference cmd/virtual_kubelet/main.go:464
createK8sClient); neither client-go nor the python kubernetes package exists
in this environment, so the API surface this kubelet needs is defined here as
an interface with two implementations:

- ``real.HttpK8sClient`` — kubeconfig/in-cluster HTTP client (httpx).
- ``fake.FakeKube``     — in-memory apiserver for hermetic tests and bench.
"""

from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Any, Dict, Iterator, List, Optional, Tuple


class ApiError(Exception):
    def __init__(self, status_code: int, message: str = ""):
        super().__init__(f"k8s api error {status_code}: {message}")
        self.status_code = status_code
        self.message = message


class NotFoundError(ApiError):
    def __init__(self, message: str = "not found"):
        super().__init__(404, message)


class ConflictError(ApiError):
    def __init__(self, message: str = "conflict"):
        super().__init__(409, message)


class GoneError(ApiError):
    """HTTP 410 — the requested resourceVersion has been compacted away
    (etcd compaction). Watchers must relist and re-watch from the fresh
    list's resourceVersion."""

    def __init__(self, message: str = "resourceVersion too old"):
        super().__init__(410, message)


def is_not_found(exc: BaseException) -> bool:
    return isinstance(exc, ApiError) and exc.status_code == 404


WatchEvent = Tuple[str, Dict[str, Any]]  # ("ADDED"|"MODIFIED"|"DELETED", obj)


class K8sClient(ABC):
    # ---- pods ----
    @abstractmethod
    def list_pods(self, namespace: Optional[str] = None,
                  field_selector: str = "", label_selector: str = "") -> List[Dict[str, Any]]: ...

    def list_pods_with_rv(self, namespace: Optional[str] = None,
                          field_selector: str = "", label_selector: str = ""
                          ) -> Tuple[List[Dict[str, Any]], str]:
        """List plus the PodList's resourceVersion — what a correct informer
        threads into its first watch (list→watch continuity). Default falls
        back to a plain list with no RV (watch then starts from 'now')."""
        return (self.list_pods(namespace=namespace,
                               field_selector=field_selector,
                               label_selector=label_selector), "")

    @abstractmethod
    def get_pod(self, namespace: str, name: str) -> Dict[str, Any]: ...

    @abstractmethod
    def create_pod(self, namespace: str, pod: Dict[str, Any]) -> Dict[str, Any]: ...

    @abstractmethod
    def update_pod(self, namespace: str, pod: Dict[str, Any]) -> Dict[str, Any]: ...

    @abstractmethod
    def patch_pod(self, namespace: str, name: str, patch: Dict[str, Any]) -> Dict[str, Any]:
        """Strategic merge patch against the pod object."""

    @abstractmethod
    def patch_pod_status(self, namespace: str, name: str,
                         patch: Dict[str, Any]) -> Dict[str, Any]:
        """Strategic merge patch against the pods/status subresource
        (reference kubelet.go:1822-1845)."""

    @abstractmethod
    def delete_pod(self, namespace: str, name: str,
                   grace_period_s: Optional[int] = None) -> None: ...

    @abstractmethod
    def watch_pods(self, namespace: Optional[str] = None, field_selector: str = "",
                   resource_version: str = "", timeout_s: float = 60.0
                   ) -> Iterator[WatchEvent]: ...

    # ---- nodes ----
    @abstractmethod
    def get_node(self, name: str) -> Dict[str, Any]: ...

    @abstractmethod
    def create_node(self, node: Dict[str, Any]) -> Dict[str, Any]: ...

    @abstractmethod
    def update_node(self, node: Dict[str, Any]) -> Dict[str, Any]: ...

    @abstractmethod
    def patch_node_status(self, name: str, patch: Dict[str, Any]) -> Dict[str, Any]: ...

    @abstractmethod
    def delete_node(self, name: str) -> None: ...

    # ---- coordination (leases) ----
    @abstractmethod
    def get_lease(self, namespace: str, name: str) -> Dict[str, Any]: ...

    @abstractmethod
    def create_lease(self, namespace: str, lease: Dict[str, Any]) -> Dict[str, Any]: ...

    @abstractmethod
    def update_lease(self, namespace: str, lease: Dict[str, Any]) -> Dict[str, Any]: ...

    def leases_supported(self) -> bool:
        return True

    # ---- workload references ----
    @abstractmethod
    def get_secret(self, namespace: str, name: str) -> Dict[str, Any]: ...

    @abstractmethod
    def get_configmap(self, namespace: str, name: str) -> Dict[str, Any]: ...

    @abstractmethod
    def get_job(self, namespace: str, name: str) -> Dict[str, Any]: ...

    # ---- events / auth ----
    @abstractmethod
    def create_event(self, namespace: str, event: Dict[str, Any]) -> Dict[str, Any]: ...

    def self_subject_review(self) -> Dict[str, Any]:
        """SelfSubjectReview for auth logging (reference main.go:92-108);
        optional — default empty."""
        return {}

    def close(self) -> None: ...
