"""Real Kubernetes API client over HTTP (httpx).

Counterpart of the reference's client-go factory
(cmd/virtual_kubelet/main.go:464-502 createK8sClient): in-cluster service
account config → ``~/.kube/config`` fallback → explicit ``--kubeconfig``.
Implements exactly the API surface this kubelet needs (see client.K8sClient)
including streaming watches and the pods/status strategic-merge PATCH, with
the reference's retry policy (3 attempts, 500 ms·n linear backoff,
runpod_client.go:268-343) on transient failures."""

from __future__ import annotations

import base64
import json
import logging
import os
import ssl
import tempfile
from typing import Any, Dict, Iterator, List, Optional

import httpx
import yaml

from ..utils.backoff import retry
from .client import (ApiError, ConflictError, GoneError, K8sClient,
                     NotFoundError, WatchEvent)

log = logging.getLogger("kube.real")

SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"


class ClusterConfig:
    def __init__(self, server: str, token: str = "", ca_path: str = "",
                 client_cert: str = "", client_key: str = "", verify: bool = True,
                 token_path: str = ""):
        self.server = server.rstrip("/")
        self.token = token
        self.ca_path = ca_path
        self.client_cert = client_cert
        self.client_key = client_key
        self.verify = verify
        # service-account token file: re-read on 401 (kubelet semantics —
        # bound SA tokens rotate; a client that caches the startup token
        # goes Unauthorized after ~1 h on a real cluster)
        self.token_path = token_path


def _materialize(data_b64: str, suffix: str) -> str:
    fd, path = tempfile.mkstemp(suffix=suffix, prefix="amdvk-")
    with os.fdopen(fd, "wb") as fh:
        fh.write(base64.b64decode(data_b64))
    return path


def load_kubeconfig(path: str) -> ClusterConfig:
    with open(path, "r", encoding="utf-8") as fh:
        doc = yaml.safe_load(fh)
    ctx_name = doc.get("current-context", "")
    contexts = {c["name"]: c["context"] for c in doc.get("contexts", [])}
    clusters = {c["name"]: c["cluster"] for c in doc.get("clusters", [])}
    users = {u["name"]: u["user"] for u in doc.get("users", [])}
    ctx = contexts.get(ctx_name) or (list(contexts.values())[0] if contexts else {})
    cluster = clusters.get(ctx.get("cluster", "")) or {}
    user = users.get(ctx.get("user", "")) or {}

    ca_path = cluster.get("certificate-authority", "")
    if not ca_path and cluster.get("certificate-authority-data"):
        ca_path = _materialize(cluster["certificate-authority-data"], ".ca.crt")
    cert = user.get("client-certificate", "")
    if not cert and user.get("client-certificate-data"):
        cert = _materialize(user["client-certificate-data"], ".crt")
    key = user.get("client-key", "")
    if not key and user.get("client-key-data"):
        key = _materialize(user["client-key-data"], ".key")
    return ClusterConfig(
        server=cluster.get("server", ""),
        token=user.get("token", ""),
        ca_path=ca_path,
        client_cert=cert,
        client_key=key,
        verify=not cluster.get("insecure-skip-tls-verify", False),
    )


def load_in_cluster() -> Optional[ClusterConfig]:
    token_path = os.path.join(SA_DIR, "token")
    if not os.path.exists(token_path):
        return None
    host = os.environ.get("KUBERNETES_SERVICE_HOST")
    port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
    if not host:
        return None
    with open(token_path, "r", encoding="utf-8") as fh:
        token = fh.read().strip()
    return ClusterConfig(
        server=f"https://{host}:{port}",
        token=token,
        ca_path=os.path.join(SA_DIR, "ca.crt"),
        token_path=token_path,
    )


def create_k8s_client(kubeconfig: str = "") -> "HttpK8sClient":
    """in-cluster → ~/.kube/config → --kubeconfig (reference main.go:464-502
    order, with the explicit flag taking precedence when given)."""
    if kubeconfig:
        return HttpK8sClient(load_kubeconfig(kubeconfig))
    cc = load_in_cluster()
    if cc is not None:
        return HttpK8sClient(cc)
    home = os.path.expanduser("~/.kube/config")
    if os.path.exists(home):
        return HttpK8sClient(load_kubeconfig(home))
    raise RuntimeError(
        "no Kubernetes config found (not in-cluster, no ~/.kube/config, "
        "no --kubeconfig)"
    )


class HttpK8sClient(K8sClient):
    def __init__(self, cc: ClusterConfig, timeout_s: float = 30.0):
        self.cc = cc
        headers = {"Accept": "application/json"}
        if cc.token:
            headers["Authorization"] = f"Bearer {cc.token}"
        verify: Any
        if not cc.verify:
            verify = False
        elif cc.ca_path:
            ctx = ssl.create_default_context(cafile=cc.ca_path)
            if cc.client_cert and cc.client_key:
                ctx.load_cert_chain(cc.client_cert, cc.client_key)
            verify = ctx
        else:
            verify = True
        cert = None
        if cc.client_cert and cc.client_key and not cc.ca_path:
            cert = (cc.client_cert, cc.client_key)
        self._http = httpx.Client(
            base_url=cc.server, headers=headers, verify=verify, cert=cert,
            timeout=timeout_s,
        )
        self._leases_supported: Optional[bool] = None

    def close(self) -> None:
        self._http.close()

    # ---- plumbing ----

    @staticmethod
    def _check(resp: httpx.Response) -> Dict[str, Any]:
        if resp.status_code == 404:
            raise NotFoundError(resp.text[:200])
        if resp.status_code == 409:
            raise ConflictError(resp.text[:200])
        if resp.status_code == 410:
            raise GoneError(resp.text[:200])
        if resp.status_code >= 400:
            raise ApiError(resp.status_code, resp.text[:500])
        if resp.status_code == 204 or not resp.content:
            return {}
        return resp.json()

    def _refresh_token(self) -> bool:
        """Re-read a rotated service-account token (in-cluster only).
        Returns True when the Authorization header changed."""
        if not self.cc.token_path:
            return False
        try:
            with open(self.cc.token_path, "r", encoding="utf-8") as fh:
                token = fh.read().strip()
        except OSError:
            return False
        if not token or token == self.cc.token:
            return False
        self.cc.token = token
        self._http.headers["Authorization"] = f"Bearer {token}"
        log.info("service-account token refreshed")
        return True

    def _request(self, method: str, path: str, *, params=None, json_body=None,
                 content_type: str = "application/json") -> Dict[str, Any]:
        def attempt() -> Dict[str, Any]:
            headers = {"Content-Type": content_type} if json_body is not None else {}
            resp = self._http.request(
                method, path, params=params,
                content=json.dumps(json_body) if json_body is not None else None,
                headers=headers,
            )
            if resp.status_code == 401 and self._refresh_token():
                resp = self._http.request(
                    method, path, params=params,
                    content=json.dumps(json_body)
                    if json_body is not None else None,
                    headers=headers,
                )
            return self._check(resp)

        # 404/409 are terminal-valid outcomes (reference treats 200/404 as
        # terminal, runpod_client.go:288-296).
        return retry(attempt, terminal=lambda e: isinstance(e, ApiError))

    # ---- pods ----

    def list_pods(self, namespace=None, field_selector="", label_selector="") -> List[Dict[str, Any]]:
        path = f"/api/v1/namespaces/{namespace}/pods" if namespace else "/api/v1/pods"
        params = {}
        if field_selector:
            params["fieldSelector"] = field_selector
        if label_selector:
            params["labelSelector"] = label_selector
        return self._request("GET", path, params=params).get("items", [])

    def list_pods_with_rv(self, namespace=None, field_selector="",
                          label_selector=""):
        path = f"/api/v1/namespaces/{namespace}/pods" if namespace else "/api/v1/pods"
        params = {}
        if field_selector:
            params["fieldSelector"] = field_selector
        if label_selector:
            params["labelSelector"] = label_selector
        body = self._request("GET", path, params=params)
        return (body.get("items", []),
                body.get("metadata", {}).get("resourceVersion", ""))

    def get_pod(self, namespace: str, name: str) -> Dict[str, Any]:
        return self._request("GET", f"/api/v1/namespaces/{namespace}/pods/{name}")

    def create_pod(self, namespace: str, pod: Dict[str, Any]) -> Dict[str, Any]:
        return self._request("POST", f"/api/v1/namespaces/{namespace}/pods",
                             json_body=pod)

    def update_pod(self, namespace: str, pod: Dict[str, Any]) -> Dict[str, Any]:
        name = pod["metadata"]["name"]
        return self._request("PUT", f"/api/v1/namespaces/{namespace}/pods/{name}",
                             json_body=pod)

    def patch_pod(self, namespace: str, name: str, patch: Dict[str, Any]) -> Dict[str, Any]:
        return self._request(
            "PATCH", f"/api/v1/namespaces/{namespace}/pods/{name}",
            json_body=patch, content_type="application/strategic-merge-patch+json",
        )

    def patch_pod_status(self, namespace: str, name: str, patch: Dict[str, Any]) -> Dict[str, Any]:
        return self._request(
            "PATCH", f"/api/v1/namespaces/{namespace}/pods/{name}/status",
            json_body=patch, content_type="application/strategic-merge-patch+json",
        )

    def delete_pod(self, namespace: str, name: str, grace_period_s=None) -> None:
        params = {}
        if grace_period_s is not None:
            params["gracePeriodSeconds"] = str(grace_period_s)
        self._request("DELETE", f"/api/v1/namespaces/{namespace}/pods/{name}",
                      params=params)

    def watch_pods(self, namespace=None, field_selector="", resource_version="",
                   timeout_s: float = 60.0) -> Iterator[WatchEvent]:
        path = f"/api/v1/namespaces/{namespace}/pods" if namespace else "/api/v1/pods"
        params = {"watch": "true", "timeoutSeconds": str(int(timeout_s)),
                  "allowWatchBookmarks": "true"}
        if field_selector:
            params["fieldSelector"] = field_selector
        if resource_version:
            params["resourceVersion"] = resource_version
        with self._http.stream("GET", path, params=params,
                               timeout=timeout_s + 10) as resp:
            if resp.status_code == 401:
                # rotated SA token: refresh for the caller's retry (the
                # informer re-establishes the watch on error)
                self._refresh_token()
            if resp.status_code >= 400:
                resp.read()
                self._check(resp)
            for line in resp.iter_lines():
                if not line:
                    continue
                try:
                    event = json.loads(line)
                except json.JSONDecodeError:
                    continue
                ev_type = event.get("type", "")
                obj = event.get("object", {})
                if ev_type in ("ADDED", "MODIFIED", "DELETED", "BOOKMARK"):
                    yield ev_type, obj
                elif ev_type == "ERROR":
                    # apiserver reports expired RVs as an in-stream Status
                    # with code 410 (Gone) — the watcher must relist
                    if obj.get("code") == 410:
                        raise GoneError(obj.get("message", ""))
                    raise ApiError(int(obj.get("code", 500)),
                                   obj.get("message", "watch error"))

    # ---- nodes ----

    def get_node(self, name: str) -> Dict[str, Any]:
        return self._request("GET", f"/api/v1/nodes/{name}")

    def create_node(self, node: Dict[str, Any]) -> Dict[str, Any]:
        return self._request("POST", "/api/v1/nodes", json_body=node)

    def update_node(self, node: Dict[str, Any]) -> Dict[str, Any]:
        name = node["metadata"]["name"]
        return self._request("PUT", f"/api/v1/nodes/{name}", json_body=node)

    def patch_node_status(self, name: str, patch: Dict[str, Any]) -> Dict[str, Any]:
        return self._request(
            "PATCH", f"/api/v1/nodes/{name}/status",
            json_body=patch, content_type="application/strategic-merge-patch+json",
        )

    def delete_node(self, name: str) -> None:
        self._request("DELETE", f"/api/v1/nodes/{name}")

    # ---- leases ----

    def leases_supported(self) -> bool:
        """Discoverability probe (reference main.go:196-204 checks the
        coordination group before enabling leases)."""
        if self._leases_supported is None:
            try:
                self._request("GET", "/apis/coordination.k8s.io/v1")
                self._leases_supported = True
            except ApiError:
                self._leases_supported = False
        return self._leases_supported

    def get_lease(self, namespace: str, name: str) -> Dict[str, Any]:
        return self._request(
            "GET",
            f"/apis/coordination.k8s.io/v1/namespaces/{namespace}/leases/{name}",
        )

    def create_lease(self, namespace: str, lease: Dict[str, Any]) -> Dict[str, Any]:
        return self._request(
            "POST", f"/apis/coordination.k8s.io/v1/namespaces/{namespace}/leases",
            json_body=lease,
        )

    def update_lease(self, namespace: str, lease: Dict[str, Any]) -> Dict[str, Any]:
        name = lease["metadata"]["name"]
        return self._request(
            "PUT",
            f"/apis/coordination.k8s.io/v1/namespaces/{namespace}/leases/{name}",
            json_body=lease,
        )

    # ---- secrets / configmaps / jobs ----

    def get_secret(self, namespace: str, name: str) -> Dict[str, Any]:
        return self._request("GET", f"/api/v1/namespaces/{namespace}/secrets/{name}")

    def get_configmap(self, namespace: str, name: str) -> Dict[str, Any]:
        return self._request("GET", f"/api/v1/namespaces/{namespace}/configmaps/{name}")

    def get_job(self, namespace: str, name: str) -> Dict[str, Any]:
        return self._request("GET", f"/apis/batch/v1/namespaces/{namespace}/jobs/{name}")

    # ---- events / auth ----

    def update_event(self, namespace: str, event: Dict[str, Any]) -> Dict[str, Any]:
        name = event.get("metadata", {}).get("name", "")
        return self._request(
            "PUT", f"/api/v1/namespaces/{namespace}/events/{name}",
            json_body=event)

    def create_event(self, namespace: str, event: Dict[str, Any]) -> Dict[str, Any]:
        return self._request("POST", f"/api/v1/namespaces/{namespace}/events",
                             json_body=event)

    def self_subject_review(self) -> Dict[str, Any]:
        try:
            return self._request(
                "POST", "/apis/authentication.k8s.io/v1/selfsubjectreviews",
                json_body={"apiVersion": "authentication.k8s.io/v1",
                           "kind": "SelfSubjectReview"},
            )
        except ApiError:
            return {}
