"""MI355X-native virtual-kubelet provider.

A brand-new framework with the capabilities of BSVogler/k8s-runpod-kubelet
(reference at /root/reference): it registers a virtual node in a Kubernetes
cluster and implements the full PodLifecycleHandler / NodeProvider surface —
but instead of translating pod specs into RunPod cloud REST/GraphQL calls
(reference pkg/virtual_kubelet/runpod_client.go), it places pods directly onto
the 8 GPUs of one MI355X node:

- GPU inventory via a native C++ probe over KFD topology + amdgpu DRM sysfs
  (``ops/csrc/probe.cpp``), advertising ``amd.com/gpu`` capacity with 288 GB
  HBM3E per GPU.
- Per-pod GPU binding through ``ROCR_VISIBLE_DEVICES``/``HIP_VISIBLE_DEVICES``
  plus render-node device scoping, with xGMI-topology-aware multi-GPU set
  selection (``gpu/binder.py``).
- An event-driven status loop (pidfd + epoll in ``ops/csrc/launcher.cpp``)
  replacing the reference's 10 s/30 s cloud-polling tickers
  (reference kubelet.go:292-303, :713-731).

The kubectl-facing contract (annotations, taint, Helm values) stays compatible
with the reference.
"""

from .version import __version__  # noqa: F401

PROVIDER_NAME = "amd-mi355x"
DEFAULT_NODE_NAME = "virtual-runpod"  # parity: reference main.go:64
