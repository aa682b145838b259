"""Annotation vocabulary — kept kubectl-compatible with the reference
(reference pkg/virtual_kubelet/runpod_client.go:37-52) plus the amd.com
extensions this node-local backend adds."""

# Reference-compatible keys (runpod_client.go:37-52)
POD_ID = "runpod.io/pod-id"
COST_PER_HR = "runpod.io/cost-per-hr"
CLOUD_TYPE = "runpod.io/cloud-type"
TEMPLATE_ID = "runpod.io/templateId"
GPU_MEMORY = "runpod.io/required-gpu-memory"
GPU_MEMORY_ALT = "runpod.io/gpu-memory"  # BASELINE.json config 4 spelling
REGISTRY_AUTH_ID = "runpod.io/container-registry-auth-id"
DATACENTER_IDS = "runpod.io/datacenter-ids"
PORTS = "runpod.io/ports"
EXTERNAL = "runpod.io/external"

# MI355X-local extensions
GPU_IDS = "amd.com/gpu-ids"          # bound GPU indices (ledger reconstruction)
GPU_RESOURCE = "amd.com/gpu"         # resource name (vs reference nvidia.com/gpu)

# Node identity (reference kubelet.go:1111-1117)
TAINT_KEY = "virtual-kubelet.io/provider"
TAINT_VALUE = "runpod"

DEFAULT_MAX_PRICE = 0.5      # reference runpod_client.go:49
DEFAULT_GPU_MEMORY_GB = 16   # reference runpod_client.go:1189

HTTP_AUTO_PORTS = {80, 443, 8080, 8000, 3000, 5000, 8888, 9000}  # runpod_client.go:1214
