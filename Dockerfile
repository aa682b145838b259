# Two-stage build (reference Dockerfile:1-22 pattern: builder -> minimal
# runtime). The builder compiles the native probe/launcher extension and the
# gfx950 podworker with hipcc; the runtime stage carries only ROCm runtime
# libs + python.
FROM rocm/dev-ubuntu-22.04:7.2 AS builder
RUN apt-get update && apt-get install -y --no-install-recommends \
        python3 python3-pip g++ && rm -rf /var/lib/apt/lists/*
RUN pip3 install --no-cache-dir pybind11 setuptools
WORKDIR /src
COPY k8s_runpod_kubelet_amd/ k8s_runpod_kubelet_amd/
COPY pyproject.toml .
ENV PYTORCH_ROCM_ARCH=gfx950
RUN python3 -m k8s_runpod_kubelet_amd.ops.build

FROM rocm/rocm-runtime-ubuntu-22.04:7.2
RUN apt-get update && apt-get install -y --no-install-recommends \
        python3 python3-pip && rm -rf /var/lib/apt/lists/*
RUN pip3 install --no-cache-dir pyyaml httpx prometheus-client
WORKDIR /app
COPY --from=builder /src/k8s_runpod_kubelet_amd/ k8s_runpod_kubelet_amd/
ENV PYTHONPATH=/app
# non-root is not possible here: the kubelet manages /dev/kfd bindings and
# cgroups (the reference runs distroless nonroot because its backend is a
# remote cloud API).
ENTRYPOINT ["python3", "-m", "k8s_runpod_kubelet_amd.cli"]
