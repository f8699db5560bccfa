# Two-stage build (reference Dockerfile:1-41 shape): builder compiles the
# gfx950 node-agent library with hipcc; runtime is a slim python image.
# The controller itself needs no ROCm — only nodes run the HIP agent — so
# the runtime stage stays small and the nodeagent DaemonSet reuses this
# image on GPU nodes where /dev/kfd exists.
#
# Multi-arch: the HIP builder stage is pinned to amd64 (ROCm toolchain)
# — under buildx it builds the gfx950 code object once and every target
# arch copies it in. MI355X nodes are amd64, so the arm64 image variant
# (control-plane-only deployments) carries the library as inert payload;
# the node agent only ever executes on amd64 GPU nodes and fails loudly
# anywhere the library cannot load.
FROM --platform=linux/amd64 rocm/dev-ubuntu-22.04:6.4 AS builder
WORKDIR /src
COPY nodeagent/ nodeagent/
RUN hipcc --offload-arch=gfx950 -O3 -shared -fPIC \
      nodeagent/agent.hip -o /src/libmi355x_nodeagent.so

FROM python:3.10-slim AS runtime
ARG VERSION=0.1.0
LABEL org.opencontainers.image.source=https://github.com/kaito-project/gpu-provisioner-amd
WORKDIR /app
RUN pip install --no-cache-dir httpx prometheus_client uvicorn starlette pyyaml
COPY gpu_provisioner_amd/ gpu_provisioner_amd/
COPY --from=builder /src/libmi355x_nodeagent.so gpu_provisioner_amd/_native/
USER 65532:65532
ENTRYPOINT ["python", "-m", "gpu_provisioner_amd"]
