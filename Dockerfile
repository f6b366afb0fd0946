# Two-stage image for the MI355X-native Kata xPU device plugin.
#
# Reference analog: Dockerfile:27-70 (2-stage on a CUDA base that linked
# nothing from CUDA). Here: a ROCm builder compiles the gfx950 HIP probe
# extension (optional at runtime — burn-in only), the runtime stage is a
# plain slim Python image: the control plane itself needs NO ROCm.

FROM rocm/dev-ubuntu-22.04:7.2 AS builder
RUN apt-get update && apt-get install -y --no-install-recommends \
        python3 python3-pip python3-dev g++ && rm -rf /var/lib/apt/lists/*
RUN pip3 install --no-cache-dir pybind11 setuptools
WORKDIR /src
COPY setup.py ./
COPY native ./native
COPY probes ./probes
COPY kata_xpu_device_plugin_amd ./kata_xpu_device_plugin_amd
RUN python3 setup.py build_ext --inplace && \
    python3 -c "from setup import build_hip; build_hip('gfx950')"

FROM python:3.10-slim
# Control-plane runtime deps only. Optional extras on the node image:
#   - amdsmi python bindings (ship with ROCm) for the amd-smi health poller
#   - ROCm runtime libs if tools/burnin probes are used in-container
RUN pip install --no-cache-dir grpcio protobuf pyyaml prometheus_client
# pci.ids fallback for naming unknown silicon (reference side-loads it to
# /usr/pci.ids, Dockerfile:66; we use the distro location first)
RUN apt-get update && apt-get install -y --no-install-recommends pciutils \
    && rm -rf /var/lib/apt/lists/*
COPY --from=builder /src/kata_xpu_device_plugin_amd /opt/kxdp/kata_xpu_device_plugin_amd
ENV PYTHONPATH=/opt/kxdp
ENTRYPOINT ["python", "-m", "kata_xpu_device_plugin_amd"]
