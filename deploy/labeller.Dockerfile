# AMD GPU node labeller image (MI355X-native).
# The bundled amdgpu.ids ships inside the package (native/amdgpu.ids), so no
# /usr/share/libdrm install is needed (cf. reference labeller.Dockerfile:34).
FROM rocm/dev-ubuntu-22.04:7.0 AS build
RUN apt-get update && apt-get install -y --no-install-recommends \
        python3 python3-pip python3-dev g++ && \
    pip3 install --no-cache-dir pybind11
COPY . /src
WORKDIR /src
RUN python3 -c "from k8s_device_plugin_amd.native.build import build_drmctl; build_drmctl()"

FROM ubuntu:22.04
RUN apt-get update && apt-get install -y --no-install-recommends \
        python3 python3-pip && \
    pip3 install --no-cache-dir requests && \
    rm -rf /var/lib/apt/lists/*
COPY --from=build /src/k8s_device_plugin_amd /opt/amdxdp/k8s_device_plugin_amd
ENV PYTHONPATH=/opt/amdxdp
ENTRYPOINT ["python3", "-c", "import sys; from k8s_device_plugin_amd.cli import labeller_main; sys.exit(labeller_main())"]
