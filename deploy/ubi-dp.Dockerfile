# UBI-based device plugin image (the reference ships ubi-dp.Dockerfile for
# OpenShift-certified deployments).
FROM registry.access.redhat.com/ubi9/ubi:latest AS build
RUN dnf install -y python3.11 python3.11-pip python3.11-devel gcc-c++ && \
    pip3.11 install --no-cache-dir pybind11
COPY . /src
WORKDIR /src
RUN python3.11 -c "from k8s_device_plugin_amd.native.build import build_drmctl, build_fastserver; build_drmctl(); build_fastserver()"

FROM registry.access.redhat.com/ubi9/ubi-minimal:latest
RUN microdnf install -y python3.11 python3.11-pip libnghttp2 && \
    pip3.11 install --no-cache-dir grpcio protobuf && \
    microdnf clean all
COPY --from=build /src/k8s_device_plugin_amd /opt/amdxdp/k8s_device_plugin_amd
ENV PYTHONPATH=/opt/amdxdp
LABEL name="amd-gpu-device-plugin" vendor="AMD" \
      summary="AMD Instinct MI355X device plugin for Kubernetes"
ENTRYPOINT ["python3.11", "-m", "k8s_device_plugin_amd.cli"]
