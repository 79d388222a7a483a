# UBI-based node labeller image (cf. reference ubi-labeller.Dockerfile).
FROM registry.access.redhat.com/ubi9/ubi:latest AS build
RUN dnf install -y python3.11 python3.11-pip python3.11-devel gcc-c++ && \
    pip3.11 install --no-cache-dir pybind11
COPY . /src
WORKDIR /src
RUN python3.11 -c "from k8s_device_plugin_amd.native.build import build_drmctl; build_drmctl()"

FROM registry.access.redhat.com/ubi9/ubi-minimal:latest
RUN microdnf install -y python3.11 python3.11-pip && \
    pip3.11 install --no-cache-dir requests && \
    microdnf clean all
COPY --from=build /src/k8s_device_plugin_amd /opt/amdxdp/k8s_device_plugin_amd
ENV PYTHONPATH=/opt/amdxdp
LABEL name="amd-gpu-node-labeller" vendor="AMD" \
      summary="AMD Instinct MI355X node labeller for Kubernetes"
ENTRYPOINT ["python3.11", "-c", "import sys; from k8s_device_plugin_amd.cli import labeller_main; sys.exit(labeller_main())"]
