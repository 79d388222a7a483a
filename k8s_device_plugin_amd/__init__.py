"""MI355X-native Kubernetes device plugin and node labeller for AMD Instinct GPUs.

A from-scratch implementation with the capabilities of ROCm/k8s-device-plugin
(reference surveyed in SURVEY.md): DevicePlugin v1beta1 gRPC (Register /
ListAndWatch / Allocate / GetPreferredAllocation / PreStartContainer), a
hand-written gfx950 kfd-sysfs topology walker, an xGMI-hive-aware preferred
allocator, per-GPU health (kfd scan + amd-metrics-exporter gRPC + an on-GPU
CDNA4 deep-probe kernel), and a node labeller.

Native components (replacing the reference's libdrm/hwloc cgo,
reference: internal/pkg/amdgpu/amdgpu.go:21-27, internal/pkg/hwloc/hwloc.go):
  - native/_drmctl.so    raw DRM_IOCTL_AMDGPU_INFO ioctls (C++/pybind11)
  - native/_healthprobe.so gfx950 MFMA/LDS/HBM deep health probe (HIP)
"""

__version__ = "0.2.0"
