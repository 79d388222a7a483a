"""Injectable sysfs root paths.

The reference makes individual functions take an optional topo-root override
(reference: internal/pkg/amdgpu/amdgpu.go:103-107,453; allocator.go:28).  We
generalise that single good testability decision: every kernel-interface path
hangs off one SysPaths value, so a whole fake /sys tree can be swapped in for
tests and benchmarks with one argument.
"""

from __future__ import annotations

import os
from dataclasses import dataclass


@dataclass(frozen=True)
class SysPaths:
    """Resolves every kernel interface path the plugin and labeller read.

    `root` is prepended to the absolute kernel paths, so `SysPaths("/tmp/fake")`
    reads `/tmp/fake/sys/class/kfd/...`; the default reads the real `/sys`.
    """

    root: str = "/"

    @property
    def kfd_class(self) -> str:
        # Existence gate for "ROCm kernel driver installed"
        # (reference: cmd/k8s-device-plugin/main.go:141-142).
        return os.path.join(self.root, "sys/class/kfd")

    @property
    def kfd_topology_nodes(self) -> str:
        return os.path.join(self.root, "sys/class/kfd/kfd/topology/nodes")

    @property
    def amdgpu_driver(self) -> str:
        # Existence gate for the amdgpu module (reference: amdgpu.go:157).
        return os.path.join(self.root, "sys/module/amdgpu/drivers")

    @property
    def amdgpu_pci(self) -> str:
        # Physical GPUs: one dir per PCI address (reference: amdgpu.go:166).
        return os.path.join(self.root, "sys/module/amdgpu/drivers/pci:amdgpu")

    @property
    def platform_devices(self) -> str:
        # MI300/MI355-style compute partitions appear as platform devices
        # amdgpu_xcp_N (reference: amdgpu.go:232).
        return os.path.join(self.root, "sys/devices/platform")

    @property
    def drm_class(self) -> str:
        # /sys/class/drm/cardN/device/... (vendor, device id, product_name,
        # driver module version) used by the labeller.
        return os.path.join(self.root, "sys/class/drm")

    def drm_card_device(self, card: int) -> str:
        return os.path.join(self.drm_class, f"card{card}", "device")


def read_text(path: str) -> str | None:
    """Read a small sysfs file; None when absent/unreadable."""
    try:
        with open(path, "r") as f:
            return f.read()
    except OSError:
        return None


def read_stripped(path: str) -> str | None:
    t = read_text(path)
    return t.strip() if t is not None else None
