"""GPU discovery: physical PCI GPUs plus amdgpu_xcp partition fan-out.

Behavioral parity with the reference's GetAMDGPUs walk (reference:
internal/pkg/amdgpu/amdgpu.go:156-279), including its subtle rules:
  - physical GPUs come from /sys/module/amdgpu/drivers/pci:amdgpu/<pciaddr>,
    keyed by the PCI address (the device-plugin device ID seen by kubelet);
  - numa_node must be readable for a physical GPU or it is skipped;
  - partitions come from /sys/devices/platform/amdgpu_xcp_N, keyed
    "amdgpu_xcp_N", inheriting partition types and NUMA node from the
    physical device with the same devID (amdgpu.go:250-260);
  - an amdgpu_xcp whose renderD is not in the kfd topology is invalid and
    skipped, as is one with no NUMA inheritance (amdgpu.go:266-274).
"""

from __future__ import annotations

import glob
import os
import re
from dataclasses import dataclass, asdict
from typing import Dict, Optional

from .kfd import KFDTopology
from .sysfs import SysPaths, read_stripped


class DriverUnavailableError(RuntimeError):
    """amdgpu kernel driver not present (reference exits with code 2,
    amdgpu.go:157-160)."""


@dataclass
class GPUDevice:
    """One schedulable device: a whole GPU or one compute partition."""

    id: str                  # PCI address or "amdgpu_xcp_N" (kubelet device ID)
    card: int = 0            # /dev/dri/card<N>
    render_d: int = 128      # /dev/dri/renderD<M>
    dev_id: str = ""         # PCI-ish id shared by all partitions of one GPU
    compute_partition: str = ""   # e.g. "spx", "cpx" (lowercased)
    memory_partition: str = ""    # e.g. "nps1", "nps4" (lowercased)
    numa_node: int = -1
    node_id: int = 0         # kfd topology node index
    # False when the device's renderD has no readable kfd topology node —
    # e.g. a peer GPU masked away by the container's cgroup (observed on
    # MI355X boxes: peer nodes' properties read "Operation not permitted").
    # Such a device is visible on PCI but not schedulable; the plugin
    # advertises it Unhealthy.
    kfd_backed: bool = True

    @property
    def is_partition(self) -> bool:
        return "amdgpu_xcp" in self.id

    @property
    def partition_key(self) -> str:
        """'<compute>_<memory>' bucket used for mixed resource naming
        (reference: plugin.go:277, amdgpu.go:289)."""
        return f"{self.compute_partition}_{self.memory_partition}"

    def as_dict(self) -> dict:
        return asdict(self)


_PCI_ADDR_RE = re.compile(r"^[0-9a-fA-F]{4}:[0-9a-fA-F:.]+$")


def _scan_drm_dir(dev_path: str) -> tuple[Optional[int], Optional[int]]:
    """Return (card, renderD) minors found under <dev_path>/drm/*."""
    card = render_d = None
    for entry in glob.glob(os.path.join(dev_path, "drm", "*")):
        name = os.path.basename(entry)
        if name.startswith("card"):
            try:
                card = int(name[4:])
            except ValueError:
                pass
        elif name.startswith("renderD"):
            try:
                render_d = int(name[7:])
            except ValueError:
                pass
    return card, render_d


def discover_gpus(
    paths: SysPaths = SysPaths(),
    topology: Optional[KFDTopology] = None,
    strict: bool = True,
) -> Dict[str, GPUDevice]:
    """Enumerate schedulable AMD GPU devices on this node.

    Returns {device_id: GPUDevice}.  With strict=True a missing amdgpu driver
    raises DriverUnavailableError (callers exit 2, matching the reference);
    strict=False returns {} (the reference's test kill-switch,
    amdgpu.go:150-153, expressed as a parameter instead of a global).
    """
    if not os.path.isdir(paths.amdgpu_driver):
        if strict:
            raise DriverUnavailableError(
                f"amdgpu driver unavailable: {paths.amdgpu_driver} missing"
            )
        return {}

    topo = topology if topology is not None else KFDTopology.load(paths)
    rd_to_devid = topo.render_minor_to_dev_id()
    rd_to_nodeid = topo.render_minor_to_node_id()

    devices: Dict[str, GPUDevice] = {}

    # --- physical GPUs ---
    for path in sorted(glob.glob(os.path.join(paths.amdgpu_pci, "*"))):
        pci_addr = os.path.basename(path)
        if not _PCI_ADDR_RE.match(pci_addr):
            continue

        compute = read_stripped(os.path.join(path, "current_compute_partition"))
        memory = read_stripped(os.path.join(path, "current_memory_partition"))
        numa_raw = read_stripped(os.path.join(path, "numa_node"))
        if numa_raw is None:
            continue  # reference skips when numa_node is unreadable (amdgpu.go:203-206)
        try:
            numa_node = int(numa_raw)
        except ValueError:
            continue

        card, render_d = _scan_drm_dir(path)
        dev = GPUDevice(
            id=pci_addr,
            card=card if card is not None else 0,
            render_d=render_d if render_d is not None else 128,
            compute_partition=(compute or "").lower(),
            memory_partition=(memory or "").lower(),
            numa_node=numa_node,
        )
        if render_d is not None and render_d in rd_to_devid:
            dev.dev_id = rd_to_devid[render_d]
            dev.node_id = rd_to_nodeid.get(render_d, 0)
        else:
            dev.kfd_backed = False
        devices[pci_addr] = dev

    # --- compute partitions (amdgpu_xcp platform devices) ---
    for path in sorted(glob.glob(os.path.join(paths.platform_devices, "amdgpu_xcp_*"))):
        xcp_name = os.path.basename(path)
        card, render_d = _scan_drm_dir(path)
        if render_d is None or render_d not in rd_to_devid:
            # not a valid partition per kfd topology (amdgpu.go:269-271)
            continue
        dev_id = rd_to_devid[render_d]

        compute = memory = ""
        numa_node = -1
        for parent in devices.values():
            if (
                parent.dev_id == dev_id
                and parent.compute_partition
                and parent.memory_partition
            ):
                compute = parent.compute_partition
                memory = parent.memory_partition
                numa_node = parent.numa_node
                break
        if numa_node == -1:
            continue  # no valid parent to inherit from (amdgpu.go:272-274)

        devices[xcp_name] = GPUDevice(
            id=xcp_name,
            card=card if card is not None else 0,
            render_d=render_d,
            dev_id=dev_id,
            compute_partition=compute,
            memory_partition=memory,
            numa_node=numa_node,
            node_id=rd_to_nodeid.get(render_d, 0),
        )

    return devices


def unique_partition_config_count(devices: Dict[str, GPUDevice]) -> Dict[str, int]:
    """Count devices per '<compute>_<memory>' partition config
    (reference: amdgpu.go:281-296)."""
    counts: Dict[str, int] = {}
    for dev in devices.values():
        if dev.compute_partition and dev.memory_partition:
            counts[dev.partition_key] = counts.get(dev.partition_key, 0) + 1
    return counts


def is_homogeneous(devices: Dict[str, GPUDevice]) -> bool:
    """Homogeneous = zero or one distinct partition config
    (reference: amdgpu.go:298-304)."""
    return len(unique_partition_config_count(devices)) <= 1


def _first_pci_gpu_path(paths: SysPaths) -> Optional[str]:
    matches = sorted(glob.glob(os.path.join(paths.amdgpu_pci, "*")))
    for m in matches:
        if _PCI_ADDR_RE.match(os.path.basename(m)):
            return m
    return None


def is_compute_partition_supported(paths: SysPaths = SysPaths()) -> bool:
    p = _first_pci_gpu_path(paths)
    return p is not None and os.path.exists(
        os.path.join(p, "available_compute_partition")
    )


def is_memory_partition_supported(paths: SysPaths = SysPaths()) -> bool:
    p = _first_pci_gpu_path(paths)
    return p is not None and os.path.exists(
        os.path.join(p, "available_memory_partition")
    )


def count_gpus_from_topology(paths: SysPaths = SysPaths()) -> int:
    """Number of kfd nodes with simd_count > 0
    (reference: plugin.go:123-159)."""
    topo = KFDTopology.load(paths)
    return sum(1 for n in topo.nodes.values() if n.simd_count > 0)


def simple_health_check(
    paths: SysPaths = SysPaths(), topology: Optional[KFDTopology] = None
) -> bool:
    """Node-level health: any kfd node with cpu_cores_count==0 and
    gfx_target_version>0 (reference: plugin.go:161-206)."""
    topo = topology if topology is not None else KFDTopology.load(paths)
    return any(n.is_gpu for n in topo.nodes.values())
