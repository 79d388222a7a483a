"""Fake /sys tree builder for hermetic tests and the CPU bench path.

The reference tests against checked-in fake sysfs trees (reference:
testdata/topology-parsing, topo-mi210-xgmi-pcie, topo-mi300-cpx; see
SURVEY.md §4).  We synthesize equivalent trees programmatically, which lets
tests build arbitrary gfx950 topologies (8-GPU single-hive MI355X, CPX
fan-out, mixed xGMI/PCIe) with a few lines instead of hundreds of fixture
files.
"""

from __future__ import annotations

import os
from typing import Dict, List, Optional

from ..topology.sysfs import SysPaths

# MI355X (gfx950) defaults
MI355X_VRAM_BYTES = 309220868096           # 288 GB HBM3E (value read from a real MI355X kfd node)
MI355X_SIMD_COUNT = 1024                   # 256 CUs x 4 SIMDs
MI355X_SIMD_PER_CU = 4
MI355X_GFX_TARGET = 90500
# kfd reports xGMI min/max_bandwidth in MB/s; 7 point-to-point links per GPU
# at ~153 GB/s each on an 8-GPU hive.
MI355X_XGMI_BW_MBPS = 153600
MI355X_XGMI_WEIGHT = 15
PCIE_WEIGHT = 40


def _write(path: str, content: str) -> None:
    os.makedirs(os.path.dirname(path), exist_ok=True)
    with open(path, "w") as f:
        f.write(content)


def _props_text(props: Dict[str, int]) -> str:
    return "".join(f"{k} {v}\n" for k, v in props.items())


class FakeSysfs:
    """Builds a fake /sys tree under `root` readable through SysPaths(root)."""

    def __init__(self, root: str):
        self.root = root
        self.paths = SysPaths(root)
        self._link_counters: Dict[int, Dict[str, int]] = {}
        os.makedirs(self.paths.kfd_topology_nodes, exist_ok=True)
        os.makedirs(self.paths.amdgpu_pci, exist_ok=True)

    # ---------- low-level ----------

    def _node_dir(self, node_id: int) -> str:
        return os.path.join(self.paths.kfd_topology_nodes, str(node_id))

    def add_kfd_node(
        self,
        node_id: int,
        props: Dict[str, int],
        mem_banks: Optional[List[Dict[str, int]]] = None,
    ) -> None:
        _write(os.path.join(self._node_dir(node_id), "properties"), _props_text(props))
        for i, bank in enumerate(mem_banks or []):
            _write(
                os.path.join(self._node_dir(node_id), "mem_banks", str(i), "properties"),
                _props_text(bank),
            )

    def add_link(
        self,
        node_from: int,
        node_to: int,
        link_type: int = 11,
        weight: int = MI355X_XGMI_WEIGHT,
        bandwidth: int = MI355X_XGMI_BW_MBPS,
        kind: str = "io_links",
        symmetric: bool = True,
    ) -> None:
        """Write a link entry under node_from (and mirrored, like real kfd)."""
        endpoints = [(node_from, node_to)]
        if symmetric:
            endpoints.append((node_to, node_from))
        for frm, to in endpoints:
            counters = self._link_counters.setdefault(frm, {})
            idx = counters.get(kind, 0)
            counters[kind] = idx + 1
            props = {
                "type": link_type,
                "version_major": 0,
                "version_minor": 0,
                "node_from": frm,
                "node_to": to,
                "weight": weight,
                "min_latency": 0,
                "max_latency": 0,
                "min_bandwidth": bandwidth,
                "max_bandwidth": bandwidth,
                "flags": 1,
            }
            _write(
                os.path.join(self._node_dir(frm), kind, str(idx), "properties"),
                _props_text(props),
            )

    # ---------- higher-level building blocks ----------

    def add_cpu_node(self, node_id: int, cores: int = 96) -> None:
        self.add_kfd_node(
            node_id,
            {
                "cpu_cores_count": cores,
                "simd_count": 0,
                "mem_banks_count": 1,
                "io_links_count": 0,
                "gfx_target_version": 0,
                "drm_render_minor": 0,
                "location_id": 0,
                "domain": 0,
            },
            mem_banks=[{"heap_type": 0, "size_in_bytes": 1 << 40}],
        )

    def add_gpu_kfd_node(
        self,
        node_id: int,
        render_minor: int,
        location_id: int,
        domain: int = 0,
        hive_id: int = 0,
        simd_count: int = MI355X_SIMD_COUNT,
        simd_per_cu: int = MI355X_SIMD_PER_CU,
        gfx_target_version: int = MI355X_GFX_TARGET,
        vram_bytes: int = MI355X_VRAM_BYTES,
        numa_node: int = 0,
        device_id: int = 0x75A3,
    ) -> None:
        self.add_kfd_node(
            node_id,
            {
                "cpu_cores_count": 0,
                "simd_count": simd_count,
                "mem_banks_count": 1,
                "simd_per_cu": simd_per_cu,
                "max_waves_per_simd": 8,
                "wave_front_size": 64,
                "gfx_target_version": gfx_target_version,
                "vendor_id": 0x1002,
                "device_id": device_id,
                "location_id": location_id,
                "domain": domain,
                "drm_render_minor": render_minor,
                "hive_id": hive_id,
                "numa_node": numa_node,
                "num_sdma_engines": 2,
                "num_xcc": 8,
            },
            mem_banks=[
                {
                    "heap_type": 1,
                    "size_in_bytes": vram_bytes,
                    "flags": 0,
                    "width": 8192,
                    "mem_clk_max": 2400,
                }
            ],
        )

    def _write_drm_device_entries(self, card: int, device_id: int, product: str) -> None:
        dev_dir = self.paths.drm_card_device(card)
        _write(os.path.join(dev_dir, "vendor"), "0x1002\n")
        _write(os.path.join(dev_dir, "device"), f"0x{device_id:04x}\n")
        _write(os.path.join(dev_dir, "product_name"), f"{product}\n")
        _write(os.path.join(dev_dir, "driver/module/version"), "6.14.14\n")
        _write(os.path.join(dev_dir, "driver/module/srcversion"), "FAKE123456789\n")

    def add_physical_gpu(
        self,
        index: int,
        node_id: int,
        compute_partition: str = "SPX",
        memory_partition: str = "NPS1",
        numa_node: int = 0,
        hive_id: int = 0,
        partition_caps: bool = True,
        product: str = "AMD Instinct MI355 OAM",
        device_id: int = 0x75A3,
        **gpu_node_kwargs,
    ) -> str:
        """Add one physical GPU: PCI dir + drm minors + its kfd node.

        Returns the PCI address (= the device-plugin device ID).
        """
        bus = 0x0C + index
        pci_addr = f"0000:{bus:02x}:00.0"
        card = index
        render_minor = 128 + index
        location_id = bus << 8

        pci_dir = os.path.join(self.paths.amdgpu_pci, pci_addr)
        _write(os.path.join(pci_dir, "current_compute_partition"), f"{compute_partition}\n")
        _write(os.path.join(pci_dir, "current_memory_partition"), f"{memory_partition}\n")
        _write(os.path.join(pci_dir, "numa_node"), f"{numa_node}\n")
        if partition_caps:
            _write(os.path.join(pci_dir, "available_compute_partition"), "SPX, DPX, QPX, CPX\n")
            _write(os.path.join(pci_dir, "available_memory_partition"), "NPS1, NPS2\n")
        os.makedirs(os.path.join(pci_dir, "drm", f"card{card}"), exist_ok=True)
        os.makedirs(os.path.join(pci_dir, "drm", f"renderD{render_minor}"), exist_ok=True)

        self._write_drm_device_entries(card, device_id, product)
        self.add_gpu_kfd_node(
            node_id,
            render_minor=render_minor,
            location_id=location_id,
            hive_id=hive_id,
            numa_node=numa_node,
            device_id=device_id,
            **gpu_node_kwargs,
        )
        return pci_addr

    def add_partition(
        self,
        xcp_index: int,
        node_id: int,
        parent_index: int,
        card: int,
        render_minor: int,
        numa_node: int = 0,
        hive_id: int = 0,
        in_kfd: bool = True,
        **gpu_node_kwargs,
    ) -> str:
        """Add one amdgpu_xcp partition whose kfd node shares the parent's
        location_id/domain (the devID join, reference: amdgpu.go:141-143).

        in_kfd=False creates the platform device without a kfd node — an
        invalid renderD that discovery must skip (amdgpu.go:269-271).
        """
        name = f"amdgpu_xcp_{xcp_index}"
        plat_dir = os.path.join(self.paths.platform_devices, name)
        os.makedirs(os.path.join(plat_dir, "drm", f"card{card}"), exist_ok=True)
        os.makedirs(os.path.join(plat_dir, "drm", f"renderD{render_minor}"), exist_ok=True)
        if in_kfd:
            bus = 0x0C + parent_index
            self.add_gpu_kfd_node(
                node_id,
                render_minor=render_minor,
                location_id=bus << 8,
                hive_id=hive_id,
                numa_node=numa_node,
                **gpu_node_kwargs,
            )
        return name


def build_mi355x_node(
    root: str,
    n_gpus: int = 8,
    partitions_per_gpu: int = 1,
    compute_partition: str = "SPX",
    memory_partition: str = "NPS1",
    hive_id: int = 7455128887705989632,
    numa_per_gpu: Optional[List[int]] = None,
) -> FakeSysfs:
    """One 8*MI355X node: CPU nodes 0-1, GPUs fully connected over xGMI.

    With partitions_per_gpu > 1 each physical GPU additionally fans out into
    that many amdgpu_xcp partitions (CPX-style); the physical GPU keeps the
    first kfd node like real MI300/MI355 CPX trees.
    """
    fs = FakeSysfs(root)
    fs.add_cpu_node(0)
    fs.add_cpu_node(1)

    numa_map = numa_per_gpu or [0 if i < n_gpus // 2 else 1 for i in range(n_gpus)]
    gpu_nodes: List[int] = []  # kfd node ids of every schedulable device
    next_node = 2
    next_xcp = 0
    next_minor = 128 + n_gpus  # partition renderDs start after the physical ones

    for i in range(n_gpus):
        # in CPX-style trees the physical GPU keeps the FIRST kfd node,
        # and that node reports per-partition values like every other
        # partition node (real MI300/MI355 CPX behavior; cf. the
        # topo-mi300-cpx reference fixture where all 8 nodes of a GPU
        # carry the divided simd counts)
        fs.add_physical_gpu(
            i,
            node_id=next_node,
            compute_partition=compute_partition,
            memory_partition=memory_partition,
            numa_node=numa_map[i],
            hive_id=hive_id,
            vram_bytes=MI355X_VRAM_BYTES // partitions_per_gpu,
            simd_count=MI355X_SIMD_COUNT // partitions_per_gpu,
        )
        gpu_nodes.append(next_node)
        next_node += 1
        for _ in range(1, partitions_per_gpu):
            fs.add_partition(
                next_xcp,
                node_id=next_node,
                parent_index=i,
                card=n_gpus + next_xcp,
                render_minor=next_minor,
                numa_node=numa_map[i],
                hive_id=hive_id,
                vram_bytes=MI355X_VRAM_BYTES // partitions_per_gpu,
                simd_count=MI355X_SIMD_COUNT // partitions_per_gpu,
            )
            gpu_nodes.append(next_node)
            next_node += 1
            next_xcp += 1
            next_minor += 1

    # Full xGMI mesh between every pair of schedulable kfd nodes (on real
    # MI355X CPX trees every partition lists links to every other node).
    for a_idx in range(len(gpu_nodes)):
        for b_idx in range(a_idx + 1, len(gpu_nodes)):
            fs.add_link(gpu_nodes[a_idx], gpu_nodes[b_idx], link_type=11)

    return fs
