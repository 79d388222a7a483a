"""Hand-written kfd sysfs topology walker for gfx950.

Parses /sys/class/kfd/kfd/topology/nodes/*/{properties,mem_banks,io_links,
p2p_links}.  Unlike the reference, which re-opens and regex-scans the same
properties file once per queried key (reference:
internal/pkg/amdgpu/amdgpu.go:453-474 called per-property from e.g.
cmd/k8s-node-labeller/main.go:254,296,332), this walker reads each file once
into a dict — one syscall pass over the whole topology, which keeps plugin
startup and the ListAndWatch hot loop cheap on 64-partition CPX nodes.

The kfd properties format is one `<name> <value>` pair per line with integer
values (examples: the fake trees built by testing/fakesysfs.py).
"""

from __future__ import annotations

import glob
import os
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from .sysfs import SysPaths

# gfx_target_version for MI355X (gfx950) is 90500-series; kept as a constant
# for labellers/tests, not used as a filter (the walker is model-agnostic).
GFX950_TARGET_VERSION = 90500


def parse_properties_text(text: str) -> Dict[str, int]:
    """Parse `<name> <value>` lines into a dict; non-integer values skipped."""
    props: Dict[str, int] = {}
    for line in text.splitlines():
        parts = line.split()
        if len(parts) != 2:
            continue
        try:
            # kfd writes decimal; accept 0x-prefixed too (ParseInt base-0
            # semantics, reference: amdgpu.go:468).
            props[parts[0]] = int(parts[1], 0)
        except ValueError:
            continue
    return props


def parse_properties_file(path: str) -> Optional[Dict[str, int]]:
    try:
        with open(path, "r") as f:
            return parse_properties_text(f.read())
    except OSError:
        return None


@dataclass
class KFDLink:
    """One io_link/p2p_link entry.

    type 11 = xGMI, type 2 = PCIe (reference: internal/pkg/allocator/
    device.go:144-150).  weight/min_bandwidth are the kfd-reported link cost
    and bandwidth (MB/s); on an 8*MI355X hive each GPU has 7 xGMI
    point-to-point links.
    """

    node_from: int = -1
    node_to: int = -1
    type: int = -1
    weight: int = 0
    min_bandwidth: int = 0
    max_bandwidth: int = 0

    XGMI = 11
    PCIE = 2

    @classmethod
    def from_props(cls, props: Dict[str, int]) -> "KFDLink":
        return cls(
            node_from=props.get("node_from", -1),
            node_to=props.get("node_to", -1),
            type=props.get("type", -1),
            weight=props.get("weight", 0),
            min_bandwidth=props.get("min_bandwidth", 0),
            max_bandwidth=props.get("max_bandwidth", 0),
        )


@dataclass
class KFDNode:
    node_id: int
    properties: Dict[str, int] = field(default_factory=dict)
    mem_banks: List[Dict[str, int]] = field(default_factory=list)
    io_links: List[KFDLink] = field(default_factory=list)
    p2p_links: List[KFDLink] = field(default_factory=list)

    @property
    def is_gpu(self) -> bool:
        # A kfd node is a GPU iff it has no CPU cores and a gfx target
        # (reference: internal/pkg/plugin/plugin.go:198).
        return (
            self.properties.get("cpu_cores_count", 0) == 0
            and self.properties.get("gfx_target_version", 0) > 0
        )

    @property
    def render_minor(self) -> int:
        return self.properties.get("drm_render_minor", 0)

    @property
    def hive_id(self) -> int:
        """xGMI hive id; 0 = not part of a hive / unknown."""
        return self.properties.get("hive_id", 0)

    @property
    def numa_node(self) -> int:
        return self.properties.get("numa_node", -1)

    @property
    def simd_count(self) -> int:
        return self.properties.get("simd_count", 0)

    @property
    def cu_count(self) -> int:
        spc = self.properties.get("simd_per_cu", 0)
        return self.simd_count // spc if spc else 0

    @property
    def vram_bytes(self) -> int:
        # mem_banks/0 size_in_bytes; 288 GB HBM3E on MI355X
        # (reference vram label source: cmd/k8s-node-labeller/main.go:262-272).
        if not self.mem_banks:
            return 0
        return self.mem_banks[0].get("size_in_bytes", 0)

    def dev_id(self) -> Optional[str]:
        """PCI-ish device id string shared by all partitions of one GPU.

        Decoded from location_id/domain exactly as the reference does
        (reference: amdgpu.go:141-143): dev = (loc>>3)&0x1f, bus = (loc>>8)&0xff,
        formatted "%04x:%02x:%02x:0".
        """
        if "location_id" not in self.properties or "domain" not in self.properties:
            return None
        loc = self.properties["location_id"]
        domain = self.properties["domain"]
        dev = (loc >> 3) & 0x1F
        bus = (loc >> 8) & 0xFF
        return f"{domain:04x}:{bus:02x}:{dev:02x}:0"

    def all_links(self) -> List[KFDLink]:
        return self.io_links + self.p2p_links


class KFDTopology:
    """The parsed node tree plus the derived maps everything else consumes."""

    def __init__(self, nodes: Dict[int, KFDNode]):
        self.nodes = nodes

    @classmethod
    def load(cls, paths: SysPaths = SysPaths()) -> "KFDTopology":
        nodes: Dict[int, KFDNode] = {}
        root = paths.kfd_topology_nodes
        for prop_file in glob.glob(os.path.join(root, "*", "properties")):
            node_dir = os.path.dirname(prop_file)
            try:
                node_id = int(os.path.basename(node_dir))
            except ValueError:
                continue
            props = parse_properties_file(prop_file)
            if props is None:
                continue
            node = KFDNode(node_id=node_id, properties=props)
            for bank_file in sorted(
                glob.glob(os.path.join(node_dir, "mem_banks", "*", "properties"))
            ):
                bank = parse_properties_file(bank_file)
                if bank is not None:
                    node.mem_banks.append(bank)
            for kind in ("io_links", "p2p_links"):
                links = getattr(node, kind)
                for link_file in glob.glob(
                    os.path.join(node_dir, kind, "[0-9]*", "properties")
                ):
                    link_props = parse_properties_file(link_file)
                    if link_props is not None:
                        links.append(KFDLink.from_props(link_props))
            nodes[node_id] = node
        return cls(nodes)

    # ---- derived maps (reference: amdgpu.go:103-148, 507-549) ----

    def render_minor_to_dev_id(self) -> Dict[int, str]:
        """drm_render_minor -> PCI-ish devID, for GPU nodes only."""
        out: Dict[int, str] = {}
        for node in self.nodes.values():
            if node.render_minor <= 0:
                continue
            dev_id = node.dev_id()
            if dev_id is not None:
                out[node.render_minor] = dev_id
        return out

    def render_minor_to_node_id(self) -> Dict[int, int]:
        return {
            n.render_minor: n.node_id
            for n in self.nodes.values()
            if n.render_minor > 0
        }

    def gpu_nodes(self) -> List[KFDNode]:
        return [n for n in self.nodes.values() if n.is_gpu]

    def node_by_render_minor(self, render_minor: int) -> Optional[KFDNode]:
        for n in self.nodes.values():
            if n.render_minor == render_minor:
                return n
        return None
