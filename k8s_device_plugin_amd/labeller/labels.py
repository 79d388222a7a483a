"""Node label generators: the 12 label kinds in both namespaces.

Parity with the reference labeller (reference:
cmd/k8s-node-labeller/main.go:37-397):
  - plain namespace  amd.com/gpu.<kind>
  - experimental     beta.amd.com/gpu.<kind> plus counter labels
    beta.amd.com/gpu.<kind>.<value>=<count>
and the same value computations (vram rounded to G from mem_banks/0,
cu-count = simd_count/simd_per_cu, partition label only on homogeneous
nodes, ...).  Firmware/family/product-name come from the raw-ioctl shim
with sysfs fallbacks instead of libdrm cgo.
"""

from __future__ import annotations

import logging
import os
from typing import Callable, Dict, Optional

from ..native import load_drmctl
from ..topology import (
    GPUDevice,
    KFDTopology,
    SysPaths,
    discover_gpus,
    is_compute_partition_supported,
    is_homogeneous,
    is_memory_partition_supported,
    unique_partition_config_count,
)
from ..topology.sysfs import read_stripped

log = logging.getLogger(__name__)

AMD_PREFIX = "amd.com"
EXPERIMENTAL_AMD_PREFIX = "beta.amd.com"

# AMDGPU_FAMILY_* -> string (kernel UAPI include/uapi/drm/amdgpu_drm.h;
# reference mapping: internal/pkg/amdgpu/amdgpu.go:44-84)
FAMILY_NAMES = {
    110: "SI",
    120: "CI",
    125: "KV",
    130: "VI",
    135: "CZ",
    141: "AI",
    142: "RV",
    143: "NV",
    144: "VGH",
    145: "GC_11_0_0",
    146: "YC",
    148: "GC_11_0_1",
    149: "GC_10_3_6",
    150: "GC_11_5_0",
    151: "GC_10_3_7",
    152: "GC_12_0_0",
}

_IDS_FILE = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                         "native", "amdgpu.ids")


def family_from_gfx_target(gfx: int) -> "str | None":
    """sysfs fallback mapping: kfd gfx_target_version -> family name.

    Used only when the DRM ioctl is unavailable; mirrors FAMILY_NAMES
    coverage for every discrete-GPU generation rather than hardcoding one
    branch, so a future-IP Instinct falls through to None (no label)
    instead of silently mislabelling.  APU-specific families (KV/CZ/RV/
    VGH/YC/GC_10_3_x) are indistinguishable from their discrete siblings
    in kfd sysfs — those resolve correctly only via the ioctl path.
    Encoding: major*10000 + minor*100 + rev (e.g. gfx950 -> 90500).
    """
    major, minor = gfx // 10000, (gfx // 100) % 100
    if major == 6:
        return "SI"
    if major == 7:
        return "CI"
    if major == 8:
        return "VI"
    if major == 9:
        return "AI"  # Vega + all CDNA incl. gfx942 (90402) and gfx950 (90500)
    if major == 10:
        return "NV"
    if major == 11:
        return "GC_11_5_0" if minor == 5 else "GC_11_0_0"
    if major == 12:
        return "GC_12_0_0"
    return None


def create_label_prefix(name: str, experimental: bool = False) -> str:
    prefix = EXPERIMENTAL_AMD_PREFIX if experimental else AMD_PREFIX
    return f"{prefix}/gpu.{name}"


def _create_labels(kind: str, counts: Dict[str, int]) -> Dict[str, str]:
    """Both-namespace label expansion (reference: main.go:87-108)."""
    labels: Dict[str, str] = {}
    beta = create_label_prefix(kind, True)
    for value, count in counts.items():
        labels[f"{beta}.{value}"] = str(count)
        if len(counts) == 1:
            labels[beta] = value
    plain = create_label_prefix(kind, False)
    for value, count in counts.items():
        if len(counts) == 1:
            labels[plain] = value
        else:
            labels[f"{plain}.{value}"] = str(count)
    return labels


def product_name_from_ids(device_id: int, revision: int = 0) -> Optional[str]:
    """Look up the marketing name in the bundled amdgpu.ids table."""
    try:
        with open(_IDS_FILE) as f:
            fallback = None
            for line in f:
                line = line.strip()
                if not line or line.startswith("#") or "," not in line:
                    continue
                parts = [p.strip() for p in line.split(",", 2)]
                if len(parts) != 3:
                    continue
                try:
                    did = int(parts[0], 16)
                    rid = int(parts[1], 16)
                except ValueError:
                    continue
                if did == device_id:
                    if rid == revision:
                        return parts[2]
                    fallback = fallback or parts[2]
            return fallback
    except OSError:
        return None


def _physical_gpus(devices: Dict[str, GPUDevice]):
    return [d for d in devices.values() if not d.is_partition]


def _sysfs_card_read(paths: SysPaths, card: int, *rel: str) -> Optional[str]:
    return read_stripped(os.path.join(paths.drm_card_device(card), *rel))


def _sanitize_name(name: str) -> str:
    # label values forbid spaces/parens (reference: main.go:211)
    return name.replace(" ", "_").replace("(", "").replace(")", "")


# ---- generators: each (devices, paths, topo) -> {label: value} ----

def _gen_firmware(devices, paths, topo) -> Dict[str, str]:
    from ..topology.firmware import debugfs_firmware_path, parse_debugfs_firmware

    drm = load_drmctl()
    counts: Dict[str, int] = {}
    for d in _physical_gpus(devices):
        if not d.kfd_backed:
            continue
        feat = fwv = None
        if drm is not None:
            try:
                fw = drm.query_firmware(f"/dev/dri/renderD{d.render_d}")
                feat, fwv = fw["feature"], fw["firmware"]
            except RuntimeError as e:
                log.error("firmware ioctl failed for %s: %s", d.id, e)
        if feat is None:
            # debugfs fallback (root-only; present in privileged DaemonSets)
            feat, fwv = parse_debugfs_firmware(
                debugfs_firmware_path(d.card, os.path.join(paths.root, "sys/kernel/debug"))
            )
        if not feat:
            continue
        for blk, ver in feat.items():
            counts[f"{blk}.feat.{ver}"] = counts.get(f"{blk}.feat.{ver}", 0) + 1
        for blk, ver in fwv.items():
            counts[f"{blk}.fw.{ver}"] = counts.get(f"{blk}.fw.{ver}", 0) + 1
    # firmware labels exist only in the experimental namespace
    # (reference: main.go:137-142)
    pfx = create_label_prefix("firmware", True)
    return {f"{pfx}.{k}": str(v) for k, v in counts.items()}


def _gen_family(devices, paths, topo) -> Dict[str, str]:
    drm = load_drmctl()
    counts: Dict[str, int] = {}
    for d in _physical_gpus(devices):
        name = None
        if drm is not None and d.kfd_backed:
            try:
                info = drm.query_device_info(f"/dev/dri/renderD{d.render_d}")
                name = FAMILY_NAMES.get(info["family"])
            except RuntimeError:
                pass
        if name is None and d.node_id in topo.nodes:
            # sysfs fallback keyed on gfx_target_version, covering every
            # discrete generation in FAMILY_NAMES (gfx950 -> AI)
            gfx = topo.nodes[d.node_id].properties.get("gfx_target_version", 0)
            name = family_from_gfx_target(gfx)
        if name:
            counts[name] = counts.get(name, 0) + 1
    return _create_labels("family", counts) if counts else {}


def _gen_driver_version(devices, paths, topo) -> Dict[str, str]:
    for d in _physical_gpus(devices):
        v = _sysfs_card_read(paths, d.card, "driver", "module", "version")
        if v:
            return {create_label_prefix("driver-version"): v}
    return {}


def _gen_driver_src_version(devices, paths, topo) -> Dict[str, str]:
    for d in _physical_gpus(devices):
        v = _sysfs_card_read(paths, d.card, "driver", "module", "srcversion")
        if v:
            return {create_label_prefix("driver-src-version"): v}
    return {}


def _gen_device_id(devices, paths, topo) -> Dict[str, str]:
    counts: Dict[str, int] = {}
    for d in _physical_gpus(devices):
        v = _sysfs_card_read(paths, d.card, "device")
        if not v:
            continue
        v = v[2:] if v.startswith("0x") else v
        counts[v] = counts.get(v, 0) + 1
    return _create_labels("device-id", counts) if counts else {}


def _gen_product_name(devices, paths, topo) -> Dict[str, str]:
    drm = load_drmctl()
    counts: Dict[str, int] = {}
    for d in _physical_gpus(devices):
        name = _sysfs_card_read(paths, d.card, "product_name") or ""
        if not name and drm is not None and d.kfd_backed:
            # ioctl + bundled amdgpu.ids fallback (replaces
            # amdgpu_get_marketing_name, reference: amdgpu.go:551-563)
            try:
                info = drm.query_device_info(f"/dev/dri/renderD{d.render_d}")
                name = product_name_from_ids(info["device_id"], info["pci_rev"]) or ""
            except RuntimeError:
                pass
        name = _sanitize_name(name.strip())
        if name:
            counts[name] = counts.get(name, 0) + 1
    return _create_labels("product-name", counts) if counts else {}


def _gen_vram(devices, paths, topo) -> Dict[str, str]:
    counts: Dict[str, int] = {}
    for d in _physical_gpus(devices):
        node = topo.node_by_render_minor(d.render_d)
        if node is None or node.vram_bytes == 0:
            continue
        # bytes -> MB -> GiB rounded (reference: main.go:262-272);
        # 309220868096 B -> 288G on MI355X
        mb = node.vram_bytes // (1024 * 1024)
        g = round(mb / 1024)
        counts[f"{g}G"] = counts.get(f"{g}G", 0) + 1
    return _create_labels("vram", counts) if counts else {}


def _gen_simd_count(devices, paths, topo) -> Dict[str, str]:
    counts: Dict[str, int] = {}
    for d in _physical_gpus(devices):
        node = topo.node_by_render_minor(d.render_d)
        if node is None or node.simd_count <= 0:
            continue
        counts[str(node.simd_count)] = counts.get(str(node.simd_count), 0) + 1
    return _create_labels("simd-count", counts) if counts else {}


def _gen_cu_count(devices, paths, topo) -> Dict[str, str]:
    counts: Dict[str, int] = {}
    for d in _physical_gpus(devices):
        node = topo.node_by_render_minor(d.render_d)
        if node is None or node.cu_count <= 0:
            continue
        counts[str(node.cu_count)] = counts.get(str(node.cu_count), 0) + 1
    return _create_labels("cu-count", counts) if counts else {}


def _gen_compute_memory_partition(devices, paths, topo) -> Dict[str, str]:
    # homogeneous nodes only (reference: main.go:356-368)
    if not is_homogeneous(devices):
        return {}
    for key, count in unique_partition_config_count(devices).items():
        if count > 0:
            return {create_label_prefix("compute-memory-partition"): key}
    return {}


def _gen_compute_partitioning_supported(devices, paths, topo) -> Dict[str, str]:
    v = str(is_compute_partition_supported(paths)).lower()
    return {create_label_prefix("compute-partitioning-supported"): v}


def _gen_memory_partitioning_supported(devices, paths, topo) -> Dict[str, str]:
    v = str(is_memory_partition_supported(paths)).lower()
    return {create_label_prefix("memory-partitioning-supported"): v}


def _gen_xgmi_hive(devices, paths, topo) -> Dict[str, str]:
    """xGMI hive tagging (beyond the reference; BASELINE config 3).

    Values are the kfd `hive_id` in hex, counted per physical GPU — on a
    healthy 8xMI355X node one hive of 8, so schedulers/operators can
    select full-hive nodes (`amd.com/gpu.xgmi-hive=<id>`) and spot
    split-hive or hive-less nodes from the counter labels.  Additionally
    emits `amd.com/gpu.xgmi-hive-count` with the number of distinct
    hives (0 = no xGMI info).
    """
    counts: Dict[str, int] = {}
    for d in _physical_gpus(devices):
        node = topo.node_by_render_minor(d.render_d)
        if node is None or node.hive_id == 0:
            continue
        key = f"{node.hive_id:x}"
        counts[key] = counts.get(key, 0) + 1
    labels = _create_labels("xgmi-hive", counts) if counts else {}
    labels[create_label_prefix("xgmi-hive-count")] = str(len(counts))
    return labels


LABEL_GENERATORS: Dict[str, Callable] = {
    "firmware": _gen_firmware,
    "family": _gen_family,
    "driver-version": _gen_driver_version,
    "driver-src-version": _gen_driver_src_version,
    "device-id": _gen_device_id,
    "product-name": _gen_product_name,
    "vram": _gen_vram,
    "simd-count": _gen_simd_count,
    "cu-count": _gen_cu_count,
    "compute-memory-partition": _gen_compute_memory_partition,
    "compute-partitioning-supported": _gen_compute_partitioning_supported,
    "xgmi-hive": _gen_xgmi_hive,
    "memory-partitioning-supported": _gen_memory_partitioning_supported,
}

LABEL_KINDS = sorted(LABEL_GENERATORS)


def generate_labels(
    enabled: Optional[Dict[str, bool]] = None,
    paths: SysPaths = SysPaths(),
) -> Dict[str, str]:
    """Compute all enabled labels from the current hardware state."""
    devices = discover_gpus(paths, strict=False)
    topo = KFDTopology.load(paths)
    out: Dict[str, str] = {}
    for kind, gen in LABEL_GENERATORS.items():
        if enabled is not None and not enabled.get(kind, False):
            continue
        try:
            out.update(gen(devices, paths, topo))
        except Exception:
            log.exception("label generator %s failed", kind)
    return out


def remove_old_node_labels(labels: Dict[str, str]) -> None:
    """Delete every label this labeller manages, in place (reference:
    main.go:55-74): plain keys directly; experimental keys plus their
    counter label beta.amd.com/gpu.<kind>.<value>."""
    for kind in LABEL_KINDS:
        labels.pop(create_label_prefix(kind, False), None)
        beta = create_label_prefix(kind, True)
        value = labels.pop(beta, None)
        if value is not None:
            labels.pop(f"{beta}.{value}", None)
    # counter/firmware labels have arbitrary value suffixes; sweep every
    # remaining <ns>/gpu.<kind>.<suffix> key.  (The reference only removes
    # the beta counter recorded in the base label's value, so multi-valued
    # counter labels can leak across restarts — main.go:68-73; the sweep
    # fixes that.)
    beta_prefixes = tuple(create_label_prefix(k, True) + "." for k in LABEL_KINDS)
    for key in [k for k in labels if k.startswith(beta_prefixes)]:
        labels.pop(key, None)
    plain_prefixes = tuple(create_label_prefix(k, False) + "." for k in LABEL_KINDS)
    for key in [k for k in labels if k.startswith(plain_prefixes)]:
        labels.pop(key, None)
