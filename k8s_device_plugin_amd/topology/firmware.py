"""debugfs firmware-info parser.

Fallback source for firmware/feature versions when the raw-ioctl shim is
unavailable (reference has the same dual path: ioctl via libdrm plus
parseDebugFSFirmwareInfo over /sys/kernel/debug/dri/<minor>/
amdgpu_firmware_info — internal/pkg/amdgpu/amdgpu.go:476-501)."""

from __future__ import annotations

import re
from typing import Dict, Tuple

_FW_LINE_RE = re.compile(
    r"(\w+) feature version: (\d+), firmware version: (0x[0-9a-fA-F]+)"
)


def parse_debugfs_firmware_text(text: str) -> Tuple[Dict[str, int], Dict[str, int]]:
    """Returns ({block: feature_version}, {block: firmware_version})."""
    feat: Dict[str, int] = {}
    fw: Dict[str, int] = {}
    for line in text.splitlines():
        m = _FW_LINE_RE.search(line)
        if m is None:
            continue
        feat[m.group(1)] = int(m.group(2))
        fw[m.group(1)] = int(m.group(3), 16)
    return feat, fw


def parse_debugfs_firmware(path: str) -> Tuple[Dict[str, int], Dict[str, int]]:
    try:
        with open(path) as f:
            return parse_debugfs_firmware_text(f.read())
    except OSError:
        return {}, {}


def debugfs_firmware_path(card: int, debugfs_root: str = "/sys/kernel/debug") -> str:
    return f"{debugfs_root}/dri/{card}/amdgpu_firmware_info"
