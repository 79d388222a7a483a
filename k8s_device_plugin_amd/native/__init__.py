"""Native extension loaders.

On a GPU machine the HIP probe must NOT silently fall back: if CUDA/HIP
devices are visible but the extension is missing, loading raises so a broken
deployment is caught loudly (the plugin's compute path is the .so, not an
eager fallback).
"""

from __future__ import annotations

import importlib
import os
from typing import Optional

_DIR = os.path.dirname(os.path.abspath(__file__))


class NativeExtensionMissing(ImportError):
    pass


def _load(name: str):
    try:
        return importlib.import_module(f"{__name__}.{name}")
    except ImportError as e:
        raise NativeExtensionMissing(
            f"{name}.so not built under {_DIR}; run "
            f"python -m k8s_device_plugin_amd.native.build (error: {e})"
        ) from e


def load_drmctl(required: bool = False):
    """Raw-ioctl shim; optional — sysfs fallbacks cover a missing shim."""
    try:
        return _load("_drmctl")
    except NativeExtensionMissing:
        if required:
            raise
        return None


def _gpu_visible() -> bool:
    # cheap check without importing torch: any KFD GPU node present
    try:
        from ..topology import simple_health_check

        return os.path.exists("/dev/kfd") and simple_health_check()
    except Exception:
        return False


def load_healthprobe(required: Optional[bool] = None):
    """gfx950 deep probe; REQUIRED whenever a GPU is actually present."""
    if required is None:
        required = _gpu_visible()
    try:
        return _load("_healthprobe")
    except NativeExtensionMissing:
        if required:
            raise
        return None


def load_fastserver(required: bool = False):
    """Native DevicePlugin gRPC server (nghttp2 over UDS)."""
    try:
        return _load("_fastserver")
    except NativeExtensionMissing:
        if required:
            raise
        return None


def deep_health_probe(device: int = 0, hbm_bytes: int = 1 << 30) -> dict:
    """Run the on-GPU MFMA/LDS/HBM probe.  Raises loudly when the extension
    is missing on a GPU machine."""
    mod = load_healthprobe(required=True)
    return mod.run_probe(device=device, hbm_bytes=hbm_bytes)
