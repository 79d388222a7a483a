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


# Performance floors for the deep probe.  Calibrated against our own
# MI355X measurements across BOTH conditions the probe runs in:
# standalone (1935-2043 TF/s bf16 MFMA, 5.1-6.3 TB/s HBM copy across
# boxes — profiles/r01_bench_mi355x.md, r01_pmc_mfma_peak.csv) and
# IN-BAND on a loaded daemon, where host-side churn and package-power
# shifting cost ~25-30% (worst healthy in-band observation: 1383 TF/s,
# r02 soak — an earlier 1450 floor false-positived there).  Floors sit
# ~25% under the worst healthy in-band number: they catch a halved
# matrix pipe or a downtrained HBM channel, never
# DVFS noise.  Overridable per deployment via env; 0 disables.
DEFAULT_MFMA_FLOOR_TFLOPS = 1000.0
DEFAULT_HBM_FLOOR_GBPS = 3800.0
MFMA_FLOOR_ENV = "AMDXDP_MFMA_FLOOR_TFLOPS"
HBM_FLOOR_ENV = "AMDXDP_HBM_FLOOR_GBPS"


def _floor_from_env(env: str, default: float) -> float:
    try:
        return float(os.environ[env])
    except (KeyError, ValueError):
        return default


def deep_health_probe(
    device: int = 0,
    hbm_bytes: int = 1 << 30,
    mfma_floor_tflops: Optional[float] = None,
    hbm_floor_gbps: Optional[float] = None,
) -> dict:
    """Run the on-GPU MFMA/LDS/HBM probe and apply performance floors.

    Raises loudly when the extension is missing on a GPU machine.  The
    result's `healthy` goes False on any correctness failure OR any floor
    violation; violations are listed in `floor_violations`.
    """
    if mfma_floor_tflops is None:
        mfma_floor_tflops = _floor_from_env(MFMA_FLOOR_ENV,
                                            DEFAULT_MFMA_FLOOR_TFLOPS)
    if hbm_floor_gbps is None:
        hbm_floor_gbps = _floor_from_env(HBM_FLOOR_ENV,
                                         DEFAULT_HBM_FLOOR_GBPS)
    mod = load_healthprobe(required=True)
    res = mod.run_probe(device=device, hbm_bytes=hbm_bytes)
    violations = []
    mfma = res.get("mfma_tflops")
    if mfma_floor_tflops > 0 and mfma is not None and mfma < mfma_floor_tflops:
        violations.append(
            f"mfma_tflops {mfma:.0f} < floor {mfma_floor_tflops:.0f}"
        )
    hbm = res.get("hbm_gbps")
    if hbm_floor_gbps > 0 and hbm is not None and hbm < hbm_floor_gbps:
        violations.append(f"hbm_gbps {hbm:.0f} < floor {hbm_floor_gbps:.0f}")
    res["floors"] = {
        "mfma_tflops": mfma_floor_tflops,
        "hbm_gbps": hbm_floor_gbps,
    }
    res["floor_violations"] = violations
    if violations:
        res["healthy"] = False
    return res
