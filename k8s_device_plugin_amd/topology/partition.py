"""Partition-mode control: write SPX/DPX/QPX/CPX and NPS1..NPS8 modes.

The reference is read-only here — it probes `available_compute_partition` /
`available_memory_partition` for capability labels and reads the `current_*`
files for bucketing (reference: internal/pkg/amdgpu/amdgpu.go:175-206,
306-339) but never changes the mode; operators flip it out-of-band with
amd-smi and restart the plugin.  This module closes that loop the MI355X
way: writing the mode name into the same sysfs files drives the amdgpu
driver's repartition, after which kfd re-enumerates nodes and the plugin's
heartbeat path (device-set change -> allocator re-init -> CDI/list rebuild)
picks up the new logical-device fan-out without a restart.

SAFETY — this is a destructive, node-wide operation (it tears down every
kfd consumer's view of the GPU).  Writes are double-gated:
  * the caller must pass ``allow=True`` (no accidental API use), and
  * the environment must set ``AMDXDP_ALLOW_REPARTITION=1`` (no accidental
    deployment use — DaemonSets must opt in explicitly).
Shared development boxes (e.g. multi-tenant GPU pools) must NEVER set the
env var; tests exercise this module exclusively against fake sysfs trees.
"""

from __future__ import annotations

import glob
import logging
import os
import time
from typing import Dict, List, Optional, Tuple

from .sysfs import SysPaths, read_stripped

log = logging.getLogger(__name__)

ALLOW_ENV = "AMDXDP_ALLOW_REPARTITION"

# amdgpu driver vocabulary (drivers/gpu/drm/amd/amdgpu/amdgpu_xcp.c)
VALID_COMPUTE = ("SPX", "DPX", "TPX", "QPX", "CPX")
VALID_MEMORY = ("NPS1", "NPS2", "NPS4", "NPS8")


class PartitionError(RuntimeError):
    pass


def _gpu_pci_dirs(paths: SysPaths) -> List[str]:
    return sorted(
        d
        for d in glob.glob(os.path.join(paths.amdgpu_pci, "*"))
        if os.path.isdir(d) and ":" in os.path.basename(d)
    )


def _parse_available(text: Optional[str]) -> List[str]:
    """`available_*_partition` is a comma-separated list, e.g.
    "SPX, DPX, QPX, CPX"."""
    if not text:
        return []
    return [t.strip().upper() for t in text.split(",") if t.strip()]


def available_partition_modes(
    paths: SysPaths = SysPaths(), pci_addr: Optional[str] = None
) -> Tuple[List[str], List[str]]:
    """(compute modes, memory modes) the node's GPUs can switch into."""
    dirs = _gpu_pci_dirs(paths)
    if pci_addr is not None:
        dirs = [d for d in dirs if os.path.basename(d) == pci_addr]
    if not dirs:
        return [], []
    d = dirs[0]  # homogeneous node assumption, same as the reference
    comp = _parse_available(
        read_stripped(os.path.join(d, "available_compute_partition"))
    )
    mem = _parse_available(
        read_stripped(os.path.join(d, "available_memory_partition"))
    )
    return comp, mem


def current_partition_modes(
    paths: SysPaths = SysPaths(),
) -> Dict[str, Tuple[str, str]]:
    """{pci_addr: (compute, memory)} for every physical GPU."""
    out: Dict[str, Tuple[str, str]] = {}
    for d in _gpu_pci_dirs(paths):
        comp = read_stripped(os.path.join(d, "current_compute_partition")) or ""
        mem = read_stripped(os.path.join(d, "current_memory_partition")) or ""
        out[os.path.basename(d)] = (comp.upper(), mem.upper())
    return out


def _check_gate(allow: bool) -> None:
    if not allow:
        raise PartitionError(
            "partition-mode write refused: caller did not pass allow=True"
        )
    if os.environ.get(ALLOW_ENV) != "1":
        raise PartitionError(
            f"partition-mode write refused: {ALLOW_ENV}=1 not set "
            "(never set it on shared GPU boxes)"
        )


def set_partition_mode(
    paths: SysPaths = SysPaths(),
    compute: Optional[str] = None,
    memory: Optional[str] = None,
    pci_addrs: Optional[List[str]] = None,
    allow: bool = False,
    settle_timeout_s: float = 30.0,
) -> Dict[str, Tuple[str, str]]:
    """Write the requested mode(s) to each GPU and wait until the
    `current_*` files report them.  Returns the final per-GPU modes.

    Memory-mode changes require the compute mode to be written too on real
    hardware (the driver re-creates the xcp devices); callers normally set
    both.  Raises PartitionError on gate failure, unknown/unsupported
    modes, write errors, or settle timeout.
    """
    _check_gate(allow)
    if compute is None and memory is None:
        raise PartitionError("nothing to do: no compute or memory mode given")
    if compute is not None:
        compute = compute.upper()
        if compute not in VALID_COMPUTE:
            raise PartitionError(f"unknown compute mode {compute!r}")
    if memory is not None:
        memory = memory.upper()
        if memory not in VALID_MEMORY:
            raise PartitionError(f"unknown memory mode {memory!r}")

    dirs = _gpu_pci_dirs(paths)
    if pci_addrs is not None:
        want = set(pci_addrs)
        dirs = [d for d in dirs if os.path.basename(d) in want]
        missing = want - {os.path.basename(d) for d in dirs}
        if missing:
            raise PartitionError(f"unknown GPUs: {sorted(missing)}")
    if not dirs:
        raise PartitionError("no amdgpu PCI devices found")

    avail_c, avail_m = available_partition_modes(paths)
    if compute is not None and compute not in avail_c:
        raise PartitionError(
            f"compute mode {compute} not offered by hardware "
            f"(available: {avail_c})"
        )
    if memory is not None and memory not in avail_m:
        raise PartitionError(
            f"memory mode {memory} not offered by hardware "
            f"(available: {avail_m})"
        )

    for d in dirs:
        # memory first: the driver rejects NPS changes while partitioned
        # into the old compute mode's xcp devices in some orders; writing
        # memory then compute matches amd-smi's sequencing
        for fname, mode in (
            ("current_memory_partition", memory),
            ("current_compute_partition", compute),
        ):
            if mode is None:
                continue
            path = os.path.join(d, fname)
            try:
                with open(path, "w") as f:
                    f.write(mode + "\n")
            except OSError as e:
                raise PartitionError(f"writing {mode} to {path} failed: {e}")
        log.info(
            "repartition requested on %s: compute=%s memory=%s",
            os.path.basename(d), compute, memory,
        )

    deadline = time.monotonic() + settle_timeout_s
    want_dirs = [os.path.basename(d) for d in dirs]
    while True:
        cur = current_partition_modes(paths)
        ok = all(
            (compute is None or cur.get(a, ("", ""))[0] == compute)
            and (memory is None or cur.get(a, ("", ""))[1] == memory)
            for a in want_dirs
        )
        if ok:
            return cur
        if time.monotonic() >= deadline:
            raise PartitionError(
                f"partition mode did not settle within {settle_timeout_s}s: "
                f"{ {a: cur.get(a) for a in want_dirs} }"
            )
        time.sleep(0.2)


def apply_partition_mode(
    paths: SysPaths = SysPaths(),
    compute: Optional[str] = None,
    memory: Optional[str] = None,
    allow: bool = False,
    settle_timeout_s: float = 30.0,
):
    """set_partition_mode + full rediscovery.  Returns (modes, devices):
    the settled per-GPU modes and the freshly discovered device map (the
    new logical fan-out after the kernel re-enumerates kfd nodes)."""
    modes = set_partition_mode(
        paths,
        compute=compute,
        memory=memory,
        allow=allow,
        settle_timeout_s=settle_timeout_s,
    )
    from .discovery import discover_gpus

    devices = discover_gpus(paths, strict=False)
    return modes, devices
