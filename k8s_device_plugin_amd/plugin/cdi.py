"""Container Device Interface (CDI) support.

Modern container runtimes (containerd >=1.7 / CRI-O with CDI enabled)
prefer CDI injection over raw device specs: the plugin writes a CDI spec
describing every GPU and Allocate returns fully-qualified CDI device names
(`amd.com/gpu=<id>`); the runtime performs the injection.  The reference
plugin predates CDI — this build supports both paths simultaneously
(device specs always, CDI names when enabled), which is what kubelet's
dual-mode handling expects.
"""

from __future__ import annotations

import json
import os
import tempfile
from typing import Iterable

from ..topology.discovery import GPUDevice

# CDI 0.7: annotations carry per-device topology
# hints (NUMA node, partition config, physical-GPU id) so operators and
# runtime hooks can read placement facts straight from the spec.
CDI_VERSION = "0.7.0"
CDI_KIND = "amd.com/gpu"
CDI_SPEC_DIR = "/var/run/cdi"
ANNOTATION_PREFIX = "cdi.amd.com"


def cdi_device_name(device_id: str, kind: str = CDI_KIND) -> str:
    return f"{kind}={device_id}"


def build_cdi_spec(devices: Iterable[GPUDevice], kind: str = CDI_KIND) -> dict:
    spec = {
        "cdiVersion": CDI_VERSION,
        "kind": kind,
        "annotations": {
            f"{ANNOTATION_PREFIX}/producer": "k8s-device-plugin-mi355x",
        },
        # /dev/kfd is shared by every GPU workload on the node
        "containerEdits": {
            "deviceNodes": [{"path": "/dev/kfd", "permissions": "rw"}]
        },
        "devices": [],
    }
    for d in sorted(devices, key=lambda x: x.id):
        annotations = {
            f"{ANNOTATION_PREFIX}/numa-node": str(d.numa_node),
            f"{ANNOTATION_PREFIX}/physical-gpu": d.dev_id or d.id,
        }
        if d.compute_partition and d.memory_partition:
            annotations[f"{ANNOTATION_PREFIX}/partition"] = d.partition_key
        spec["devices"].append(
            {
                "name": d.id,
                "annotations": annotations,
                "containerEdits": {
                    "deviceNodes": [
                        {"path": f"/dev/dri/card{d.card}", "permissions": "rw"},
                        {"path": f"/dev/dri/renderD{d.render_d}",
                         "permissions": "rw"},
                    ]
                },
            }
        )
    return spec


def write_cdi_spec(
    devices: Iterable[GPUDevice],
    spec_dir: str = CDI_SPEC_DIR,
    kind: str = CDI_KIND,
) -> str:
    """Atomically write the CDI spec file; returns its path."""
    os.makedirs(spec_dir, exist_ok=True)
    path = os.path.join(spec_dir, kind.replace("/", "-") + ".json")
    spec = build_cdi_spec(devices, kind)
    fd, tmp = tempfile.mkstemp(dir=spec_dir, suffix=".tmp")
    try:
        with os.fdopen(fd, "w") as f:
            json.dump(spec, f, indent=2)
        os.replace(tmp, path)
    except BaseException:
        if os.path.exists(tmp):
            os.unlink(tmp)
        raise
    return path
