from .sysfs import SysPaths
from .kfd import (
    KFDNode,
    KFDTopology,
    parse_properties_file,
    parse_properties_text,
)
from .partition import (
    PartitionError,
    apply_partition_mode,
    available_partition_modes,
    current_partition_modes,
    set_partition_mode,
)
from .discovery import (
    DriverUnavailableError,
    GPUDevice,
    discover_gpus,
    is_homogeneous,
    unique_partition_config_count,
    is_compute_partition_supported,
    is_memory_partition_supported,
    count_gpus_from_topology,
    simple_health_check,
)

__all__ = [
    "SysPaths",
    "KFDNode",
    "KFDTopology",
    "parse_properties_file",
    "parse_properties_text",
    "DriverUnavailableError",
    "GPUDevice",
    "discover_gpus",
    "is_homogeneous",
    "unique_partition_config_count",
    "is_compute_partition_supported",
    "is_memory_partition_supported",
    "count_gpus_from_topology",
    "simple_health_check",
    "PartitionError",
    "apply_partition_mode",
    "available_partition_modes",
    "current_partition_modes",
    "set_partition_mode",
]
