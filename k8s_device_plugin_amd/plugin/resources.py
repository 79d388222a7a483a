"""Resource naming strategy: single vs mixed.

Parity with the reference CLI logic (reference:
cmd/k8s-device-plugin/main.go:35-91):
  - single + homogeneous            -> ["gpu"]
  - single + heterogeneous          -> error
  - mixed + partitioned             -> one resource per '<compute>_<memory>'
  - mixed + unpartitioned           -> ["gpu"]
"""

from __future__ import annotations

from typing import Dict, List

from ..topology.discovery import (
    GPUDevice,
    is_homogeneous,
    unique_partition_config_count,
)

STRATEGY_SINGLE = "single"
STRATEGY_MIXED = "mixed"


class StrategyError(ValueError):
    pass


def parse_strategy(s: str) -> str:
    if s in (STRATEGY_SINGLE, STRATEGY_MIXED):
        return s
    raise StrategyError(f"invalid resource naming strategy: {s}")


def get_resource_list(devices: Dict[str, GPUDevice], strategy: str) -> List[str]:
    if not devices:
        return []
    partition_counts = unique_partition_config_count(devices)
    if is_homogeneous(devices):
        if strategy == STRATEGY_SINGLE:
            return ["gpu"]
        if not partition_counts:
            # partitioning unsupported: always report "gpu"
            return ["gpu"]
        return sorted(k for k, v in partition_counts.items() if v > 0)
    if strategy == STRATEGY_SINGLE:
        raise StrategyError(
            "partitions of different styles across GPUs are not supported "
            "with the single strategy; use mixed"
        )
    return sorted(k for k, v in partition_counts.items() if v > 0)
