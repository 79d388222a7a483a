from .exporter import get_gpu_health, populate_per_gpu_health
from .checker import HeartbeatTicker

__all__ = ["get_gpu_health", "populate_per_gpu_health", "HeartbeatTicker"]
