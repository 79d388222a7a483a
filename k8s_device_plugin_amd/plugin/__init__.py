from .resources import parse_strategy, get_resource_list, StrategyError
from .server import AMDGPUPlugin
from .manager import PluginManager

__all__ = [
    "parse_strategy",
    "get_resource_list",
    "StrategyError",
    "AMDGPUPlugin",
    "PluginManager",
]
