from .config import DistriConfig, check_env, is_power_of_2
from .comm import PatchParallelismCommManager

__all__ = ["DistriConfig", "PatchParallelismCommManager", "check_env", "is_power_of_2"]
