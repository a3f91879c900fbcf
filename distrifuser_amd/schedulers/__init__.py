from .ddim import DDIMScheduler
from .euler import EulerDiscreteScheduler
from .dpm_solver import DPMSolverMultistepScheduler


def get_scheduler(name: str):
    name = name.lower().replace("_", "-")
    if name == "ddim":
        return DDIMScheduler()
    if name == "euler":
        return EulerDiscreteScheduler()
    if name in ("dpm-solver", "dpmsolver", "dpm"):
        return DPMSolverMultistepScheduler()
    raise ValueError(f"unknown scheduler {name!r} (ddim | euler | dpm-solver)")


__all__ = [
    "DDIMScheduler",
    "EulerDiscreteScheduler",
    "DPMSolverMultistepScheduler",
    "get_scheduler",
]
