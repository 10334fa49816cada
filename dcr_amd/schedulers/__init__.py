from .ddpm import DDPMScheduler
from .ddim import DDIMScheduler
from .dpm_solver import DPMSolverMultistepScheduler

__all__ = ["DDPMScheduler", "DDIMScheduler", "DPMSolverMultistepScheduler"]
