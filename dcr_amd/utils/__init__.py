from .logging import SmoothedValue, MetricLogger, Tracker
from .image import tensor_to_pil, concat_h, image_grid

__all__ = ["SmoothedValue", "MetricLogger", "Tracker", "tensor_to_pil", "concat_h", "image_grid"]
from .profiler import PhaseProfiler
from .misc import bool_flag
