from .core import Scheduler, FilterResult, BindResult  # noqa: F401
from .score import NodeUsage, calc_score, fit_in_certain_device, fit_in_devices  # noqa: F401
