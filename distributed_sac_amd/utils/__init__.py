from .tfevents import TFEventWriter  # noqa: F401
from .logger import MetricLogger  # noqa: F401
from .timers import StepTimer  # noqa: F401
