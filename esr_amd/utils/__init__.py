from .logging import setup_logging, MetricWriter  # noqa: F401
from .tracker import MetricTracker  # noqa: F401
from .timers import Timer, CudaTimer  # noqa: F401
