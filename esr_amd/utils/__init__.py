from .logging import setup_logging, MetricWriter  # noqa: F401
from .tracker import MetricTracker  # noqa: F401
from .timers import Timer, CudaTimer  # noqa: F401
from .misc import normalize_nonzero, inf_loop  # noqa: F401,E402
