from .logging import setup_logging  # noqa: F401
from .timing import StopWatch, PhaseTimer  # noqa: F401
