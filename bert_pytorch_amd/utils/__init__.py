from . import checkpoint  # noqa: F401
from .logging import MetricLogger  # noqa: F401
