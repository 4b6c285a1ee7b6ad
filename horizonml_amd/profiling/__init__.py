from .timers import StepProfiler  # noqa: F401
from .metrics import MetricsWriter, EpochMetrics, REFERENCE_COLUMNS  # noqa: F401
