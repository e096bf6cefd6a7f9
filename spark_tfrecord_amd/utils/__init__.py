from .metrics import IOMetrics, StageTimer, get_logger, last_metrics

__all__ = ["IOMetrics", "StageTimer", "get_logger", "last_metrics"]
