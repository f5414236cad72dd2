from .metrics import get_metric_with_unit, METRIC_UNITS
from .errors import error_percentiles, format_error_table

__all__ = ["get_metric_with_unit", "METRIC_UNITS", "error_percentiles", "format_error_table"]
