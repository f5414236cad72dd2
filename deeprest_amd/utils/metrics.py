"""Resource-type display names and units.

Same resource vocabulary as the reference (resource-estimation/utils.py:8-26):
cpu (millicores), memory working set (MB), write IOps, write throughput (KB),
disk usage (MB).
"""

from __future__ import annotations

from typing import Tuple

METRIC_UNITS = {
    "cpu": ("CPU (millicores)", "(millicores)"),
    "memory": ("Working Set Size (MB)", "(MB)"),
    "write-iops": ("Write IOps", ""),
    "write-tp": ("Write Throughput (KB)", "(KB)"),
    "usage": ("Disk Usage (MB)", "(MB)"),
}


def get_metric_with_unit(metric: str) -> Tuple[str, str]:
    return METRIC_UNITS.get(metric, (metric, ""))
