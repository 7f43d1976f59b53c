"""RFC3339 time helpers (k8s metav1.Time wire format)."""
from __future__ import annotations

import calendar
import time
from typing import Optional, Union


def format_time(t: Optional[float] = None) -> str:
    """Unix seconds -> RFC3339 UTC string, k8s style (second precision)."""
    if t is None:
        t = time.time()
    return time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime(t))


def parse_time(s: Union[str, float, None]) -> Optional[float]:
    """RFC3339 string (or unix seconds passthrough) -> unix seconds."""
    if s is None or s == "":
        return None
    if isinstance(s, (int, float)):
        return float(s)
    s = s.strip()
    # Tolerate fractional seconds and offsets: 2020-01-01T00:00:00(.123)?(Z|+hh:mm)
    if s.endswith("Z"):
        base = s[:-1]
        offset = 0
    elif len(s) >= 6 and s[-6] in "+-" and s[-3] == ":":
        sign = -1 if s[-6] == "-" else 1
        offset = sign * (int(s[-5:-3]) * 3600 + int(s[-2:]) * 60)
        base = s[:-6]
    else:
        base = s
        offset = 0
    frac = 0.0
    if "." in base:
        base, frac_s = base.split(".", 1)
        frac = float("0." + frac_s) if frac_s else 0.0
    st = time.strptime(base, "%Y-%m-%dT%H:%M:%S")
    return calendar.timegm(st) + frac - offset
