"""Reporters — the reference's stdout log grammar plus a real CSV emitter.

Grammar kept verbatim so scripts/parse.py (and the reference's own parser)
consume our logs (reference concurency/main.cpp:277,310-319, parse.py:17-26):

    # <mode> | <commands> | Starting Benchmarking...
    ## <mode> | <commands> | SUCCESS: ... | FAILURE: ...
"""

from __future__ import annotations

import csv
from dataclasses import dataclass
from typing import IO, Sequence

TOL_SPEEDUP = 0.3  # reference main.cpp:12


def format_time_info(time_us: float, bytes_moved: int = 0) -> str:
    s = f"{int(time_us)}us"
    if bytes_moved:
        gbps = 1e-3 * bytes_moved / time_us
        s += f" ({gbps:g} GBytes/s)"
    return s


def speedup_verdict(serial_us: float, concurrent_us: float,
                    theoretical_speedup: float,
                    bandwidth_gbps: float | None = None,
                    min_bandwidth: float | None = None) -> tuple[bool, str]:
    """Reference pass/fail: FAILURE if bandwidth floor missed, or if the
    theoretical speedup exceeds measured by more than 30% (main.cpp:311-319)."""
    if (min_bandwidth is not None and min_bandwidth >= 0
            and bandwidth_gbps is not None and bandwidth_gbps < min_bandwidth):
        return False, "FAILURE: Minimum Bandwidth not reached"
    speedup = serial_us / max(concurrent_us, 1e-9)
    if theoretical_speedup >= (1.0 + TOL_SPEEDUP) * speedup:
        return False, "FAILURE: Far from Theoretical Speedup"
    return True, "SUCCESS: Close from Theoretical Speedup"


def verdict_line(mode: str, commands: Sequence[str], verdict: str) -> str:
    return f"## {mode} | {' '.join(commands)} | {verdict}"


@dataclass
class CsvReporter:
    """Machine-readable sibling of the log grammar."""

    fh: IO
    _writer: object = None

    FIELDS = ["mode", "commands", "serial_us", "concurrent_us",
              "theoretical_speedup", "speedup", "bandwidth_gbps", "verdict"]

    def __post_init__(self):
        self._writer = csv.DictWriter(self.fh, fieldnames=self.FIELDS)
        self._writer.writeheader()

    def row(self, **kw):
        self._writer.writerow({k: kw.get(k, "") for k in self.FIELDS})
        self.fh.flush()
