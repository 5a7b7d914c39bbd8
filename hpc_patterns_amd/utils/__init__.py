"""utils — timing harnesses, reporters, log parser, config knobs."""

from .timing import MinOverReps, gpu_timer  # noqa: F401
from .report import verdict_line, format_time_info, CsvReporter  # noqa: F401
from .logparse import parse_log  # noqa: F401
