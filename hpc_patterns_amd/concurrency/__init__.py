"""concurrency — the single-GPU stream-concurrency benchmark.

Python face of the native hipStream/hipGraph engine (native/conc.hip) plus
the CPU-testable pieces: command-DSL validation, default parameters, the
autotuner's linear rescale, and the verdict logic.
"""

from .engine import (  # noqa: F401
    ALLOWED_MODES,
    DEFAULT_COPY_FLOATS,
    DEFAULT_TRIPCOUNT,
    autotune_rescale,
    default_params,
    run_bench,
    sanitize_command,
    tuned_param_name,
    validate_command,
)
