"""Concurrency-bench engine wrapper + pure logic (CPU-testable).

The measurement itself runs in native/conc.hip; this module owns the
command DSL, defaults, and the autotuner's linear model — the same logic the
standalone hpk_conc binary implements in C++ (cpp/conc_main.cpp), re-exposed
for Python callers and for CPU unit tests.

Reference semantics being preserved (re-implemented, not copied):
- command letters M/D/H/S -> malloc/hipMalloc/hipHostMalloc/hipMallocManaged
  (reference bench_sycl.cpp:55-72);
- defaults: tripcount_C=40000, globalsize_C=1, copies ~1 GB of floats
  (reference main.cpp:96-105);
- autotune: run the unique commands serially once, rescale every '-1'
  parameter linearly so each command takes the time of the fastest copy
  (reference main.cpp:226-258).
"""

from __future__ import annotations

from typing import Mapping, Sequence

ALLOWED_MODES = ("serial", "in_order", "out_of_order", "graph",
                 "graph_explicit", "host_threads", "nowait")

DEFAULT_TRIPCOUNT = 40_000
DEFAULT_GLOBALSIZE_C = 1
DEFAULT_COPY_FLOATS = int(1e9 / 4)  # ~1 GB
MEMORY_LETTERS = "MDHS"
_BANNED = {"HM", "MH", "MM", "HH"}  # host->host measures nothing on-GPU


def sanitize_command(cmd: str) -> str:
    """'M2D' -> 'MD', 'C' -> 'C' (reference main.cpp sanitize_command)."""
    return cmd.replace("2", "")


def validate_command(cmd: str) -> str:
    """Returns the sanitized command or raises ValueError."""
    sc = sanitize_command(cmd)
    if sc == "C":
        return sc
    if (len(sc) == 2 and all(c in MEMORY_LETTERS for c in sc)
            and sc not in _BANNED):
        return sc
    raise ValueError(f"unsupported COMMAND '{cmd}' "
                     f"(C or A2B with A,B in {{{MEMORY_LETTERS}}})")


def tuned_param_name(cmd: str) -> str:
    return "tripcount_C" if cmd == "C" else f"globalsize_{cmd}"


def default_params(commands: Sequence[str],
                   overrides: Mapping[str, int] | None = None,
                   default_memory: int = -1) -> dict[str, int]:
    """Resolve the parameter map for a command list; -1/absent means default."""
    overrides = dict(overrides or {})
    params: dict[str, int] = {}

    def resolve(name: str, dflt: int):
        v = overrides.get(name, -1)
        params[name] = dflt if v == -1 else int(v)

    resolve("tripcount_C", DEFAULT_TRIPCOUNT)
    resolve("globalsize_C", DEFAULT_GLOBALSIZE_C)
    copy_default = default_memory if default_memory > 0 else DEFAULT_COPY_FLOATS
    for cmd in commands:
        sc = validate_command(cmd)
        if sc != "C":
            resolve(f"globalsize_{sc}", copy_default)
    return params


def autotune_rescale(unique_commands: Sequence[str],
                     measured_us: Sequence[float],
                     params: Mapping[str, int],
                     auto_flags: Mapping[str, bool]) -> dict[str, int]:
    """Linear rescale of every auto-tuned parameter so each command matches
    the fastest copy command's time (pure function; the linearity assumption:
    time is proportional to tripcount for C and to buffer size for copies).

    auto_flags[name] is True when the user left that parameter at -1.
    """
    new = dict(params)
    copies = [t for c, t in zip(unique_commands, measured_us) if c != "C"]
    target = min(copies) if copies else max(measured_us)
    for cmd, t in zip(unique_commands, measured_us):
        name = tuned_param_name(cmd)
        if auto_flags.get(name, False) and t > 0:
            new[name] = max(int(target / t * params[name]), 1)
    return new


COPY_ENGINES = {"auto": 0, "shader": 1, "sdma": 2}


def run_bench(mode: str, commands: Sequence[str],
              params: Mapping[str, int] | None = None,
              enable_profiling: bool = False, n_queues: int = -1,
              n_repetitions: int = 10, verbose: bool = False,
              use_copy_kernel: bool = False,
              copy_engine: str = "auto") -> dict:
    """Run the native engine. Requires the _hpk extension and a GPU.

    copy_engine: auto (hipMemcpyAsync) | shader (K2 copy kernel) | sdma
    (explicit hsa_amd_memory_async_copy_on_engine). use_copy_kernel=True is
    a legacy alias for copy_engine="shader"."""
    from .._native import native

    if mode not in ALLOWED_MODES:
        raise ValueError(f"mode '{mode}' not in {ALLOWED_MODES}")
    if copy_engine not in COPY_ENGINES:
        raise ValueError(f"copy_engine '{copy_engine}' not in {list(COPY_ENGINES)}")
    engine = COPY_ENGINES["shader"] if use_copy_kernel else COPY_ENGINES[copy_engine]
    cmds = [validate_command(c) for c in commands]
    p = default_params(cmds, params)
    # pass through engine-level extras (e.g. payload_C_mfma) untouched
    for k, v in dict(params or {}).items():
        if k not in p and v != -1:
            p[k] = int(v)
    return native().conc_bench(mode, cmds, {k: int(v) for k, v in p.items()},
                               enable_profiling, n_queues, n_repetitions,
                               verbose, engine)
