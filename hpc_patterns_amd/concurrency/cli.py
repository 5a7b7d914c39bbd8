"""Reference-compatible concurrency-bench CLI (python twin of cpp/conc_main).

    python -m hpc_patterns_amd.concurrency.cli <mode> [options] --commands ...

Same surface and log grammar as the reference driver
(reference concurency/main.cpp:115-322) and as the native hpk_conc binary;
useful where the engine should run inside a Python process (e.g. under
torch, or from the test-suite) instead of the standalone binary.
"""

from __future__ import annotations

import argparse
import sys

from ..utils.report import format_time_info, speedup_verdict, verdict_line
from .engine import (
    ALLOWED_MODES,
    autotune_rescale,
    default_params,
    run_bench,
    tuned_param_name,
    validate_command,
)


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(
        prog="hpk-conc",
        description="MI355X stream-concurrency benchmark "
                    f"(modes: {'|'.join(ALLOWED_MODES)})")
    p.add_argument("mode", choices=ALLOWED_MODES)
    p.add_argument("--enable_profiling", action="store_true")
    p.add_argument("--verbose", action="store_true")
    p.add_argument("--copy_kernel", action="store_true",
                   help="legacy alias for --copy_engine shader")
    p.add_argument("--copy_engine", choices=["auto", "shader", "sdma"],
                   default="auto",
                   help="copy path for A2B commands: runtime-picked | "
                        "hand-written kernel | explicit SDMA engine")
    p.add_argument("--tripcount_C", type=int, default=-1)
    p.add_argument("--globalsize_C", type=int, default=-1)
    p.add_argument("--globalsize_default_memory", type=int, default=-1)
    p.add_argument("--queues", type=int, default=-1)
    p.add_argument("--repetitions", type=int, default=10)
    p.add_argument("--min_bandwidth", type=float, default=-1)
    p.add_argument("--commands", action="append", nargs="+", required=True,
                   metavar="CMD")
    return p


def parse_argv(argv):
    """Also accept --globalsize_<CMD> for any command (dynamic flags)."""
    parser = build_parser()
    known, unknown = parser.parse_known_args(argv)
    overrides = {}
    it = iter(unknown)
    for tok in it:
        if tok.startswith("--globalsize_") or tok.startswith("--tripcount_"):
            try:
                overrides[tok[2:]] = int(next(it))
            except StopIteration:
                parser.error(f"missing value for {tok}")
        else:
            parser.error(f"unrecognized argument {tok}")
    return known, overrides


def main(argv=None) -> int:
    args, extra_overrides = parse_argv(argv if argv is not None else sys.argv[1:])

    l_commands = [[validate_command(c) for c in lst] for lst in args.commands]
    overrides = {"tripcount_C": args.tripcount_C,
                 "globalsize_C": args.globalsize_C}
    overrides.update(extra_overrides)

    all_cmds = sorted({c for lst in l_commands for c in lst})
    params = default_params(all_cmds, overrides,
                            default_memory=args.globalsize_default_memory)
    auto_flags = {tuned_param_name(c): overrides.get(tuned_param_name(c), -1) == -1
                  for c in all_cmds}

    if any(auto_flags.values()) and len(all_cmds) > 1:
        print("# Performing Autotuning to Balance Commands Times")
        base = run_bench("serial", all_cmds, params, n_repetitions=args.repetitions,
                         use_copy_kernel=args.copy_kernel,
                         copy_engine=args.copy_engine)
        params = autotune_rescale(all_cmds, base["per_cmd_us"], params, auto_flags)

    print("Parameters used:")
    for c in all_cmds:
        print(f"  {tuned_param_name(c)}: {params[tuned_param_name(c)]}")
        if c == "C":
            print(f"  globalsize_C: {params['globalsize_C']}")

    exit_code = 0
    for cmds in l_commands:
        label = f"{args.mode} | {' '.join(cmds)} "
        print(f"# {label}| Starting Benchmarking...")
        serial = run_bench("serial", cmds, params,
                           enable_profiling=args.enable_profiling,
                           n_queues=args.queues, n_repetitions=args.repetitions,
                           verbose=args.verbose, use_copy_kernel=args.copy_kernel,
                           copy_engine=args.copy_engine)
        print(f"Minimum Measured Total Time Serial: {serial['total_us']}us")
        for i, c in enumerate(cmds):
            nbytes = params[f"globalsize_{c}"] * 4 if c != "C" else 0
            print(f"  Minimum Time Command {i} ({c:>3}): "
                  f"{format_time_info(serial['per_cmd_us'][i], nbytes)}")
        theoretical = serial["total_us"] / max(max(serial["per_cmd_us"]), 1)
        print(f"Maximum Theoretical Speedup: {theoretical:g}x")
        if len(cmds) > 1 and theoretical <= 1.5:
            print("  WARNING: Large Unbalance Between Commands", file=sys.stderr)

        conc = run_bench(args.mode, cmds, params,
                         enable_profiling=args.enable_profiling,
                         n_queues=args.queues, n_repetitions=args.repetitions,
                         verbose=args.verbose, use_copy_kernel=args.copy_kernel,
                         copy_engine=args.copy_engine)
        nbytes = sum(params[f"globalsize_{c}"] * 4 for c in cmds if c != "C")
        gbps = 1e-3 * nbytes / conc["total_us"] if nbytes else None
        print(f"Minimum Measured Total Time //: "
              f"{format_time_info(conc['total_us'], nbytes)}")
        speedup = serial["total_us"] / max(conc["total_us"], 1)
        print(f"Speedup Relative to Serial: {speedup:g}x")
        ok, verdict = speedup_verdict(serial["total_us"], conc["total_us"],
                                      theoretical, gbps, args.min_bandwidth)
        print(verdict_line(args.mode, cmds, verdict))
        if not ok:
            exit_code = 1
    return exit_code


if __name__ == "__main__":
    sys.exit(main())
