#!/usr/bin/env python3
"""Log -> table parser (reference concurency/parse.py CLI contract:
`parse.py <logfile> [tablefmt]`, groups verdict lines by `export` env
markers). Implementation lives in hpc_patterns_amd.utils.logparse."""

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from hpc_patterns_amd.utils.logparse import parse_log, render_table  # noqa: E402


def main() -> int:
    if len(sys.argv) < 2:
        print("usage: parse.py <logfile> [style]", file=sys.stderr)
        return 1
    with open(sys.argv[1]) as f:
        parsed = parse_log(f.readlines())
    try:
        from tabulate import tabulate

        style = sys.argv[2] if len(sys.argv) > 2 else "simple"
        for env, groups in parsed.items():
            rows = [{"commands": cmds, **modes} for cmds, modes in groups.items()]
            print(env)
            print(tabulate(rows, headers="keys", tablefmt=style))
            print()
    except ImportError:
        print(render_table(parsed))
    return 0


if __name__ == "__main__":
    sys.exit(main())
