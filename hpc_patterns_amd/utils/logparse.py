"""Log parser — groups `## mode | commands | SUCCESS/FAILURE` verdict lines
by the preceding `export ...` environment marker.

Same log grammar as the reference's concurency/parse.py:1-32, implemented
fresh (dict output + optional table rendering instead of tabulate-only)."""

from __future__ import annotations

from collections import defaultdict
from typing import Iterable


def parse_log(lines: Iterable[str]) -> dict:
    """Returns {env_marker: {commands: {mode: SUCCESS|FAILURE}}}."""
    result: dict = defaultdict(lambda: defaultdict(dict))
    env = None
    for line in lines:
        if "export" in line:
            env = line.split("export", 1)[1].strip()
        elif "FAILURE" in line or "SUCCESS" in line:
            verdict = "FAILURE" if "FAILURE" in line else "SUCCESS"
            body = line.strip()
            if body.startswith("##"):
                body = body[2:]
            parts = body.split("|")
            if len(parts) < 2:
                continue
            mode = parts[0].strip()
            commands = " ".join(parts[1].split())
            result[env][commands][mode] = verdict
    return {k: {c: dict(m) for c, m in v.items()} for k, v in result.items()}


def render_table(parsed: dict) -> str:
    """Plain-text table per env group (tabulate-free)."""
    out = []
    for env, groups in parsed.items():
        out.append(str(env))
        modes = sorted({m for g in groups.values() for m in g})
        header = ["commands"] + modes
        rows = [[cmds] + [groups[cmds].get(m, "") for m in modes]
                for cmds in groups]
        widths = [max(len(str(r[i])) for r in [header] + rows)
                  for i in range(len(header))]
        fmt = "  ".join(f"{{:<{w}}}" for w in widths)
        out.append(fmt.format(*header))
        out.append("  ".join("-" * w for w in widths))
        for r in rows:
            out.append(fmt.format(*r))
        out.append("")
    return "\n".join(out)
