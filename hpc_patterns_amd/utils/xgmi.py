"""Per-link xGMI traffic sampling — scaffolding for the RCCL channel/link
tuning study (a ring all-reduce is bound by ONE ~153 GB/s
xGMI link; RCCL spreads channels across all 7 — the sweep needs per-link
evidence, not just one bus-bandwidth number).

`amd-smi xgmi` (and the underlying amdsmi library) expose cumulative
per-link read/write counters on MI300+-class parts. The exact JSON schema
varies by tool version, so the parser is schema-tolerant: it walks the JSON
recursively and collects every numeric leaf whose key mentions read/write,
keyed by its path. A before/after delta around a collective then shows
which links carried traffic. Returns None cleanly when the tool or the
counters are absent (CPU boxes, driver without xgmi metrics).
"""

from __future__ import annotations

import json
import subprocess

_KEYS = ("read", "write", "tx", "rx")


def sample_xgmi() -> dict[str, float] | None:
    """One flattened sample of the per-link counters, or None."""
    for cmd in (["amd-smi", "xgmi", "--json"],
                ["amd-smi", "metric", "--xgmi", "--json"]):
        try:
            res = subprocess.run(cmd, capture_output=True, text=True,
                                 timeout=20)
        except (OSError, subprocess.TimeoutExpired):
            return None
        if res.returncode != 0 or not res.stdout.strip():
            continue
        try:
            data = json.loads(res.stdout)
        except json.JSONDecodeError:
            continue
        flat = flatten_counters(data)
        if flat:
            return flat
    return None


def flatten_counters(data, prefix: str = "") -> dict[str, float]:
    """Pure: collect numeric leaves whose key path mentions read/write
    traffic. CPU-testable on canned JSON."""
    out: dict[str, float] = {}
    if isinstance(data, dict):
        for k, v in data.items():
            p = f"{prefix}.{k}" if prefix else str(k)
            out.update(flatten_counters(v, p))
    elif isinstance(data, list):
        for i, v in enumerate(data):
            out.update(flatten_counters(v, f"{prefix}[{i}]"))
    elif isinstance(data, (int, float)) and not isinstance(data, bool):
        low = prefix.lower()
        if any(k in low for k in _KEYS):
            out[prefix] = float(data)
    return out


def delta(before: dict[str, float] | None,
          after: dict[str, float] | None,
          min_delta: float = 1.0) -> dict[str, float]:
    """Pure: per-counter increase between two samples (only keys present in
    both; only deltas >= min_delta, so idle links drop out)."""
    if not before or not after:
        return {}
    out = {}
    for k, b in before.items():
        a = after.get(k)
        if a is not None and a - b >= min_delta:
            out[k] = a - b
    return out
