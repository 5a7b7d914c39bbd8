"""hpc_patterns_amd — MI355X-native GPU parallel-patterns miniapp suite.

A brand-new framework with the capabilities of argonne-lcf/HPC-Patterns,
designed for CDNA4/gfx950 from scratch (see SURVEY.md for the structural map
of the reference suite):

- :mod:`hpc_patterns_amd.ops` — hand-written HIP kernels (busy-wait FMA/MFMA,
  shader copy, accumulate, fills, exact device reductions) exposed on torch
  tensors.
- :mod:`hpc_patterns_amd.concurrency` — the multi-hipStream / hipGraph
  stream-concurrency benchmark engine with the reference's CLI, autotuner and
  pass/fail criteria.
- :mod:`hpc_patterns_amd.parallel` — distributed patterns: hand ring
  all-reduce (plain + chunked-pipelined), pairwise P2P exchange, device-buffer
  ping-pong, xGMI topology discovery and rank->GPU placement policies. One
  process per GPU over torch.distributed (RCCL on ROCm, gloo on CPU).
- :mod:`hpc_patterns_amd.models` — the flagship "pattern step" workloads
  benchmarked by bench.py.
- :mod:`hpc_patterns_amd.utils` — timing harnesses (min-over-reps, hipEvent),
  the reference-compatible log/CSV reporters and the log parser.

Native core: ``hpc_patterns_amd._hpk`` (pybind11 + HIP, built in-tree by
``make ext`` / ``__graft_entry__.build()``); standalone binaries in ``bin/``.
"""

__version__ = "0.1.0"

from . import utils  # noqa: F401
