"""models — flagship pattern-step workloads benchmarked by bench.py.

The reference suite has no neural models; its "models" are
GPU parallel patterns. The flagship workload bundles them into one
benchmarkable step per GPU: compute/copy stream overlap + xGMI P2P exchange +
ring/native all-reduce — the BASELINE.json headline metrics in one step.
"""

from .flagship import FlagshipPatternStep, SMOKE_CONFIG  # noqa: F401
