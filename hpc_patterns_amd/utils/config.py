"""Config surface of the suite — the three tiers the reference uses
(SURVEY.md §5.6), with ROCm names:

1. CLI flags (per binary/module — see docs/).
2. Environment knobs (below): runtime/tuning surface.
3. Compile-time: Makefile vars (GPU_ARCH, HIPCC) and CMake cache entries.

`env_knobs()` returns every knob with its current value; `check_env()`
flags suspicious settings (e.g. the blit-copy mode that serializes
H2D||D2H — measured in profiles/h2d_d2h_matrix_r4.log).
"""

from __future__ import annotations

import os
from dataclasses import dataclass


@dataclass(frozen=True)
class Knob:
    name: str
    purpose: str
    owner: str  # rocm | rccl | suite


KNOBS = [
    Knob("HIP_VISIBLE_DEVICES", "device selection (reference ZE_AFFINITY_MASK)", "rocm"),
    Knob("GPU_MAX_HW_QUEUES", "hardware queues per device (stream concurrency width)", "rocm"),
    Knob("HSA_ENABLE_SDMA", "SDMA copy engines; 0 forces shader-blit copies "
         "(measured: breaks H2D||D2H overlap)", "rocm"),
    Knob("HSA_ENABLE_IPC_MODE_LEGACY", "must be 0 on this stack: dmabuf IPC "
         "(hipIpc* fails with legacy mode)", "rocm"),
    Knob("HSA_XNACK", "page-fault-capable memory (managed-memory debugging)", "rocm"),
    Knob("NCCL_MIN_NCHANNELS", "RCCL channel floor — spread collectives over "
         "more xGMI links", "rccl"),
    Knob("NCCL_MAX_NCHANNELS", "RCCL channel ceiling", "rccl"),
    Knob("NCCL_DEBUG", "RCCL logging (INFO/TRACE)", "rccl"),
    Knob("HPK_PINNED_FLAGS", "pinned-alloc flavour for H buffers: default|nc|wc", "suite"),
    Knob("HPK_LAUNCH_TIMEOUT", "fork-launcher watchdog seconds (default 600)", "suite"),
    Knob("HPK_NGPUS", "GPU-count override for gpu_mapping.sh", "suite"),
    Knob("HPK_COPY_STREAM_PRIORITY", "high = raise copy-pool stream priority "
         "(co-scheduling-limited pods)", "suite"),
    Knob("HPK_PLACEMENT_POLICY", "default rank->GPU policy for bench.py: "
         "compact|spread|topo", "suite"),
    Knob("HPK_NCCL_TIMEOUT_S", "process-group collective timeout for bench.py "
         "(default 300)", "suite"),
    Knob("HPK_BENCH_FAULT", "fault injection: named bench component raises "
         "(CI rehearsal of null-field degradation)", "suite"),
    Knob("MASTER_ADDR", "torch.distributed rendezvous address (use 127.0.0.1)", "suite"),
    Knob("MASTER_PORT", "torch.distributed rendezvous port", "suite"),
]


def env_knobs() -> dict[str, dict]:
    return {k.name: {"value": os.environ.get(k.name), "purpose": k.purpose,
                     "owner": k.owner} for k in KNOBS}


def check_env() -> list[str]:
    """Returns human-readable warnings for settings known to hurt."""
    warnings = []
    if os.environ.get("HSA_ENABLE_SDMA") == "0":
        warnings.append(
            "HSA_ENABLE_SDMA=0 forces shader-blit copies: H2D||D2H overlap "
            "regresses ~30% and copies steal CUs from compute "
            "(profiles/h2d_d2h_matrix_r4.log)")
    if os.environ.get("HSA_ENABLE_IPC_MODE_LEGACY") not in (None, "0"):
        warnings.append(
            "HSA_ENABLE_IPC_MODE_LEGACY must be 0 on this driver: legacy IPC "
            "fails with hipIpcGetMemHandle: invalid argument")
    return warnings
