"""Loader for the native core extension (_hpk).

Policy (matches the driver's "native code must actually load" check):

- On a machine WITH a GPU, the extension is mandatory: any op touching the
  GPU raises immediately if `_hpk` is missing, rather than silently falling
  back to eager PyTorch.
- On CPU-only machines (the build container), importing the package works
  without the extension so CPU tests and tooling run; only GPU paths demand
  it.
"""

from __future__ import annotations

import importlib
import subprocess
from pathlib import Path

_REPO_ROOT = Path(__file__).resolve().parent.parent

_hpk = None
_import_error: Exception | None = None

try:
    _hpk = importlib.import_module("hpc_patterns_amd._hpk")
except ImportError as e:  # extension not built (or wrong arch)
    _import_error = e


def build_native(verbose: bool = False) -> None:
    """Build the in-tree native extension + binaries with make/hipcc."""
    global _hpk, _import_error
    cmd = ["make", "-C", str(_REPO_ROOT), "-j", "ext", "bins"]
    res = subprocess.run(cmd, capture_output=not verbose, text=True)
    if res.returncode != 0:
        raise RuntimeError(
            f"native build failed (rc={res.returncode}):\n"
            f"{res.stdout or ''}\n{res.stderr or ''}"
        )
    if _hpk is None:
        _hpk = importlib.import_module("hpc_patterns_amd._hpk")
        _import_error = None


def have_native() -> bool:
    return _hpk is not None


def native():
    """Return the _hpk module, raising loudly if it is unavailable."""
    if _hpk is None:
        raise RuntimeError(
            "hpc_patterns_amd native extension (_hpk) is not built — run "
            "`make ext` (or __graft_entry__.build()). Refusing to fall back "
            f"to an eager path. Original import error: {_import_error}"
        ) from _import_error
    return _hpk


def gpu_available() -> bool:
    try:
        import torch

        return torch.cuda.is_available()
    except Exception:
        return False


def require_native_on_gpu():
    """On a GPU machine the native path is the only allowed path."""
    if gpu_available():
        return native()
    raise RuntimeError("no GPU available")
