"""CPU tests of the real-MPI miniapps (hpk_mpi_allreduce / hpk_mpi_p2p)
in their -M host-buffer mode: genuine mpirun-launched multi-rank MPI
pt2pt, collective and RMA runs with no GPU required — BASELINE.json
config[0] ("MPI host-buffer ping-pong on CPU, world_size=2") made real.

The -H/-D/-S GPU modes of the same binaries run under ctest and ci_gpu.sh
on the MI355X box.
"""

import shutil
import subprocess
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent
MPIRUN = "/opt/conda/bin/mpirun"

pytestmark = pytest.mark.timeout(300)

needs_mpi = pytest.mark.skipif(
    not Path(MPIRUN).exists(), reason="no MPICH in image")


def _build(name: str) -> Path:
    binpath = REPO / "bin" / name
    if not binpath.exists():
        res = subprocess.run(["make", f"bin/{name}"], cwd=REPO,
                             capture_output=True, text=True, timeout=600)
        assert res.returncode == 0, res.stdout + res.stderr
    return binpath


def _mpirun(np: int, *args: str):
    return subprocess.run([MPIRUN, "-np", str(np), *args],
                          capture_output=True, text=True, timeout=240,
                          cwd=REPO)


@needs_mpi
@pytest.mark.parametrize("np", [2, 4])
def test_mpi_allreduce_ring_host(np):
    b = _build("hpk_mpi_allreduce")
    res = _mpirun(np, str(b), "-M", "-p", "16", "-i", "2")
    assert res.returncode == 0, res.stdout + res.stderr
    for r in range(np):
        assert f"Passed rank {r}" in res.stdout
    assert f"ranks={np}" in res.stdout


@needs_mpi
@pytest.mark.parametrize("dtype", ["float", "int"])
def test_mpi_allreduce_native_host(dtype):
    b = _build("hpk_mpi_allreduce")
    res = _mpirun(4, str(b), "-M", "-p", "16", "-i", "2", "-a", "-t", dtype)
    assert res.returncode == 0, res.stdout + res.stderr
    assert "algo=allreduce" in res.stdout
    assert "Passed rank 3" in res.stdout


@needs_mpi
def test_mpi_allreduce_odd_ranks_rejected():
    b = _build("hpk_mpi_allreduce")
    res = _mpirun(3, str(b), "-M", "-p", "10", "-i", "1")
    assert res.returncode != 0
    assert "even number of ranks" in res.stdout + res.stderr


@needs_mpi
@pytest.mark.parametrize("engine", ["isend", "win"])
def test_mpi_p2p_host(engine):
    """Two-sided and one-sided RMA engines: both phases, checksummed
    in-binary (reference peer2pear protocol over real MPI)."""
    b = _build("hpk_mpi_p2p")
    res = _mpirun(4, str(b), "--engine", engine, "-M",
                  "--floats", str(1 << 20))
    assert res.returncode == 0, res.stdout + res.stderr
    assert f"mpi-{engine} Unidirectional Bandwidth" in res.stdout
    assert f"mpi-{engine} Bidirectional Bandwidth" in res.stdout
    assert "pairs=2" in res.stdout


@needs_mpi
def test_mpi_p2p_world2(engine="isend"):
    b = _build("hpk_mpi_p2p")
    res = _mpirun(2, str(b), "--engine", engine, "-M",
                  "--floats", str(1 << 18))
    assert res.returncode == 0, res.stdout + res.stderr
    assert "pairs=1" in res.stdout
