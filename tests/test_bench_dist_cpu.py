"""End-to-end CPU check of the bench.py driver contract under torchrun:
2 ranks, gloo, exactly the launch line the driver uses for N>1 (with --cpu).
Validates rendezvous, the timed-region bracketing, MAX-over-ranks, and the
one-JSON-line output."""

import json
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent

pytestmark = pytest.mark.timeout(300)


def test_bench_torchrun_cpu(dist_env):
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node", "2",
        "--master-addr", "127.0.0.1",
        "--master-port", dist_env["MASTER_PORT"],
        str(REPO / "bench.py"), "--gpus", "2", "--steps", "3",
        "--warmup", "1", "--cpu",
    ]
    res = subprocess.run(cmd, capture_output=True, text=True, timeout=240,
                         cwd=REPO)
    assert res.returncode == 0, res.stdout + res.stderr
    json_lines = [l for l in res.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1, res.stdout  # exactly one line, from rank 0
    data = json.loads(json_lines[0])
    assert data["metric"] == "cpu_plumbing_check"
    assert data["n_gpus"] == 2
    assert data["steps"] == 3
    assert data["config"]["parallelism"] == "dp2"
    assert "pingpong_us" in data["components"]
    assert data["components"]["p2p_checksum_ok"] is True


def test_bench_single_rank_cpu():
    res = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--steps", "2",
         "--warmup", "1", "--cpu"],
        capture_output=True, text=True, timeout=120, cwd=REPO)
    assert res.returncode == 0, res.stdout + res.stderr
    data = json.loads([l for l in res.stdout.splitlines()
                       if l.startswith("{")][-1])
    assert data["n_gpus"] == 1 and data["value"] > 0


def test_p2p_latency_sweep_cli(dist_env):
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node", "2",
        "--master-addr", "127.0.0.1",
        "--master-port", dist_env["MASTER_PORT"],
        "-m", "hpc_patterns_amd.parallel.p2p",
        "--max-bytes", "1024", "--iters", "5", "--pair-floats", "4096",
    ]
    res = subprocess.run(cmd, capture_output=True, text=True, timeout=240,
                         cwd=REPO)
    assert res.returncode == 0, res.stdout + res.stderr
    assert "bytes,oneway_us" in res.stdout
    assert "pairwise unidirectional" in res.stdout
    assert "checksum_ok=True" in res.stdout


def test_allreduce_sweep_cli(dist_env):
    import os

    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node", "2",
        "--master-addr", "127.0.0.1",
        "--master-port", dist_env["MASTER_PORT"],
        "-m", "hpc_patterns_amd.parallel.sweep",
        "--min-mb", "0.01", "--max-mb", "0.02", "--iters", "2",
        "--algos", "rccl,ring,rsag",
    ]
    # the channel setting must land in the CSV's channels column (the
    # channel-study contract, scripts/run_allreduce_sweep.sh --channels)
    env = dict(os.environ)
    env["NCCL_MIN_NCHANNELS"] = "4"
    env["NCCL_MAX_NCHANNELS"] = "4"
    res = subprocess.run(cmd, capture_output=True, text=True, timeout=240,
                         cwd=REPO, env=env)
    assert res.returncode == 0, res.stdout + res.stderr
    assert "algo,channels,bytes,time_s,alg_GBps,bus_GBps" in res.stdout
    for algo in ("rccl", "ring", "rsag"):
        assert f"\n{algo},4," in res.stdout, algo


def test_bench_torchrun_cpu_world8(dist_env):
    """The driver's N=8 launch shape, rehearsed end to end on gloo."""
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node", "8",
        "--master-addr", "127.0.0.1",
        "--master-port", dist_env["MASTER_PORT"],
        str(REPO / "bench.py"), "--gpus", "8", "--steps", "2",
        "--warmup", "1", "--cpu",
    ]
    res = subprocess.run(cmd, capture_output=True, text=True, timeout=240,
                         cwd=REPO)
    assert res.returncode == 0, res.stdout + res.stderr
    data = json.loads([l for l in res.stdout.splitlines()
                       if l.startswith("{")][0])
    assert data["n_gpus"] == 8
    assert data["config"]["parallelism"] == "dp8"
    assert data["components"]["p2p_checksum_ok"] is True


def test_bench_component_fault_degrades_to_null(dist_env):
    """Fault-injection rehearsal: a failing diagnostic
    component must yield a null JSON field and rc 0, not kill the bench."""
    import os

    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node", "2",
        "--master-addr", "127.0.0.1",
        "--master-port", dist_env["MASTER_PORT"],
        str(REPO / "bench.py"), "--gpus", "2", "--steps", "2",
        "--warmup", "1", "--cpu",
    ]
    env = dict(os.environ)
    env["HPK_BENCH_FAULT"] = "pingpong_us"
    res = subprocess.run(cmd, capture_output=True, text=True, timeout=240,
                         cwd=REPO, env=env)
    assert res.returncode == 0, res.stdout + res.stderr
    data = json.loads([l for l in res.stdout.splitlines()
                       if l.startswith("{")][0])
    assert data["components"]["pingpong_us"] is None
    # the OTHER components still measured
    assert data["components"]["p2p_checksum_ok"] is True
    assert "injected fault" in res.stderr + res.stdout


def test_bench_policy_flag_cpu():
    """--policy must parse and land in the config record."""
    res = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--steps", "2",
         "--warmup", "1", "--cpu", "--policy", "spread"],
        capture_output=True, text=True, timeout=120, cwd=REPO)
    assert res.returncode == 0, res.stdout + res.stderr
    data = json.loads([l for l in res.stdout.splitlines()
                       if l.startswith("{")][-1])
    assert data["config"]["placement_policy"] == "spread"


def test_policy_sweep_cli_writes_csv(dist_env, tmp_path):
    """policy_sweep CLI end to end on gloo: CSV file with header + one row."""
    import os

    csv = tmp_path / "policy.csv"
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node", "2",
        "--master-addr", "127.0.0.1",
        "--master-port", dist_env["MASTER_PORT"],
        "-m", "hpc_patterns_amd.parallel.policy_sweep",
        "--policy", "spread", "--floats", "4096", "--iters", "2",
        "--backend", "gloo", "--csv", str(csv),
    ]
    res = subprocess.run(cmd, capture_output=True, text=True, timeout=240,
                         cwd=REPO)
    assert res.returncode == 0, res.stdout + res.stderr
    lines = csv.read_text().strip().splitlines()
    assert lines[0].startswith("policy,world,engine,bytes")
    assert lines[1].startswith("spread,2,gloo,16384,")
