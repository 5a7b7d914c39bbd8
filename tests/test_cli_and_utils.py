"""CPU tests: concurrency CLI parsing, scripts/parse.py round-trip, timing
harness utilities."""

import subprocess
import sys
import time
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent


def test_cli_parse_modes_and_lists():
    from hpc_patterns_amd.concurrency.cli import parse_argv

    args, overrides = parse_argv(
        ["graph", "--commands", "C", "M2D", "--commands", "D2D",
         "--globalsize_MD", "1000", "--tripcount_C", "7",
         "--queues", "4", "--min_bandwidth", "12.5"])
    assert args.mode == "graph"
    assert args.commands == [["C", "M2D"], ["D2D"]]
    assert overrides == {"globalsize_MD": 1000}
    assert args.tripcount_C == 7
    assert args.queues == 4
    assert args.min_bandwidth == 12.5


def test_cli_rejects_bad_mode():
    from hpc_patterns_amd.concurrency.cli import parse_argv

    with pytest.raises(SystemExit):
        parse_argv(["warp_speed", "--commands", "C"])


def test_cli_rejects_unknown_flag():
    from hpc_patterns_amd.concurrency.cli import parse_argv

    with pytest.raises(SystemExit):
        parse_argv(["serial", "--commands", "C", "--bogus", "1"])


def test_parse_script_roundtrip(tmp_path):
    log = tmp_path / "x.log"
    log.write_text(
        "export GPU_MAX_HW_QUEUES=8\n"
        "## in_order | C DD | SUCCESS: Close from Theoretical Speedup\n")
    res = subprocess.run([sys.executable, str(REPO / "scripts/parse.py"),
                          str(log)], capture_output=True, text=True)
    assert res.returncode == 0
    assert "SUCCESS" in res.stdout and "C DD" in res.stdout


def test_min_over_reps():
    from hpc_patterns_amd.utils.timing import MinOverReps

    calls = []

    def fn():
        calls.append(1)
        time.sleep(0.001)

    m = MinOverReps(reps=5, warmup=2)
    best = m.run(fn)
    assert len(calls) == 7
    assert len(m.times) == 5
    assert best == min(m.times) > 0


def test_dist_interval_max_no_dist():
    from hpc_patterns_amd.utils.timing import dist_interval_max

    assert dist_interval_max(1.25) == 1.25


def test_csv_reporter(tmp_path):
    import io

    from hpc_patterns_amd.utils.report import CsvReporter

    buf = io.StringIO()
    r = CsvReporter(buf)
    r.row(mode="graph", commands="C DD", serial_us=100, concurrent_us=60,
          theoretical_speedup=2.0, speedup=1.67, bandwidth_gbps=5.5,
          verdict="SUCCESS")
    lines = buf.getvalue().strip().splitlines()
    assert lines[0].startswith("mode,commands")
    assert "graph" in lines[1] and "SUCCESS" in lines[1]


def test_bench_help():
    res = subprocess.run([sys.executable, str(REPO / "bench.py"), "--help"],
                         capture_output=True, text=True)
    assert res.returncode == 0
    for flag in ("--gpus", "--steps", "--warmup"):
        assert flag in res.stdout


def test_gpu_mapping_script_policies(tmp_path):
    script = REPO / "scripts/gpu_mapping.sh"
    env = {"PATH": "/usr/bin:/bin", "LOCAL_RANK": "3", "WORLD_SIZE": "8",
           "HPK_NGPUS": "8"}
    res = subprocess.run(["bash", str(script), "compact", "sh", "-c",
                          "echo dev=$HIP_VISIBLE_DEVICES"],
                         capture_output=True, text=True, env=env)
    assert res.returncode == 0, res.stderr
    assert "dev=3" in res.stdout

    env["LOCAL_RANK"] = "1"
    env["WORLD_SIZE"] = "2"
    res = subprocess.run(["bash", str(script), "spread", "sh", "-c",
                          "echo dev=$HIP_VISIBLE_DEVICES"],
                         capture_output=True, text=True, env=env)
    assert res.returncode == 0, res.stderr
    assert "dev=4" in res.stdout


def test_env_knobs_listing(monkeypatch):
    from hpc_patterns_amd.utils.config import check_env, env_knobs

    knobs = env_knobs()
    assert "HSA_ENABLE_SDMA" in knobs and "GPU_MAX_HW_QUEUES" in knobs
    monkeypatch.setenv("HSA_ENABLE_SDMA", "0")
    monkeypatch.setenv("HSA_ENABLE_IPC_MODE_LEGACY", "1")
    warns = check_env()
    assert len(warns) == 2
    monkeypatch.setenv("HSA_ENABLE_SDMA", "1")
    monkeypatch.setenv("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    assert check_env() == []


def test_native_binary_help_paths():
    """The native CLIs print usage before any HIP call, so their argument
    handling is testable without a GPU (skipped if binaries not built)."""
    conc = REPO / "bin/hpk_conc"
    if not conc.exists():
        pytest.skip("binaries not built")
    res = subprocess.run([str(conc)], capture_output=True, text=True)
    assert res.returncode == 1 and "Usage" in res.stdout
    res = subprocess.run([str(conc), "bogus_mode", "--commands", "C"],
                         capture_output=True, text=True)
    assert res.returncode == 1 and "unknown mode" in res.stdout
    res = subprocess.run([str(conc), "in_order", "--commands", "Q2D"],
                         capture_output=True, text=True)
    assert res.returncode == 1 and "unsupported COMMAND" in res.stdout
    for b in ("hpk_allreduce", "hpk_p2p", "hpk_membench"):
        res = subprocess.run([str(REPO / "bin" / b), "--help"],
                             capture_output=True, text=True)
        assert res.returncode == 0 and "Usage" in res.stdout, b
