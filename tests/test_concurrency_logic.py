"""CPU tests of the concurrency bench's pure logic: command DSL, defaults,
autotuner rescale, verdict criteria (reference main.cpp semantics)."""

import pytest

from hpc_patterns_amd.concurrency import (
    ALLOWED_MODES,
    DEFAULT_COPY_FLOATS,
    DEFAULT_TRIPCOUNT,
    autotune_rescale,
    default_params,
    sanitize_command,
    tuned_param_name,
    validate_command,
)
from hpc_patterns_amd.utils.report import (
    TOL_SPEEDUP,
    format_time_info,
    speedup_verdict,
    verdict_line,
)


def test_sanitize():
    assert sanitize_command("M2D") == "MD"
    assert sanitize_command("C") == "C"
    assert sanitize_command("D2D") == "DD"


@pytest.mark.parametrize("cmd", ["C", "M2D", "D2M", "H2D", "D2H", "D2D",
                                 "S2D", "D2S", "HD", "DH"])
def test_valid_commands(cmd):
    assert validate_command(cmd) == sanitize_command(cmd)


@pytest.mark.parametrize("cmd", ["X", "M2M", "H2H", "M2H", "H2M", "C2D",
                                 "DDD", ""])
def test_invalid_commands(cmd):
    with pytest.raises(ValueError):
        validate_command(cmd)


def test_modes():
    for m in ("serial", "in_order", "out_of_order", "graph", "host_threads",
              "nowait"):
        assert m in ALLOWED_MODES


def test_default_params():
    p = default_params(["C", "MD"])
    assert p["tripcount_C"] == DEFAULT_TRIPCOUNT
    assert p["globalsize_C"] == 1
    assert p["globalsize_MD"] == DEFAULT_COPY_FLOATS


def test_default_params_overrides():
    p = default_params(["C", "MD"], overrides={"tripcount_C": 5,
                                               "globalsize_MD": 100})
    assert p["tripcount_C"] == 5
    assert p["globalsize_MD"] == 100


def test_default_memory():
    p = default_params(["MD"], default_memory=12345)
    assert p["globalsize_MD"] == 12345


def test_tuned_param_name():
    assert tuned_param_name("C") == "tripcount_C"
    assert tuned_param_name("MD") == "globalsize_MD"


def test_autotune_rescale_linear():
    # C took 10x the fastest copy -> tripcount shrinks 10x
    cmds = ["C", "MD", "DH"]
    measured = [10_000.0, 1_000.0, 2_000.0]
    params = {"tripcount_C": 40_000, "globalsize_MD": 1_000_000,
              "globalsize_DH": 1_000_000}
    flags = {k: True for k in params}
    new = autotune_rescale(cmds, measured, params, flags)
    assert new["tripcount_C"] == 4_000
    assert new["globalsize_MD"] == 1_000_000  # already the target
    assert new["globalsize_DH"] == 500_000


def test_autotune_respects_user_fixed_params():
    cmds = ["C", "MD"]
    measured = [10_000.0, 1_000.0]
    params = {"tripcount_C": 40_000, "globalsize_MD": 1_000_000}
    flags = {"tripcount_C": False, "globalsize_MD": True}
    new = autotune_rescale(cmds, measured, params, flags)
    assert new["tripcount_C"] == 40_000  # untouched


def test_autotune_only_compute():
    # no copy commands: target = max measured, C unchanged relative to itself
    new = autotune_rescale(["C"], [5_000.0], {"tripcount_C": 100},
                           {"tripcount_C": True})
    assert new["tripcount_C"] == 100


def test_speedup_verdict_success():
    ok, msg = speedup_verdict(serial_us=1000, concurrent_us=550,
                              theoretical_speedup=2.0)
    assert ok and "SUCCESS" in msg


def test_speedup_verdict_failure_far():
    # measured speedup 1.0, theoretical 2.0 > 1.3*1.0 -> failure
    ok, msg = speedup_verdict(serial_us=1000, concurrent_us=1000,
                              theoretical_speedup=2.0)
    assert not ok and "Far from Theoretical" in msg


def test_speedup_verdict_tolerance_boundary():
    # measured speedup s, fails iff theoretical >= 1.3*s (reference >= test)
    s = 1000 / 800.0
    ok, _ = speedup_verdict(1000, 800, theoretical_speedup=1.3 * s)
    assert not ok
    ok, _ = speedup_verdict(1000, 800, theoretical_speedup=1.3 * s - 1e-6)
    assert ok
    assert TOL_SPEEDUP == 0.3


def test_speedup_verdict_bandwidth_floor():
    ok, msg = speedup_verdict(1000, 500, 1.5, bandwidth_gbps=10.0,
                              min_bandwidth=20.0)
    assert not ok and "Bandwidth" in msg
    ok, _ = speedup_verdict(1000, 500, 1.5, bandwidth_gbps=30.0,
                            min_bandwidth=20.0)
    assert ok


def test_format_time_info():
    assert format_time_info(1000) == "1000us"
    s = format_time_info(1000, bytes_moved=4_000_000)
    assert s.startswith("1000us (") and "GBytes/s" in s


def test_verdict_line_grammar():
    line = verdict_line("in_order", ["C", "MD"], "SUCCESS: x")
    assert line == "## in_order | C MD | SUCCESS: x"


def test_run_bench_rejects_bad_copy_engine():
    from hpc_patterns_amd.concurrency import run_bench

    with pytest.raises(ValueError, match="copy_engine"):
        run_bench("serial", ["C"], copy_engine="warp_drive")


def test_run_bench_rejects_bad_mode_before_native():
    from hpc_patterns_amd.concurrency import run_bench

    with pytest.raises(ValueError, match="mode"):
        run_bench("hyperspeed", ["C"])
