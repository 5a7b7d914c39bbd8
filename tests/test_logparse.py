"""Log-grammar parser tests (reference parse.py behavior)."""

from hpc_patterns_amd.utils.logparse import parse_log, render_table

SAMPLE = """\
export HIP_VISIBLE_DEVICES=0
# in_order | C MD | Starting Benchmarking...
Minimum Measured Total Time Serial: 1000us
## in_order | C MD | SUCCESS: Close from Theoretical Speedup
## in_order | C DM | FAILURE: Far from Theoretical Speedup
export GPU_MAX_HW_QUEUES=8
## graph | C MD | SUCCESS: Close from Theoretical Speedup
"""


def test_parse_groups_by_env():
    parsed = parse_log(SAMPLE.splitlines())
    assert parsed["HIP_VISIBLE_DEVICES=0"]["C MD"]["in_order"] == "SUCCESS"
    assert parsed["HIP_VISIBLE_DEVICES=0"]["C DM"]["in_order"] == "FAILURE"
    assert parsed["GPU_MAX_HW_QUEUES=8"]["C MD"]["graph"] == "SUCCESS"


def test_parse_ignores_non_verdict_lines():
    parsed = parse_log(SAMPLE.splitlines())
    assert len(parsed["HIP_VISIBLE_DEVICES=0"]) == 2


def test_render_table_contains_cells():
    parsed = parse_log(SAMPLE.splitlines())
    table = render_table(parsed)
    assert "C MD" in table and "SUCCESS" in table and "FAILURE" in table


def test_reference_style_log_lines():
    # the reference's own verdict format parses identically
    lines = [
        "export ZE_AFFINITY_MASK=0.0",
        "## out_of_order | C C | SUCCESS: Close from Theoretical Speedup",
    ]
    parsed = parse_log(lines)
    assert parsed["ZE_AFFINITY_MASK=0.0"]["C C"]["out_of_order"] == "SUCCESS"


def test_parse_tolerates_malformed_verdict_lines():
    lines = [
        "export X=1",
        "## only-one-field SUCCESS",       # no | separator -> skipped
        "## a | b | SUCCESS: ok",
    ]
    parsed = parse_log(lines)
    assert parsed["X=1"] == {"b": {"a": "SUCCESS"}}
