"""CPU tests for the per-link xGMI counter helpers (utils/xgmi.py)."""

from hpc_patterns_amd.utils.xgmi import delta, flatten_counters

CANNED = {
    "gpu": [
        {"bdf": "0000:0c:00.0",
         "xgmi": {"link_0": {"read_kb": 1000, "write_kb": 2000},
                  "link_1": {"read_kb": 0, "write_kb": 0}}},
        {"bdf": "0000:22:00.0",
         "xgmi": {"link_0": {"read_kb": 500, "write_kb": 700}}},
    ],
    "meta": {"version": "26.2"},
}


def test_flatten_collects_traffic_leaves():
    flat = flatten_counters(CANNED)
    assert flat["gpu[0].xgmi.link_0.read_kb"] == 1000.0
    assert flat["gpu[1].xgmi.link_0.write_kb"] == 700.0
    # non-traffic leaves (version strings, bdf) are not collected
    assert all("version" not in k and "bdf" not in k for k in flat)


def test_delta_filters_idle_links():
    before = flatten_counters(CANNED)
    import copy

    after_raw = copy.deepcopy(CANNED)
    after_raw["gpu"][0]["xgmi"]["link_0"]["read_kb"] += 4096
    d = delta(before, flatten_counters(after_raw))
    assert d == {"gpu[0].xgmi.link_0.read_kb": 4096.0}


def test_delta_handles_none():
    assert delta(None, {"a.read": 1.0}) == {}
    assert delta({"a.read": 1.0}, None) == {}
