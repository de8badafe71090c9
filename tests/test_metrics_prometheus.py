"""Prometheus exposition format + reference metric-name parity
(reference pkg/scheduler/metrics/metrics.go:57-191)."""

import math
import re

from volcano_amd.utils.metrics import (MetricsRegistry,
                                       exponential_buckets_range)

# the subset of the text exposition grammar promtool enforces
_SERIES = re.compile(
    r'^[a-zA-Z_:][a-zA-Z0-9_:]*(\{[a-zA-Z_][a-zA-Z0-9_]*="[^"\n]*"'
    r'(,[a-zA-Z_][a-zA-Z0-9_]*="[^"\n]*")*\})? [0-9eE+.\-]+$|'
    r'^[a-zA-Z_:][a-zA-Z0-9_:]*\{.*le="\+Inf".*\} [0-9]+$')


def _mk() -> MetricsRegistry:
    m = MetricsRegistry()
    m.observe("e2e_scheduling_latency", 0.120)
    m.observe("e2e_scheduling_latency", 0.450)
    m.observe("open_session_duration", 0.030)
    m.observe("action_scheduling_latency:allocate", 0.200)
    m.observe("action_scheduling_latency:enqueue", 0.004)
    m.observe("plugin_scheduling_latency:gang:OnSessionOpen", 0.0002)
    m.inc("schedule_attempts_total:scheduled", 5)
    m.inc("total_preemption_attempts")
    m.load_rocprof_stats({"vamd::score_cap_kernel": 7500.0})
    return m


def test_exposition_parses():
    text = _mk().export_prometheus()
    lines = [ln for ln in text.strip().splitlines()]
    assert lines
    for ln in lines:
        if ln.startswith("#"):
            assert re.match(r"^# (HELP|TYPE) [a-zA-Z_:][a-zA-Z0-9_:]*", ln), ln
        else:
            assert _SERIES.match(ln), f"unparseable series line: {ln!r}"


def test_reference_names_and_buckets():
    text = _mk().export_prometheus()
    for name in ("volcano_e2e_scheduling_latency_milliseconds",
                 "volcano_open_session_duration_milliseconds",
                 "volcano_action_scheduling_latency_milliseconds",
                 "volcano_plugin_scheduling_latency_milliseconds",
                 "volcano_schedule_attempts_total",
                 "volcano_total_preemption_attempts"):
        assert name in text, f"missing reference metric {name}"
    # labels
    assert 'action="allocate"' in text
    assert 'plugin="gang"' in text and 'OnSession="OnSessionOpen"' in text
    assert 'result="scheduled"' in text
    # histogram anatomy: cumulative buckets ending at +Inf == count
    assert 'le="+Inf"' in text
    m = re.search(r'volcano_e2e_scheduling_latency_milliseconds_bucket'
                  r'\{le="\+Inf"\} (\d+)', text)
    c = re.search(r"volcano_e2e_scheduling_latency_milliseconds_count (\d+)",
                  text)
    assert m and c and m.group(1) == c.group(1) == "2"
    # kernel time gauges (rocprof fold-in)
    assert 'volcano_kernel_time_microseconds{kernel="vamd::score_cap_kernel"}' \
        in text


def test_exponential_buckets_range():
    b = exponential_buckets_range(1, 5000, 20)
    assert len(b) == 20
    assert abs(b[0] - 1) < 1e-9 and abs(b[-1] - 5000) < 1e-6
    assert all(b[i] < b[i + 1] for i in range(19))


def test_histogram_cumulative_counts():
    m = MetricsRegistry()
    for v in (0.001, 0.010, 0.100, 1.0, 10.0):
        m.observe("e2e_scheduling_latency", v)
    text = m.export_prometheus()
    counts = [int(x) for x in re.findall(
        r'volcano_e2e_scheduling_latency_milliseconds_bucket\{le="[^"]*"\} '
        r'(\d+)', text)]
    assert counts == sorted(counts)          # cumulative
    assert counts[-1] == 5
