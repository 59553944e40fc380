"""PhaseTimer / CycleMetrics unit tests (SURVEY §5 observability).

The reference had no metrics at all [R: main.py — log lines only];
these cover the per-phase timers and counters the loop exports."""

import time

from mi355x_autoscaler.utils.metrics import CycleMetrics, PhaseTimer


def test_phase_timer_accumulates_and_cycles():
    t = PhaseTimer()
    with t.phase("ingest"):
        time.sleep(0.002)
    with t.phase("decide"):
        time.sleep(0.001)
    with t.phase("decide"):  # same phase twice in one cycle accumulates
        time.sleep(0.001)
    snap = t.end_cycle()
    assert snap["ingest"] >= 0.002
    assert snap["decide"] >= 0.002
    assert abs(snap["total"] - (snap["ingest"] + snap["decide"])) < 1e-9
    assert t.current == {}
    assert len(t.history) == 1


def test_phase_timer_records_on_exception():
    t = PhaseTimer()
    try:
        with t.phase("fit"):
            raise RuntimeError("boom")
    except RuntimeError:
        pass
    assert t.current["fit"] >= 0.0
    snap = t.end_cycle()
    assert "fit" in snap


def test_phase_timer_percentiles():
    t = PhaseTimer()
    for ms in (1, 2, 3, 4, 100):
        t.current = {"decide": ms / 1000.0}
        t.end_cycle()
    p50 = t.percentile("decide", 0.5)
    p99 = t.percentile("decide", 0.99)
    assert p50 is not None and 0.002 <= p50 <= 0.004
    assert p99 == 0.1
    assert t.percentile("missing_phase", 0.5) == 0.0  # absent == 0 rows
    assert PhaseTimer().percentile("decide", 0.5) is None


def test_cycle_metrics_counters_without_prometheus():
    m = CycleMetrics(prometheus_port=None)  # no exporter started
    m.count("cycles")
    m.count("pods_scheduled", 7)
    m.count("pods_scheduled", 3)
    m.gauge("pods_pending", 42)
    m.observe_latency(0.25)
    assert m.counters["cycles"] == 1
    assert m.counters["pods_scheduled"] == 10
    assert m.counters["pods_pending"] == 42
    assert m.counters["last_decision_latency_s"] == 0.25


def test_main_module_entrypoint_imports():
    # ``python -m mi355x_autoscaler`` wiring (reference: python main.py)
    import mi355x_autoscaler.__main__ as entry

    assert callable(entry.main)
