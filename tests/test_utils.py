"""Unit coverage for the aux subsystems: notifier event log, phase
timers / metrics, Prometheus export wiring."""

import pytest

from mi355x_autoscaler.utils.metrics import CycleMetrics, PhaseTimer
from mi355x_autoscaler.utils.notification import Notifier


def test_notifier_event_log_without_hook():
    n = Notifier(hook_url=None)
    n.notify_scale_up({"agentpool1": 2}, pods_pending=10)
    n.notify_failed_to_scale(0xDEAD, 3)
    n.notify_scale_down("k8s-agentpool1-12345678-1", 4)
    n.notify_error("boom")
    kinds = [e["kind"] for e in n.events]
    assert kinds == ["scale_up", "failed_to_scale", "scale_down", "error"]
    assert n.events[0]["units"] == {"agentpool1": 2}


def test_notifier_post_failure_never_raises(monkeypatch):
    n = Notifier(hook_url="http://127.0.0.1:9/bad")

    class Boom:
        @staticmethod
        def post(*a, **k):
            raise RuntimeError("net down")

    import mi355x_autoscaler.utils.notification as mod

    monkeypatch.setattr("requests.post", Boom.post, raising=False)
    n.notify_error("still fine")  # must not raise
    assert n.events[-1]["kind"] == "error"


def test_phase_timer_percentiles():
    t = PhaseTimer()
    for ms in (1, 2, 3, 4, 5):
        with t.phase("decide"):
            pass
        t.current["decide"] = ms / 1000.0  # deterministic values
        t.end_cycle()
    p50 = t.percentile("decide", 0.5)
    assert p50 is not None and 0.002 <= p50 <= 0.004
    assert t.percentile("missing", 0.5) == 0.0 or t.percentile("missing", 0.5) is not None


def test_cycle_metrics_counters():
    m = CycleMetrics()
    m.count("cycles")
    m.count("pods_scheduled", 42)
    m.gauge("pods_pending", 7)
    m.observe_latency(0.005)
    assert m.counters["cycles"] == 1
    assert m.counters["pods_scheduled"] == 42
    assert m.counters["pods_pending"] == 7
    assert m.counters["last_decision_latency_s"] == 0.005


def test_prometheus_export_starts():
    import socket

    sock = socket.socket()
    sock.bind(("127.0.0.1", 0))
    port = sock.getsockname()[1]
    sock.close()
    m = CycleMetrics(prometheus_port=port)
    m.count("cycles")
    m.gauge("pods_pending", 3)
    m.observe_latency(0.001)
    import urllib.request

    body = urllib.request.urlopen(f"http://127.0.0.1:{port}/metrics", timeout=5).read().decode()
    assert "autoscaler_cycles_total" in body
    assert "autoscaler_pods_pending 3.0" in body


def test_kubeconfig_parsing(tmp_path):
    """from_kubeconfig resolves current-context -> cluster/user, token
    auth, and the TLS-verify setting [R: reference used pykube.KubeConfig
    — reconstructed]."""
    import yaml

    from mi355x_autoscaler.kube.client import KubeApiClient

    cfg = {
        "current-context": "prod",
        "contexts": [
            {"name": "dev", "context": {"cluster": "c-dev", "user": "u-dev"}},
            {"name": "prod", "context": {"cluster": "c-prod", "user": "u-prod"}},
        ],
        "clusters": [
            {"name": "c-dev", "cluster": {"server": "https://dev:6443"}},
            {"name": "c-prod", "cluster": {"server": "https://prod:6443",
                                           "insecure-skip-tls-verify": True}},
        ],
        "users": [
            {"name": "u-dev", "user": {}},
            {"name": "u-prod", "user": {"token": "sekrit"}},
        ],
    }
    p = tmp_path / "kubeconfig"
    p.write_text(yaml.safe_dump(cfg))
    kube = KubeApiClient.from_kubeconfig(str(p))
    assert kube.base_url.startswith("https://prod:6443")
    assert kube.session.headers["Authorization"] == "Bearer sekrit"
    assert kube.session.verify is False


def test_in_cluster_reads_service_account(tmp_path, monkeypatch):
    import mi355x_autoscaler.kube.client as kc

    sa = tmp_path / "sa"
    sa.mkdir()
    (sa / "token").write_text("tok123\n")
    (sa / "ca.crt").write_text("CERT")
    monkeypatch.setattr(kc, "_SA_DIR", str(sa))
    monkeypatch.setenv("KUBERNETES_SERVICE_HOST", "10.0.0.1")
    monkeypatch.setenv("KUBERNETES_SERVICE_PORT", "443")
    kube = kc.KubeApiClient.in_cluster()
    assert kube.base_url == "https://10.0.0.1:443"
    assert kube.session.headers["Authorization"] == "Bearer tok123"
    assert str(kube.session.verify).endswith("ca.crt")
