"""CLI end-to-end: the flag-compatible entrypoint drives a full cycle
against the in-process fake API + fake cloud."""

from click.testing import CliRunner

from mi355x_autoscaler.cli import main
from mi355x_autoscaler.kube.fake_api import FakeKubeServer
from mi355x_autoscaler.sim.synth import burst_trace


def test_cli_one_cycle_fake_backends():
    with FakeKubeServer() as srv:
        pods, nodes = burst_trace(20, n_nodes_per_pool=(3,), seed=0)
        srv.state.load(pods, nodes)
        runner = CliRunner()
        result = runner.invoke(main, [
            "--kube-api", srv.base_url, "--cloud", "fake", "--backend", "cpu",
            "--sleep", "0", "--idle-threshold", "0", "--no-maintenance",
            "--max-cycles", "1", "-v",
        ])
        assert result.exit_code == 0, result.output


def test_cli_reference_flags_accepted():
    runner = CliRunner()
    result = runner.invoke(main, ["--help"])
    assert result.exit_code == 0
    for flag in ("--resource-group", "--service-principal-app-id", "--kubeconfig",
                 "--sleep", "--spare-agents", "--idle-threshold", "--over-provision",
                 "--ignore-pools", "--no-scale", "--no-maintenance", "--slack-hook",
                 "--dry-run", "--verbose", "--debug"):
        assert flag in result.output, flag


def test_toml_config_file(tmp_path):
    """TOML file supplies defaults; explicit CLI flags win."""
    cfg = tmp_path / "autoscaler.toml"
    cfg.write_text(
        "[autoscaler]\nsleep = 42\nspare_agents = 3\ncloud = \"fake\"\n"
        "no_scale = true\nno_maintenance = true\ndry_run = true\n"
    )
    with FakeKubeServer() as srv:
        pods, nodes = burst_trace(5, n_nodes_per_pool=(2,), seed=1)
        srv.state.load(pods, nodes)
        runner = CliRunner()
        result = runner.invoke(main, [
            "--config", str(cfg), "--kube-api", srv.base_url,
            "--max-cycles", "1", "--sleep", "0",  # CLI overrides file's 42
        ])
        assert result.exit_code == 0, result.output


def test_toml_config_validation(tmp_path):
    from mi355x_autoscaler.utils.config import load_config

    bad = tmp_path / "bad.toml"
    bad.write_text("[autoscaler]\nbackend = \"tpu\"\n")
    import pytest as _pytest

    with _pytest.raises(Exception):
        load_config(str(bad))
    good = tmp_path / "good.toml"
    good.write_text("[autoscaler]\nbackend = \"gpu\"\ngpus = 8\nidle_threshold = 600\n")
    cfg = load_config(str(good))
    assert cfg.backend == "gpu" and cfg.gpus == 8 and cfg.idle_threshold == 600
