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
