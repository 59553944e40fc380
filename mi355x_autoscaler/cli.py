"""CLI entrypoint — flag-compatible with the reference's ``main.py``.

[R: main.py · click command — reconstructed, unverified]: same flag
names and env-var fallbacks for everything that still has meaning
(resource-group, service-principal credentials, kubeconfig, sleep,
spare-agents, idle-threshold, over-provision, ignore-pools, no-scale,
no-maintenance, slack-hook, dry-run, verbose, debug), plus the
MI355X-native additions namespaced as new flags (--backend, --gpus,
--kube-api, --prometheus-port, --utilization-threshold).
"""

from __future__ import annotations

import logging

import click

from .cloud.azure_stub import AzureEngineProvider
from .cloud.fake import FakeCloudProvider
from .ctl.loop import Cluster, ClusterOptions
from .kube.client import KubeApiClient
from .sim.engine import make_engine
from .utils.metrics import CycleMetrics
from .utils.notification import Notifier

logger = logging.getLogger(__name__)


@click.command()
# -- reference flag surface [R: main.py — reconstructed] ---------------------
@click.option("--resource-group", envvar="AZURE_RESOURCE_GROUP", default="", help="Azure resource group of the ACS cluster")
@click.option("--service-principal-app-id", envvar="AZURE_SP_APP_ID", default=None)
@click.option("--service-principal-secret", envvar="AZURE_SP_SECRET", default=None)
@click.option("--service-principal-tenant-id", envvar="AZURE_SP_TENANT_ID", default=None)
@click.option("--kubeconfig", default=None, help="path to kubeconfig; omit for in-cluster config")
@click.option("--sleep", default=10, show_default=True, help="seconds between reconcile cycles")
@click.option("--spare-agents", default=1, show_default=True, help="idle nodes to keep as spares")
@click.option("--idle-threshold", default=1800, show_default=True, help="seconds a node must stay idle before scale-down")
@click.option("--over-provision", default=0, show_default=True, help="extra agents to add on every scale-up")
@click.option("--ignore-pools", default="", help="comma-separated pool names to leave alone")
@click.option("--no-scale", is_flag=True, help="disable scale-up")
@click.option("--no-maintenance", is_flag=True, help="disable scale-down")
@click.option("--slack-hook", envvar="SLACK_HOOK", default=None)
@click.option("--dry-run", is_flag=True)
@click.option("-v", "--verbose", is_flag=True)
@click.option("--debug", is_flag=True)
# -- MI355X-native additions -------------------------------------------------
@click.option("--backend", type=click.Choice(["cpu", "gpu"]), default="cpu", show_default=True,
              help="scheduling-simulation engine: CPU oracle or HIP kernels")
@click.option("--gpus", default=1, show_default=True, help="controller shard count (RCCL over xGMI); >1 requires torchrun launch")
@click.option("--cloud", type=click.Choice(["fake", "azure"]), default="azure", show_default=True,
              help="actuation backend (fake = in-process, for benchmarks/tests)")
@click.option("--kube-api", default=None, help="explicit kube API base URL (e.g. the in-process fake server)")
@click.option("--utilization-threshold", default=300, show_default=True,
              help="scale-down utilization bound, per-mille of the worst axis")
@click.option("--max-pool-size", default=100, show_default=True)
@click.option("--prometheus-port", default=None, type=int, help="expose /metrics on this port")
@click.option("--watch/--no-watch", default=False, show_default=True,
              help="delta ingest: one full relist, then pod watch events patch the cached snapshot")
@click.option("--config", "config_path", default=None,
              help="TOML config file; explicit CLI flags override file values")
@click.option("--max-cycles", default=None, type=int, hidden=True, help="stop after N cycles (tests)")
@click.pass_context
def main(ctx,
    **params,
):
    """MI355X-native Kubernetes GPU-node autoscaler for ACS-Engine pools."""
    config_path = params.pop("config_path", None)
    if config_path:
        from .utils.config import apply_config_defaults, load_config

        params = {k: v for k, v in apply_config_defaults(ctx, load_config(config_path)).items()}
    (resource_group, service_principal_app_id, service_principal_secret,
     service_principal_tenant_id, kubeconfig, sleep, spare_agents,
     idle_threshold, over_provision, ignore_pools, no_scale, no_maintenance,
     slack_hook, dry_run, verbose, debug, backend, gpus, cloud, kube_api,
     utilization_threshold, max_pool_size, prometheus_port, watch, max_cycles) = (
        params["resource_group"], params["service_principal_app_id"],
        params["service_principal_secret"], params["service_principal_tenant_id"],
        params["kubeconfig"], params["sleep"], params["spare_agents"],
        params["idle_threshold"], params["over_provision"], params["ignore_pools"],
        params["no_scale"], params["no_maintenance"], params["slack_hook"],
        params["dry_run"], params["verbose"], params["debug"], params["backend"],
        params["gpus"], params["cloud"], params["kube_api"],
        params["utilization_threshold"], params["max_pool_size"],
        params["prometheus_port"], params["watch"], params["max_cycles"])
    level = logging.DEBUG if debug else (logging.INFO if verbose else logging.WARNING)
    logging.basicConfig(level=level, format="%(asctime)s %(levelname)s %(name)s %(message)s")

    if kube_api:
        kube = KubeApiClient(kube_api)
    elif kubeconfig:
        kube = KubeApiClient.from_kubeconfig(kubeconfig)
    else:
        kube = KubeApiClient.in_cluster()

    if cloud == "fake":
        provider = FakeCloudProvider()
    else:
        provider = _azure_provider(
            resource_group,
            service_principal_app_id,
            service_principal_secret,
            service_principal_tenant_id,
        )

    engine = make_engine(backend)
    if gpus > 1:
        from .parallel.engine_sharded import ShardedEngine

        engine = ShardedEngine(engine)

    options = ClusterOptions(
        resource_group=resource_group,
        sleep_seconds=sleep,
        spare_agents=spare_agents,
        idle_threshold_seconds=idle_threshold,
        over_provision=over_provision,
        utilization_threshold_permille=utilization_threshold,
        ignore_pools=[p for p in ignore_pools.split(",") if p],
        max_pool_size=max_pool_size,
        no_scale=no_scale,
        no_maintenance=no_maintenance,
        dry_run=dry_run,
        watch=watch,
    )
    cluster = Cluster(
        kube,
        provider,
        engine,
        options=options,
        notifier=Notifier(slack_hook, dry_run=dry_run),
        metrics=CycleMetrics(prometheus_port=prometheus_port),
    )
    logger.warning(
        "autoscaler starting: backend=%s cloud=%s sleep=%ss dry_run=%s", backend, cloud, sleep, dry_run
    )
    from .parallel.elastic import RankLossError

    try:
        cluster.loop(max_cycles=max_cycles, debug=debug)
    except RankLossError as e:
        # distinct exit code so torchrun --max-restarts / a pod
        # restartPolicy relaunches the rank set (stateless by design:
        # the next incarnation re-lists the world)
        logger.error("exiting for supervisor relaunch after rank loss: %s", e)
        raise SystemExit(75)  # EX_TEMPFAIL


def _azure_provider(resource_group, app_id, secret, tenant):
    """Wire real azure-mgmt clients when available; otherwise the stub
    errors loudly on first actuation (benchmarks use --cloud fake)."""
    resource_client = compute_client = network_client = None
    try:  # pragma: no cover - azure SDK not installable offline
        from azure.common.credentials import ServicePrincipalCredentials  # type: ignore
        from azure.mgmt.compute import ComputeManagementClient  # type: ignore
        from azure.mgmt.network import NetworkManagementClient  # type: ignore
        from azure.mgmt.resource import ResourceManagementClient  # type: ignore

        creds = ServicePrincipalCredentials(client_id=app_id, secret=secret, tenant=tenant)
        sub = None  # subscription inferred from env in real deployments
        resource_client = ResourceManagementClient(creds, sub)
        compute_client = ComputeManagementClient(creds, sub)
        network_client = NetworkManagementClient(creds, sub)
    except ImportError:
        logger.warning("azure SDKs unavailable; actuation will fail loudly (use --cloud fake)")
    return AzureEngineProvider(
        resource_group,
        resource_client=resource_client,
        compute_client=compute_client,
        network_client=network_client,
    )


if __name__ == "__main__":
    main()
