"""Optional TOML config file (SURVEY §5 config plan): identical
semantics to the CLI flags, validated with pydantic; explicit
command-line flags always win over file values.

    [autoscaler]
    resource_group = "my-rg"
    sleep = 10
    spare_agents = 1
    idle_threshold = 1800
    backend = "gpu"
    cloud = "azure"
"""

from __future__ import annotations

from typing import Optional

from pydantic import BaseModel, Field, field_validator


class AutoscalerConfig(BaseModel):
    resource_group: str = ""
    service_principal_app_id: Optional[str] = None
    service_principal_secret: Optional[str] = None
    service_principal_tenant_id: Optional[str] = None
    kubeconfig: Optional[str] = None
    sleep: int = Field(10, ge=0)
    spare_agents: int = Field(1, ge=0)
    idle_threshold: int = Field(1800, ge=0)
    over_provision: int = Field(0, ge=0)
    ignore_pools: str = ""
    no_scale: bool = False
    no_maintenance: bool = False
    slack_hook: Optional[str] = None
    dry_run: bool = False
    verbose: bool = False
    debug: bool = False
    backend: str = "cpu"
    gpus: int = Field(1, ge=1, le=8)
    cloud: str = "azure"
    kube_api: Optional[str] = None
    utilization_threshold: int = Field(300, ge=0, le=1000)
    max_pool_size: int = Field(100, ge=1)
    prometheus_port: Optional[int] = None
    watch: bool = False

    @field_validator("backend")
    @classmethod
    def _backend(cls, v):
        if v not in ("cpu", "gpu"):
            raise ValueError("backend must be cpu|gpu")
        return v

    @field_validator("cloud")
    @classmethod
    def _cloud(cls, v):
        if v not in ("fake", "azure"):
            raise ValueError("cloud must be fake|azure")
        return v


def load_config(path: str) -> AutoscalerConfig:
    import tomli

    with open(path, "rb") as f:
        data = tomli.load(f)
    section = data.get("autoscaler", data)
    return AutoscalerConfig(**section)


def apply_config_defaults(ctx, config: AutoscalerConfig) -> dict:
    """Merge: CLI-explicit params win; file fills the rest. Returns the
    resolved kwargs dict for main()'s body."""
    import click

    resolved = {}
    for name, value in ctx.params.items():
        if name == "config":
            continue
        src = ctx.get_parameter_source(name)
        if src == click.core.ParameterSource.COMMANDLINE or not hasattr(config, name):
            resolved[name] = value
        else:
            resolved[name] = getattr(config, name)
    return resolved
