"""Azure configuration from environment variables.

Spec: reference pkg/auth/config.go:75-137 — the same env-var surface
(`LOCATION`, `ARM_RESOURCE_GROUP`, `AZURE_TENANT_ID`, `AZURE_CLIENT_ID`,
`AZURE_CLUSTER_NAME`, `ARM_SUBSCRIPTION_ID`, `DEPLOYMENT_MODE`) validated at
boot, so the reference's Helm values drive this controller unchanged.
"""
from __future__ import annotations

import os
from dataclasses import dataclass

DEPLOYMENT_MODE_MANAGED = "managed"
DEPLOYMENT_MODE_SELF_HOSTED = "self-hosted"


class ConfigError(ValueError):
    pass


@dataclass
class AzureConfig:
    tenant_id: str = ""
    subscription_id: str = ""
    resource_group: str = ""
    location: str = ""
    cluster_name: str = ""
    client_id: str = ""
    deployment_mode: str = DEPLOYMENT_MODE_SELF_HOSTED
    # workload identity (self-hosted): AZURE_FEDERATED_TOKEN_FILE is projected
    # by the webhook; managed mode uses the node's managed identity via IMDS.
    federated_token_file: str = ""
    authority_host: str = "https://login.microsoftonline.com/"
    arm_endpoint: str = "https://management.azure.com"
    user_agent: str = "gpu-provisioner-amd/0.1.0"

    @property
    def node_resource_group(self) -> str:
        # AKS puts agent-pool infra in MC_<rg>_<cluster>_<location>
        return f"MC_{self.resource_group}_{self.cluster_name}_{self.location}"

    def validate(self) -> None:
        missing = [
            name
            for name, val in (
                ("AZURE_TENANT_ID", self.tenant_id),
                ("ARM_SUBSCRIPTION_ID", self.subscription_id),
                ("ARM_RESOURCE_GROUP", self.resource_group),
                ("LOCATION", self.location),
                ("AZURE_CLUSTER_NAME", self.cluster_name),
            )
            if not val
        ]
        if missing:
            raise ConfigError(f"missing required Azure config: {', '.join(missing)}")
        if self.deployment_mode not in (DEPLOYMENT_MODE_MANAGED, DEPLOYMENT_MODE_SELF_HOSTED):
            raise ConfigError(
                f"DEPLOYMENT_MODE must be {DEPLOYMENT_MODE_MANAGED!r} or "
                f"{DEPLOYMENT_MODE_SELF_HOSTED!r}, got {self.deployment_mode!r}"
            )
        if self.deployment_mode == DEPLOYMENT_MODE_SELF_HOSTED and not self.client_id:
            raise ConfigError("AZURE_CLIENT_ID is required in self-hosted deployment mode")


def build_azure_config(environ: dict = None) -> AzureConfig:
    """BuildAzureConfig (reference pkg/auth/config.go:87-106)."""
    env = environ if environ is not None else os.environ
    cfg = AzureConfig(
        tenant_id=env.get("AZURE_TENANT_ID", ""),
        subscription_id=env.get("ARM_SUBSCRIPTION_ID", ""),
        resource_group=env.get("ARM_RESOURCE_GROUP", ""),
        location=env.get("LOCATION", ""),
        cluster_name=env.get("AZURE_CLUSTER_NAME", ""),
        client_id=env.get("AZURE_CLIENT_ID", ""),
        deployment_mode=env.get("DEPLOYMENT_MODE", DEPLOYMENT_MODE_SELF_HOSTED),
        federated_token_file=env.get(
            "AZURE_FEDERATED_TOKEN_FILE", "/var/run/secrets/azure/tokens/azure-identity-token"
        ),
        authority_host=env.get("AZURE_AUTHORITY_HOST", "https://login.microsoftonline.com/"),
        arm_endpoint=env.get("ARM_ENDPOINT", "https://management.azure.com"),
    )
    cfg.validate()
    return cfg
