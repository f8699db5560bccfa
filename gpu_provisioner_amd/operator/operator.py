"""Operator: builds Azure config + clients + the instance provider.

Spec: reference pkg/operator/operator.go:30-68 — wraps the runtime manager
with the Azure wiring, panicking with a federated-credential hint when auth
is misconfigured, exactly the failure operators actually hit when the
workload-identity webhook or federated credential is missing.
"""
from __future__ import annotations

import logging
import os

from ..auth.config import AzureConfig, ConfigError, build_azure_config
from ..auth.cred import new_credential
from ..cloudprovider.azure import AzureCloudProvider
from ..kube.client import KubeClient
from ..providers.instance.armclient import ARMAgentPoolsClient
from ..providers.instance.provider import InstanceProvider
from ..providers.instancetype.catalog import InstanceTypeProvider

log = logging.getLogger(__name__)

FEDERATED_CREDENTIAL_HINT = (
    "failed to build Azure config/credential. In self-hosted mode this "
    "usually means the federated identity credential is not configured for "
    "the controller's service account: create one with `az identity "
    "federated-credential create --identity-name <id> --issuer <oidc-issuer> "
    "--subject system:serviceaccount:<ns>:gpu-provisioner-amd` and label the "
    "pod with azure.workload.identity/use: 'true'"
)


class Operator:
    def __init__(self, kube: KubeClient, environ=None):
        try:
            self.config: AzureConfig = build_azure_config(environ)
        except ConfigError as e:
            raise SystemExit(f"{e}\n{FEDERATED_CREDENTIAL_HINT}") from e
        self.credential = new_credential(self.config)
        # E2E pipelines stamp scenario headers on every ARM request
        # (reference azure_client.go:113-141 injects them via a pipeline
        # policy when E2E_TEST_MODE is set)
        env = environ if environ is not None else os.environ
        extra_headers = {}
        if env.get("E2E_TEST_MODE", "").lower() == "true":
            extra_headers["X-Kaito-E2E"] = env.get("E2E_SCENARIO", "true")
        # pinned ARM api-version + request schema travel together
        # (armschema.py; ARM_API_PROFILE env selects stable/gpu-preview)
        from ..providers.instance.armschema import profile_from_env

        arm_profile = profile_from_env(environ)
        self.agent_pools = ARMAgentPoolsClient(
            self.credential,
            self.config.subscription_id,
            endpoint=self.config.arm_endpoint,
            user_agent=self.config.user_agent,
            extra_headers=extra_headers,
            api_version=arm_profile.api_version,
        )
        self.catalog = InstanceTypeProvider(region=self.config.location)
        self.instances = InstanceProvider(
            self.agent_pools,
            kube,
            self.catalog,
            resource_group=self.config.resource_group,
            cluster_name=self.config.cluster_name,
            arm_profile=arm_profile,
        )
        self.cloud_provider = AzureCloudProvider(self.instances, self.catalog)
        log.info(
            "operator: cluster=%s rg=%s location=%s mode=%s",
            self.config.cluster_name,
            self.config.resource_group,
            self.config.location,
            self.config.deployment_mode,
        )
