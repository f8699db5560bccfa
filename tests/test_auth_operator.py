"""Auth config matrix (reference pkg/auth/config_test.go), credential flows
(mocked AAD/IMDS), options/feature gates, logging format, leader election."""
import asyncio
import json
import logging
import time

import httpx
import pytest

from gpu_provisioner_amd.auth.config import (
    ConfigError,
    build_azure_config,
)
from gpu_provisioner_amd.auth.cred import (
    CredentialError,
    ManagedIdentityCredential,
    WorkloadIdentityCredential,
)
from gpu_provisioner_amd.fake.apiserver import InMemoryAPIServer, InMemoryClient
from gpu_provisioner_amd.operator.leaderelection import LeaderElector
from gpu_provisioner_amd.operator.logging import JSONFormatter
from gpu_provisioner_amd.operator.options import FeatureGates, Options
from tests.conftest import run

GOOD_ENV = {
    "AZURE_TENANT_ID": "tenant",
    "ARM_SUBSCRIPTION_ID": "sub",
    "ARM_RESOURCE_GROUP": "rg",
    "LOCATION": "eastus2",
    "AZURE_CLUSTER_NAME": "cluster",
    "AZURE_CLIENT_ID": "client",
}


# ------------------------------------------------------------------- config


def test_config_happy_path():
    cfg = build_azure_config(GOOD_ENV)
    assert cfg.tenant_id == "tenant"
    assert cfg.deployment_mode == "self-hosted"
    assert cfg.node_resource_group == "MC_rg_cluster_eastus2"


@pytest.mark.parametrize("missing", sorted(set(GOOD_ENV) - {"AZURE_CLIENT_ID"}))
def test_config_missing_required(missing):
    env = {k: v for k, v in GOOD_ENV.items() if k != missing}
    with pytest.raises(ConfigError):
        build_azure_config(env)


def test_config_self_hosted_requires_client_id():
    env = {k: v for k, v in GOOD_ENV.items() if k != "AZURE_CLIENT_ID"}
    env["DEPLOYMENT_MODE"] = "self-hosted"
    with pytest.raises(ConfigError):
        build_azure_config(env)
    env["DEPLOYMENT_MODE"] = "managed"
    assert build_azure_config(env).deployment_mode == "managed"


def test_config_invalid_mode():
    env = dict(GOOD_ENV, DEPLOYMENT_MODE="cloudy")
    with pytest.raises(ConfigError):
        build_azure_config(env)


# -------------------------------------------------------------- credentials


def test_workload_identity_token_exchange(tmp_path):
    token_file = tmp_path / "token"
    token_file.write_text("jwt-assertion")
    cfg = build_azure_config(dict(GOOD_ENV, AZURE_FEDERATED_TOKEN_FILE=str(token_file)))
    calls = {"n": 0}

    def handler(request: httpx.Request) -> httpx.Response:
        calls["n"] += 1
        body = dict(p.split("=", 1) for p in request.content.decode().split("&"))
        assert body["client_id"] == "client"
        assert "jwt-assertion" in body["client_assertion"]
        # RFC 7523 client-assertion-type urn, URL-encoded (AAD rejects the
        # jwt-bearer GRANT urn here)
        assert body["client_assertion_type"] == (
            "urn%3Aietf%3Aparams%3Aoauth%3Aclient-assertion-type%3Ajwt-bearer"
        )
        assert body["grant_type"] == "client_credentials"
        assert "tenant" in str(request.url)
        return httpx.Response(200, json={"access_token": "aad-token", "expires_in": 3600})

    async def main():
        cred = WorkloadIdentityCredential(
            cfg, httpx.AsyncClient(transport=httpx.MockTransport(handler))
        )
        assert await cred.get_token() == "aad-token"
        # cached: no second AAD round-trip
        assert await cred.get_token() == "aad-token"
        assert calls["n"] == 1

    run(main())


def test_workload_identity_missing_token_file():
    cfg = build_azure_config(dict(GOOD_ENV, AZURE_FEDERATED_TOKEN_FILE="/nonexistent/token"))

    async def main():
        cred = WorkloadIdentityCredential(cfg)
        with pytest.raises(CredentialError, match="federated token file"):
            await cred.get_token()

    run(main())


def test_managed_identity_imds():
    cfg = build_azure_config(dict(GOOD_ENV, DEPLOYMENT_MODE="managed"))

    def handler(request: httpx.Request) -> httpx.Response:
        assert request.headers["Metadata"] == "true"
        return httpx.Response(
            200, json={"access_token": "imds-token", "expires_on": time.time() + 3600}
        )

    async def main():
        cred = ManagedIdentityCredential(
            cfg, httpx.AsyncClient(transport=httpx.MockTransport(handler))
        )
        assert await cred.get_token() == "imds-token"

    run(main())


# ------------------------------------------------------------------ options


def test_options_defaults_and_env(monkeypatch):
    opts = Options.from_env_and_args(argv=[], environ={})
    assert opts.metrics_port == 8080
    assert opts.health_probe_port == 8081
    assert opts.kube_client_qps == 200.0
    assert opts.leader_elect is False
    assert opts.feature_gates.node_repair is True
    opts = Options.from_env_and_args(
        argv=[],
        environ={
            "METRICS_PORT": "9090",
            "FEATURE_GATES": "NodeRepair=false",
            "LEADER_ELECT": "true",
        },
    )
    assert opts.metrics_port == 9090
    assert opts.feature_gates.node_repair is False
    assert opts.leader_elect is True


def test_feature_gates_parse():
    assert FeatureGates.parse("NodeRepair=true").node_repair
    assert not FeatureGates.parse("NodeRepair=false").node_repair
    assert FeatureGates.parse("Other=false").node_repair  # default preserved
    assert FeatureGates.parse("").node_repair


# ------------------------------------------------------------------ logging


def test_json_log_format():
    rec = logging.LogRecord("x", logging.INFO, "f.py", 1, "hello %s", ("world",), None)
    entry = json.loads(JSONFormatter().format(rec))
    assert entry["level"] == "info"
    assert entry["message"] == "hello world"
    assert entry["logger"] == "x"
    assert "ts" in entry


# ---------------------------------------------------------- leader election


def test_leader_election_single_holder_and_failover():
    async def main():
        server = InMemoryAPIServer()
        c1, c2 = InMemoryClient(server), InMemoryClient(server)
        e1 = LeaderElector(c1, "lock", "kube-system", identity="a",
                           lease_duration=1.0, renew_interval=0.1)
        e2 = LeaderElector(c2, "lock", "kube-system", identity="b",
                           lease_duration=1.0, renew_interval=0.1)
        assert await e1._try_acquire() is True
        assert await e2._try_acquire() is False  # held and fresh
        assert await e1._renew() == "ok"
        # holder dies; after lease_duration the lock is stealable
        await asyncio.sleep(1.2)
        assert await e2._try_acquire() is True
        assert await e1._renew() == "lost"  # a lost the lease

    run(main())
