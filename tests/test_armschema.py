"""Pinned ARM agentPools schema: builder contract + recorded-fixture tests.

VERDICT r01 #3: every agent-pool PUT body must validate against the pinned
api-version's schema; speculative gpuProfile fields live behind the
explicitly-selected gpu-preview profile; recorded fixtures exercise real ARM
wire shapes (error bodies, header casing, api-version query)."""

import httpx
import pytest

from gpu_provisioner_amd.auth.cred import StaticCredential
from gpu_provisioner_amd.providers.instance import bootstrap
from gpu_provisioner_amd.providers.instance.armclient import ARMAgentPoolsClient
from gpu_provisioner_amd.providers.instance.armschema import (
    PROFILE_GPU_PREVIEW,
    PROFILE_STABLE,
    SchemaViolation,
    profile_from_env,
    validate_agent_pool,
)
from tests.conftest import run
from tests.test_instance_provider import make_provider, nodeclaim, VM


# ------------------------------------------------------------------ schema


def test_builder_output_validates_on_stable_profile():
    provider, *_ = make_provider()
    pool = provider.new_agent_pool_object(nodeclaim(), VM)  # validates internally
    validate_agent_pool(pool, PROFILE_STABLE)
    assert "gpuProfile" not in pool["properties"]


def test_builder_emits_gpu_profile_only_under_preview_profile():
    provider, pools, kube, aks = make_provider()
    provider.arm_profile = PROFILE_GPU_PREVIEW
    pool = provider.new_agent_pool_object(nodeclaim(), VM)
    gp = pool["properties"]["gpuProfile"]
    assert gp["driver"] == "Install"
    assert gp["driverType"] == "ROCm"
    assert set(gp) <= PROFILE_GPU_PREVIEW.gpu_profile_fields
    validate_agent_pool(pool, PROFILE_GPU_PREVIEW)
    # the same body is INVALID under the stable pin
    with pytest.raises(SchemaViolation, match="gpuProfile"):
        validate_agent_pool(pool, PROFILE_STABLE)


def test_unknown_property_rejected():
    with pytest.raises(SchemaViolation, match="rocmInstall"):
        validate_agent_pool(
            {"name": "x", "properties": {"count": 1, "vmSize": "v", "rocmInstall": True}}
        )


def test_enum_violation_rejected():
    with pytest.raises(SchemaViolation, match="scaleSetPriority"):
        validate_agent_pool(
            {"name": "x", "properties": {"scaleSetPriority": "Cheap"}}
        )


def test_unknown_kubelet_and_sysctl_fields_rejected():
    with pytest.raises(SchemaViolation, match="kubeletConfig"):
        validate_agent_pool(
            {"name": "x", "properties": {"kubeletConfig": {"gpuPinning": True}}}
        )
    with pytest.raises(SchemaViolation, match="sysctls"):
        validate_agent_pool(
            {
                "name": "x",
                "properties": {"linuxOSConfig": {"sysctls": {"vm.max_map_count": 1}}},
            }
        )


def test_builder_schema_regression_fails_loudly():
    """If the bootstrap ever grows a field the pinned schema doesn't define,
    the builder itself must raise — that is the 'a test fails' guarantee."""
    provider, *_ = make_provider()
    orig = bootstrap.rocm_kubelet_config
    bootstrap.rocm_kubelet_config = lambda: {"podMaxPids": -1, "gpuPinning": "strict"}
    try:
        with pytest.raises(SchemaViolation):
            provider.new_agent_pool_object(nodeclaim(), VM)
    finally:
        bootstrap.rocm_kubelet_config = orig


def test_profile_from_env_selection():
    assert profile_from_env({}) is PROFILE_STABLE
    assert profile_from_env({"ARM_API_PROFILE": "gpu-preview"}) is PROFILE_GPU_PREVIEW
    assert profile_from_env({"ARM_API_PROFILE": "2024-09-01"}) is PROFILE_STABLE
    with pytest.raises(ValueError, match="ARM_API_PROFILE"):
        profile_from_env({"ARM_API_PROFILE": "bogus"})


def test_rocm_gpu_profile_none_on_stable():
    assert bootstrap.rocm_gpu_profile(PROFILE_STABLE) is None
    gp = bootstrap.rocm_gpu_profile(PROFILE_GPU_PREVIEW)
    assert gp is not None and gp["rocmVersion"] == bootstrap.ROCM_VERSION


# -------------------------------------------- recorded ARM wire fixtures


def make_client(handler, **kw) -> ARMAgentPoolsClient:
    return ARMAgentPoolsClient(
        StaticCredential("tok"),
        "sub",
        http=httpx.AsyncClient(transport=httpx.MockTransport(handler)),
        lro_poll_interval=0.01,
        **kw,
    )


def test_api_version_query_matches_pinned_profile():
    seen = {}

    def handler(request: httpx.Request) -> httpx.Response:
        seen["api-version"] = request.url.params.get("api-version")
        return httpx.Response(200, json={"name": "p", "properties": {}})

    async def main():
        client = make_client(handler)
        await client.get("rg", "c", "p")
        assert seen["api-version"] == PROFILE_STABLE.api_version
        client2 = make_client(handler, api_version=PROFILE_GPU_PREVIEW.api_version)
        await client2.get("rg", "c", "p")
        assert seen["api-version"] == PROFILE_GPU_PREVIEW.api_version

    run(main())


def test_recorded_arm_error_body_casing():
    """Recorded fixture: ARM CloudError body (code/message nested under
    'error', PascalCase code values) maps onto ARMError faithfully."""
    fixture = {
        "error": {
            "code": "InvalidParameter",
            "message": (
                "Provided parameter gpuProfile is not recognized for "
                "api-version 2024-09-01."
            ),
        }
    }

    def handler(request: httpx.Request) -> httpx.Response:
        return httpx.Response(400, json=fixture)

    async def main():
        client = make_client(handler)
        from gpu_provisioner_amd.providers.instance.armapi import ARMError

        with pytest.raises(ARMError) as ei:
            await client.get("rg", "c", "p")
        assert ei.value.status == 400
        assert ei.value.code == "InvalidParameter"
        assert "gpuProfile" in ei.value.message

    run(main())


def test_lro_header_casing_insensitive():
    """Recorded fixture: ARM emits 'Azure-AsyncOperation' but proxies may
    lowercase headers; the client must read them case-insensitively."""
    state = {"polls": 0}

    def handler(request: httpx.Request) -> httpx.Response:
        url = str(request.url)
        if request.method == "PUT":
            return httpx.Response(
                201,
                headers={"azure-asyncoperation": "https://arm/operations/op9"},
                json={"name": "p", "properties": {"provisioningState": "Creating"}},
            )
        if "/operations/op9" in url:
            state["polls"] += 1
            return httpx.Response(200, json={"status": "Succeeded"})
        return httpx.Response(
            200, json={"name": "p", "properties": {"provisioningState": "Succeeded"}}
        )

    async def main():
        client = make_client(handler)
        poller = await client.begin_create_or_update("rg", "c", "p", {"properties": {}})
        result = await poller.result()
        assert state["polls"] == 1
        assert result["properties"]["provisioningState"] == "Succeeded"

    run(main())


def test_recorded_lro_provisioning_state_payload():
    """Recorded fixture: some ARM operations report progress via the
    resource's properties.provisioningState instead of a status field."""
    state = {"polls": 0}

    def handler(request: httpx.Request) -> httpx.Response:
        url = str(request.url)
        if request.method == "PUT":
            return httpx.Response(
                201,
                headers={"Location": "https://arm/operationresults/r1"},
                json={"name": "p", "properties": {"provisioningState": "Creating"}},
            )
        if "/operationresults/r1" in url:
            state["polls"] += 1
            st = "Succeeded" if state["polls"] >= 2 else "Creating"
            return httpx.Response(
                200, json={"name": "p", "properties": {"provisioningState": st}}
            )
        return httpx.Response(
            200, json={"name": "p", "properties": {"provisioningState": "Succeeded"}}
        )

    async def main():
        client = make_client(handler)
        poller = await client.begin_create_or_update("rg", "c", "p", {"properties": {}})
        result = await poller.result()
        assert result["properties"]["provisioningState"] == "Succeeded"

    run(main())
