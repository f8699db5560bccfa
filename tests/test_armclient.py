"""ARM AgentPools client tests against a mocked ARM endpoint: retry policy,
LRO polling, pager, error mapping. Spec: reference pkg/utils/opts/ (retry
20×5s exp) + armutils.go LRO handling."""
import json

import httpx
import pytest

from gpu_provisioner_amd.auth.cred import StaticCredential
from gpu_provisioner_amd.providers.instance.armapi import ARMError
from gpu_provisioner_amd.providers.instance.armclient import ARMAgentPoolsClient
from tests.conftest import run


def make_client(handler, **kw) -> ARMAgentPoolsClient:
    return ARMAgentPoolsClient(
        StaticCredential("tok"),
        "sub",
        http=httpx.AsyncClient(transport=httpx.MockTransport(handler)),
        lro_poll_interval=0.01,
        **kw,
    )


def test_create_lro_flow():
    """PUT → 201 + Azure-AsyncOperation → poll until Succeeded → GET resource."""
    state = {"polls": 0}

    def handler(request: httpx.Request) -> httpx.Response:
        url = str(request.url)
        assert request.headers["Authorization"] == "Bearer tok"
        if request.method == "PUT":
            assert "/agentPools/gpu1" in url
            body = json.loads(request.content)
            assert body["properties"]["count"] == 1
            return httpx.Response(
                201,
                headers={"Azure-AsyncOperation": "https://arm/operations/op1"},
                json={"name": "gpu1", "properties": {"provisioningState": "Creating"}},
            )
        if "/operations/op1" in url:
            state["polls"] += 1
            status = "Succeeded" if state["polls"] >= 3 else "InProgress"
            return httpx.Response(200, json={"status": status})
        if request.method == "GET" and "/agentPools/gpu1" in url:
            return httpx.Response(
                200, json={"name": "gpu1", "properties": {"provisioningState": "Succeeded"}}
            )
        raise AssertionError(f"unexpected {request.method} {url}")

    async def main():
        client = make_client(handler)
        poller = await client.begin_create_or_update(
            "rg", "cluster", "gpu1", {"properties": {"count": 1, "vmSize": "x"}}
        )
        result = await poller.result()
        assert result["properties"]["provisioningState"] == "Succeeded"
        assert state["polls"] == 3

    run(main())


def test_retry_on_5xx_and_429():
    attempts = {"n": 0}

    def handler(request: httpx.Request) -> httpx.Response:
        attempts["n"] += 1
        if attempts["n"] == 1:
            return httpx.Response(503, json={"error": {"code": "ServerBusy"}})
        if attempts["n"] == 2:
            return httpx.Response(
                429, headers={"Retry-After": "0"}, json={"error": {"code": "Throttled"}}
            )
        return httpx.Response(200, json={"name": "gpu1", "properties": {}})

    async def main():
        # shrink the backoff so the test is fast
        import gpu_provisioner_amd.providers.instance.armclient as ac

        old = ac.RETRY_BASE_SECONDS
        ac.RETRY_BASE_SECONDS = 0.001
        try:
            client = make_client(handler)
            pool = await client.get("rg", "cluster", "gpu1")
            assert pool["name"] == "gpu1"
            assert attempts["n"] == 3
        finally:
            ac.RETRY_BASE_SECONDS = old

    run(main())


def test_arm_error_mapping():
    def handler(request: httpx.Request) -> httpx.Response:
        return httpx.Response(
            404, json={"error": {"code": "NotFound", "message": "no such pool"}}
        )

    async def main():
        client = make_client(handler)
        with pytest.raises(ARMError) as exc:
            await client.get("rg", "cluster", "nope")
        assert exc.value.status == 404
        assert exc.value.code == "NotFound"

    run(main())


def test_list_pager_follows_next_link():
    def handler(request: httpx.Request) -> httpx.Response:
        url = str(request.url)
        if "skip=1" in url:
            return httpx.Response(200, json={"value": [{"name": "b"}]})
        return httpx.Response(
            200,
            json={
                "value": [{"name": "a"}],
                "nextLink": url.split("?")[0] + "?skip=1&api-version=x",
            },
        )

    async def main():
        client = make_client(handler)
        names = [p["name"] async for p in client.list("rg", "cluster")]
        assert names == ["a", "b"]

    run(main())


def test_delete_immediate_204_completes_without_lro():
    def handler(request: httpx.Request) -> httpx.Response:
        assert request.method == "DELETE"
        return httpx.Response(204)

    async def main():
        client = make_client(handler)
        poller = await client.begin_delete("rg", "cluster", "gpu1")
        assert poller.done()
        await poller.result()

    run(main())


def test_lro_failure_raises_with_code():
    def handler(request: httpx.Request) -> httpx.Response:
        if request.method == "PUT":
            return httpx.Response(
                201, headers={"Azure-AsyncOperation": "https://arm/op"}, json={}
            )
        return httpx.Response(
            200,
            json={
                "status": "Failed",
                "error": {"code": "SkuNotAvailable", "message": "no MI355X capacity"},
            },
        )

    async def main():
        client = make_client(handler)
        poller = await client.begin_create_or_update("rg", "c", "p", {"properties": {}})
        with pytest.raises(ARMError) as exc:
            await poller.result()
        assert exc.value.code == "SkuNotAvailable"

    run(main())


def test_extra_headers_injected_on_every_request():
    """E2E pipeline header injection (reference azure_client.go:113-141)."""
    import httpx

    from gpu_provisioner_amd.providers.instance.armclient import ARMAgentPoolsClient

    seen = {}

    async def handler(request: httpx.Request) -> httpx.Response:
        seen.update(dict(request.headers))
        return httpx.Response(200, json={"name": "p1", "properties": {}})

    class FakeCred:
        async def get_token(self):
            return "tok"

    async def main():
        client = ARMAgentPoolsClient(
            FakeCred(), "sub1",
            http=httpx.AsyncClient(transport=httpx.MockTransport(handler)),
            extra_headers={"X-Kaito-E2E": "scenario-7"},
        )
        await client.get("rg", "cluster", "p1")
        assert seen.get("x-kaito-e2e") == "scenario-7"
        assert seen.get("authorization") == "Bearer tok"
        await client.close()

    run(main())
