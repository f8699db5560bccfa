"""Real ARM AgentPools client: httpx against the Azure ARM REST API.

The production implementation of the AgentPoolsAPI seam (armapi.py).
Replaces the reference's armcontainerservice.AgentPoolsClient + azcore
pipeline (pkg/providers/instance/azure_client.go, pkg/utils/opts/): bearer
auth from a TokenCredential, the reference's retry policy (20 retries, 5 s
exponential base — armopts.go:34-40), a pooled HTTP client (the armbalancer
shape — init_http_client.go:29-52), the telemetry user agent, and LRO
polling via Azure-AsyncOperation/Location headers with Retry-After.
"""
from __future__ import annotations

import asyncio
import logging
import random
from typing import AsyncIterator, Optional

import httpx

from ...auth.cred import TokenCredential
from ...metrics.registry import ARM_REQUEST_DURATION, ARM_RETRIES
from .armapi import AgentPoolsAPI, ARMError, LROPoller
from .armschema import PROFILE_STABLE

log = logging.getLogger(__name__)

# default pinned api-version; overridden per-client by the ArmApiProfile
# (armschema.py) so the request schema and the api-version always travel
# together
API_VERSION = PROFILE_STABLE.api_version
MAX_RETRIES = 20  # reference armopts.go:34-40
RETRY_BASE_SECONDS = 5.0
RETRY_CAP_SECONDS = 60.0
_RETRYABLE_STATUS = {408, 429, 500, 502, 503, 504}


class ARMLROPoller(LROPoller):
    """Polls an Azure long-running operation to completion."""

    def __init__(self, client: "ARMAgentPoolsClient", poll_url: str, resource_url: str):
        self.client = client
        self.poll_url = poll_url
        self.resource_url = resource_url
        self._done = False
        self._status = ""

    def done(self) -> bool:
        return self._done

    async def poll(self) -> Optional[dict]:
        resp = await self.client._request("GET", self.poll_url, operation="poll")
        body = resp.json() if resp.content else {}
        self._status = (
            body.get("status") or body.get("properties", {}).get("provisioningState", "")
        )
        if self._status.lower() in ("succeeded", "failed", "canceled"):
            self._done = True
        if self._status.lower() == "succeeded":
            return await self._fetch_resource()
        if self._done:
            err = body.get("error", {})
            raise ARMError(
                200,
                err.get("code", self._status or "OperationFailed"),
                err.get("message", f"LRO ended with status {self._status}"),
            )
        return None

    async def result(self) -> dict:
        while not self._done:
            result = await self.poll()
            if result is not None:
                return result
            if not self._done:
                await asyncio.sleep(self.client.lro_poll_interval)
        return await self._fetch_resource()

    async def _fetch_resource(self) -> dict:
        try:
            resp = await self.client._request("GET", self.resource_url, operation="get")
        except ARMError as e:
            if e.status == 404:
                return {}  # delete LROs: the resource is gone on success
            raise
        return resp.json() if resp.content else {}


class ARMAgentPoolsClient(AgentPoolsAPI):
    def __init__(
        self,
        credential: TokenCredential,
        subscription_id: str,
        *,
        endpoint: str = "https://management.azure.com",
        user_agent: str = "gpu-provisioner-amd/0.1.0",
        http: Optional[httpx.AsyncClient] = None,
        lro_poll_interval: float = 5.0,
        max_retries: int = MAX_RETRIES,
        extra_headers: Optional[dict] = None,
        api_version: str = API_VERSION,
    ):
        self.credential = credential
        self.subscription_id = subscription_id
        self.api_version = api_version
        self.endpoint = endpoint.rstrip("/")
        self.lro_poll_interval = lro_poll_interval
        self.max_retries = max_retries
        # per-request header injection — the reference's E2E pipeline policy
        # (azure_client.go:113-141) stamps test-scenario headers this way
        self.extra_headers = dict(extra_headers or {})
        self.http = http or httpx.AsyncClient(
            timeout=httpx.Timeout(30.0, read=120.0),
            # the reference fans requests over an armbalancer pool of 100
            # (init_http_client.go:29-52): same envelope via httpx limits
            limits=httpx.Limits(max_connections=100, max_keepalive_connections=100),
            headers={"User-Agent": user_agent},
        )

    def _pool_url(self, rg: str, cluster: str, pool: str = "") -> str:
        base = (
            f"{self.endpoint}/subscriptions/{self.subscription_id}/resourceGroups/{rg}"
            f"/providers/Microsoft.ContainerService/managedClusters/{cluster}/agentPools"
        )
        return f"{base}/{pool}" if pool else base

    async def _request(
        self, method: str, url: str, json_body: Optional[dict] = None, operation: str = ""
    ) -> httpx.Response:
        """One ARM call with auth + the reference's retry envelope."""
        import time as _time

        last_exc: Optional[Exception] = None
        for attempt in range(self.max_retries + 1):
            token = await self.credential.get_token()
            start = _time.monotonic()
            try:
                # nextLink/Azure-AsyncOperation URLs already carry their query
                # string; httpx `params` would replace it
                params = None if "?" in url else {"api-version": self.api_version}
                resp = await self.http.request(
                    method,
                    url,
                    json=json_body,
                    params=params,
                    headers={"Authorization": f"Bearer {token}", **self.extra_headers},
                )
            except httpx.TransportError as e:
                last_exc = e
                ARM_RETRIES.labels(operation=operation).inc()
                await asyncio.sleep(self._backoff(attempt))
                continue
            ARM_REQUEST_DURATION.labels(
                operation=operation, code=str(resp.status_code)
            ).observe(_time.monotonic() - start)
            if resp.status_code in _RETRYABLE_STATUS and attempt < self.max_retries:
                ARM_RETRIES.labels(operation=operation).inc()
                retry_after = float(resp.headers.get("Retry-After", 0)) or self._backoff(attempt)
                await asyncio.sleep(retry_after)
                continue
            if resp.status_code >= 400:
                raise self._arm_error(resp)
            return resp
        raise ARMError(599, "TransportError", f"{operation}: {last_exc}")

    @staticmethod
    def _backoff(attempt: int) -> float:
        return min(RETRY_BASE_SECONDS * (2**attempt), RETRY_CAP_SECONDS) * (
            1 + random.uniform(-0.2, 0.2)
        )

    @staticmethod
    def _arm_error(resp: httpx.Response) -> ARMError:
        try:
            body = resp.json()
            err = body.get("error", body)
            return ARMError(
                resp.status_code, err.get("code", "Unknown"), err.get("message", "")
            )
        except Exception:
            return ARMError(resp.status_code, "Unknown", resp.text[:300])

    # ----------------------------------------------------------------- API

    async def begin_create_or_update(
        self, resource_group: str, cluster_name: str, pool_name: str, agent_pool: dict
    ) -> LROPoller:
        url = self._pool_url(resource_group, cluster_name, pool_name)
        resp = await self._request("PUT", url, json_body=agent_pool, operation="create")
        poll_url = resp.headers.get("Azure-AsyncOperation") or resp.headers.get("Location") or url
        return ARMLROPoller(self, poll_url, url)

    async def begin_delete(
        self, resource_group: str, cluster_name: str, pool_name: str
    ) -> LROPoller:
        url = self._pool_url(resource_group, cluster_name, pool_name)
        resp = await self._request("DELETE", url, operation="delete")
        poll_url = resp.headers.get("Azure-AsyncOperation") or resp.headers.get("Location")
        if resp.status_code in (200, 204) and not poll_url:
            return _CompletedPoller({"name": pool_name})
        return ARMLROPoller(self, poll_url or url, url)

    async def get(self, resource_group: str, cluster_name: str, pool_name: str) -> dict:
        resp = await self._request(
            "GET", self._pool_url(resource_group, cluster_name, pool_name), operation="get"
        )
        return resp.json()

    async def list(self, resource_group: str, cluster_name: str) -> AsyncIterator[dict]:
        url: Optional[str] = self._pool_url(resource_group, cluster_name)
        while url:
            resp = await self._request("GET", url, operation="list")
            body = resp.json()
            for item in body.get("value", []):
                yield item
            url = body.get("nextLink")

    async def close(self) -> None:
        await self.http.aclose()


class _CompletedPoller(LROPoller):
    def __init__(self, result: dict):
        self._result = result

    def done(self) -> bool:
        return True

    async def poll(self) -> Optional[dict]:
        return self._result

    async def result(self) -> dict:
        return self._result
