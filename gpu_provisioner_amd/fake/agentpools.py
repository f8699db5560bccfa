"""Fake AKS: in-memory AgentPoolsAPI + cluster simulator.

The multi-actor test substrate, modeled on the reference's programmable fakes
(pkg/fake/azure_client.go MockAgentPoolsAPI, pkg/fake/types.go MockedLRO with
output override/error injection/call counting, pkg/fake/atomic.go AtomicError
with MaxCalls) — but as one coherent simulator: agent pools move through
Creating→Succeeded LRO states on a scripted latency, and the AKSSimulator
plays kubelet + AMD device plugin, materializing Node objects into the
in-memory apiserver (agentpool labels, providerID, Ready flip, amd.com/gpu
allocatable) exactly as a real cluster would — so the full controller stack
runs e2e in-process.
"""
from __future__ import annotations

import asyncio
import hashlib
import logging
import time
from typing import AsyncIterator, Optional

from ..apis import v1 as karpv1
from ..kube import objects as ko
from ..kube.client import (
    AlreadyExistsError,
    APIError,
    KubeClient,
    NotFoundError,
)
from ..providers.instance.armapi import (
    AgentPoolsAPI,
    ARMError,
    LROPoller,
    pool_name,
    taint_from_string,
)
from ..utils.utils import build_provider_id

log = logging.getLogger(__name__)


class ScriptedError:
    """Error injection with call budget (reference pkg/fake/atomic.go:108-130
    AtomicError + MaxCalls)."""

    def __init__(self):
        self._err: Optional[Exception] = None
        self._calls_left = 0

    def set(self, err: Exception, max_calls: int = 1) -> None:
        self._err = err
        self._calls_left = max_calls

    def clear(self) -> None:
        self._err = None
        self._calls_left = 0

    def check(self) -> None:
        if self._err is not None and self._calls_left > 0:
            self._calls_left -= 1
            err = self._err
            if self._calls_left == 0:
                self._err = None
            raise err


class FakeLRO(LROPoller):
    def __init__(self, fn, latency: float, result_error: Optional[Exception] = None):
        self._fn = fn  # () -> dict, runs at completion
        self._deadline = time.monotonic() + latency
        self._result: Optional[dict] = None
        self._error = result_error

    def done(self) -> bool:
        return time.monotonic() >= self._deadline

    async def poll(self) -> Optional[dict]:
        if self.done():
            return await self._complete()
        return None

    async def result(self) -> dict:
        remaining = self._deadline - time.monotonic()
        if remaining > 0:
            await asyncio.sleep(remaining)
        return await self._complete()

    async def _complete(self) -> dict:
        if self._error is not None:
            raise self._error
        if self._result is None:
            self._result = await self._fn()
        return self._result


class FakeAgentPools(AgentPoolsAPI):
    """In-memory agent-pool store with scripted latencies and error injection."""

    def __init__(
        self,
        *,
        create_latency: float = 0.0,
        delete_latency: float = 0.0,
        api_latency: float = 0.0,
    ):
        self.pools: dict = {}  # name -> agent pool dict
        self.create_latency = create_latency
        self.delete_latency = delete_latency
        self.api_latency = api_latency
        self.create_error = ScriptedError()
        # the LRO ends "Failed" AFTER accept — how real Azure surfaces most
        # allocation failures (discovered during provisioning, not at PUT)
        self.create_lro_error = ScriptedError()
        self.delete_error = ScriptedError()
        self.get_error = ScriptedError()
        self.list_error = ScriptedError()
        self.create_calls = 0
        self.delete_calls = 0
        self.on_pool_ready = None  # hook(pool dict) — wired by AKSSimulator
        self.on_pool_deleted = None  # hook(name)

    async def _lat(self) -> None:
        if self.api_latency:
            await asyncio.sleep(self.api_latency)

    async def begin_create_or_update(
        self, resource_group: str, cluster_name: str, pool_name_: str, agent_pool: dict
    ) -> LROPoller:
        await self._lat()
        self.create_calls += 1
        self.create_error.check()
        stored = ko.deep_copy(agent_pool)
        stored["name"] = pool_name_
        stored["id"] = (
            f"/subscriptions/sub/resourceGroups/{resource_group}/providers/"
            f"Microsoft.ContainerService/managedClusters/{cluster_name}/agentPools/{pool_name_}"
        )
        stored.setdefault("properties", {})["provisioningState"] = "Creating"
        self.pools[pool_name_] = stored

        async def complete() -> dict:
            cur = self.pools.get(pool_name_)
            if cur is None:
                raise ARMError(404, "NotFound", f"agent pool {pool_name_} was deleted mid-create")
            try:
                self.create_lro_error.check()
            except Exception:
                cur["properties"]["provisioningState"] = "Failed"
                raise
            cur["properties"]["provisioningState"] = "Succeeded"
            if self.on_pool_ready is not None:
                await self.on_pool_ready(cur)
            return ko.deep_copy(cur)

        return FakeLRO(complete, self.create_latency)

    async def begin_delete(
        self, resource_group: str, cluster_name: str, pool_name_: str
    ) -> LROPoller:
        await self._lat()
        self.delete_calls += 1
        self.delete_error.check()
        cur = self.pools.get(pool_name_)
        if cur is None:
            raise ARMError(404, "NotFound", f"agent pool {pool_name_} not found")
        cur["properties"]["provisioningState"] = "Deleting"

        async def complete() -> dict:
            gone = self.pools.pop(pool_name_, None)
            if self.on_pool_deleted is not None and gone is not None:
                await self.on_pool_deleted(pool_name_)
            return {"name": pool_name_}

        return FakeLRO(complete, self.delete_latency)

    async def get(self, resource_group: str, cluster_name: str, pool_name_: str) -> dict:
        await self._lat()
        self.get_error.check()
        cur = self.pools.get(pool_name_)
        if cur is None:
            raise ARMError(404, "NotFound", f"agent pool {pool_name_} not found")
        return ko.deep_copy(cur)

    async def list(self, resource_group: str, cluster_name: str) -> AsyncIterator[dict]:
        await self._lat()
        self.list_error.check()
        for p in list(self.pools.values()):
            yield ko.deep_copy(p)


class AKSSimulator:
    """Plays the AKS control plane + kubelet + AMD device plugin: when an
    agent pool completes its create LRO, a Node appears (agentpool labels,
    VMSS providerID, NotReady), flips Ready after `ready_latency`, and gains
    amd.com/gpu capacity/allocatable after `plugin_latency` — the sequence the
    lifecycle controller's registration + initialization gates consume."""

    def __init__(
        self,
        kube: KubeClient,
        agent_pools: FakeAgentPools,
        *,
        subscription: str = "sub",
        resource_group: str = "MC_rg_cluster_loc",
        ready_latency: float = 0.0,
        plugin_latency: float = 0.0,
        gpu_count_for=None,  # fn(vm_size) -> int
    ):
        self.kube = kube
        self.agent_pools = agent_pools
        self.subscription = subscription
        self.resource_group = resource_group
        self.ready_latency = ready_latency
        self.plugin_latency = plugin_latency
        self.gpu_count_for = gpu_count_for or (lambda vm_size: 8)
        self._tasks: set = set()
        agent_pools.on_pool_ready = self._on_pool_ready
        agent_pools.on_pool_deleted = self._on_pool_deleted

    def node_name(self, pool: str) -> str:
        return f"aks-{pool}-{self._hash(pool)}-vmss000000"

    def provider_id(self, pool: str) -> str:
        return build_provider_id(self.subscription, self.resource_group, pool, self._hash(pool))

    @staticmethod
    def _hash(pool: str) -> str:
        return hashlib.sha1(pool.encode()).hexdigest()[:8]

    async def _on_pool_ready(self, pool: dict) -> None:
        name = pool_name(pool)
        props = pool.get("properties", {})
        node = {
            "apiVersion": "v1",
            "kind": "Node",
            "metadata": {
                "name": self.node_name(name),
                "labels": {
                    karpv1.AGENTPOOL_LABEL_KEY: name,
                    karpv1.AZURE_AGENTPOOL_LABEL_KEY: name,
                    karpv1.HOSTNAME_LABEL_KEY: self.node_name(name),
                    karpv1.INSTANCE_TYPE_LABEL_KEY: props.get("vmSize", ""),
                    **{k: v for k, v in (props.get("nodeLabels") or {}).items()},
                },
            },
            "spec": {
                "providerID": self.provider_id(name),
                "taints": [taint_from_string(s) for s in props.get("nodeTaints") or []]
                + [{"key": "node.kubernetes.io/not-ready", "effect": "NoSchedule"}],
            },
            "status": {
                "conditions": [
                    {"type": "Ready", "status": "False", "reason": "KubeletNotReady"}
                ],
                "capacity": {"cpu": "128", "memory": "2048Gi", "pods": "250"},
                "allocatable": {"cpu": "127", "memory": "2036Gi", "pods": "250"},
                "nodeInfo": {
                    "osImage": f"{props.get('osSKU', 'Ubuntu')} (ROCm)",
                    "kubeletVersion": "v1.31.0",
                },
            },
        }
        # kubelet semantics: node registration retries until it lands
        for _ in range(200):
            try:
                await self.kube.create(node)
                break
            except AlreadyExistsError:
                break
            except APIError:
                await asyncio.sleep(0.02)
        else:
            log.warning("node %s never registered", self.node_name(name))
            return
        self._spawn(self._become_ready(name, props))

    async def _become_ready(self, pool: str, props: dict) -> None:
        if self.ready_latency:
            await asyncio.sleep(self.ready_latency)
        node_name = self.node_name(pool)
        # kubelet Ready + drop the not-ready taint. The removal is
        # read-modify-write on spec.taints, so it carries the node's
        # resourceVersion as an optimistic lock (a raced unconditioned merge
        # would clobber the registration controller's taint sync — the same
        # lost-update bug in the other direction) and retries like a real
        # kubelet until it lands.
        for _ in range(200):
            try:
                node = await self.kube.get("v1", "Node", node_name)
            except NotFoundError:
                return
            except APIError:
                await asyncio.sleep(0.02)
                continue
            taints = [
                t
                for t in node.get("spec", {}).get("taints") or []
                if t.get("key") != "node.kubernetes.io/not-ready"
            ]
            try:
                await self.kube.patch(
                    "v1",
                    "Node",
                    node_name,
                    {
                        "metadata": {
                            "resourceVersion": node["metadata"].get("resourceVersion")
                        },
                        "spec": {"taints": taints or None},
                    },
                )
                break
            except NotFoundError:
                return
            except APIError:
                await asyncio.sleep(0.02)
        await self._retry_status_patch(
            node_name,
            {
                "status": {
                    "conditions": [
                        {"type": "Ready", "status": "True", "reason": "KubeletReady"}
                    ]
                }
            },
        )
        if self.plugin_latency:
            await asyncio.sleep(self.plugin_latency)
        # AMD device plugin registers amd.com/gpu
        gpus = str(self.gpu_count_for(props.get("vmSize", "")))
        await self._retry_status_patch(
            node_name,
            {
                "status": {
                    "capacity": {karpv1.AMD_GPU_RESOURCE: gpus},
                    "allocatable": {karpv1.AMD_GPU_RESOURCE: gpus},
                }
            },
        )

    async def _retry_status_patch(self, node_name: str, patch: dict) -> None:
        for _ in range(200):
            try:
                await self.kube.patch("v1", "Node", node_name, patch, subresource="status")
                return
            except NotFoundError:
                return
            except APIError:
                await asyncio.sleep(0.02)

    async def _on_pool_deleted(self, pool: str) -> None:
        for _ in range(200):
            try:
                await self.kube.delete("v1", "Node", self.node_name(pool))
                return
            except NotFoundError:
                return
            except APIError:
                await asyncio.sleep(0.02)

    def _spawn(self, coro) -> None:
        task = asyncio.get_event_loop().create_task(coro)
        self._tasks.add(task)
        task.add_done_callback(self._tasks.discard)

    async def drain(self) -> None:
        while self._tasks:
            await asyncio.gather(*list(self._tasks), return_exceptions=True)
