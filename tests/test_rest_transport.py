"""Production transport over real sockets — the "envtest-lite" tier.

The PRODUCTION wiring (main.build_manager: informers, all controllers,
eviction queue) runs with kube/http.py's HTTPClient speaking to the kube
REST surface served by fake/restserver.py over 127.0.0.1 — actual HTTP,
actual chunked-JSON watch streams, actual Status error bodies. The AKS
simulator plays the cloud + kubelet actors on a direct in-memory client,
exactly like real out-of-process actors.

This closes the round-1 gap "kube/http.py is exercised solely via
httpx.MockTransport": every verb the controllers use now crosses the wire.
"""
import asyncio

import pytest

from gpu_provisioner_amd.apis import v1 as karpv1
from gpu_provisioner_amd.cloudprovider.azure import AzureCloudProvider
from gpu_provisioner_amd.fake.agentpools import AKSSimulator, FakeAgentPools
from gpu_provisioner_amd.fake.apiserver import InMemoryAPIServer, InMemoryClient
from gpu_provisioner_amd.fake.restserver import RESTServerHandle
from gpu_provisioner_amd.kube import objects as ko
from gpu_provisioner_amd.kube.client import ConflictError, InvalidError, NotFoundError
from gpu_provisioner_amd.kube.http import HTTPClient
from gpu_provisioner_amd.main import build_manager
from gpu_provisioner_amd.operator.options import Options
from gpu_provisioner_amd.providers.instance.provider import InstanceProvider
from gpu_provisioner_amd.providers.instancetype.catalog import InstanceTypeProvider
from tests.conftest import run

VM = "Standard_ND128isr_MI355X_v6"


def spec_nodeclaim(name: str) -> dict:
    nc = karpv1.new_nodeclaim(name, labels={karpv1.KAITO_WORKSPACE_LABEL_KEY: "w"})
    nc["spec"] = {
        "requirements": [
            {"key": karpv1.INSTANCE_TYPE_LABEL_KEY, "operator": "In", "values": [VM]}
        ],
        "resources": {"requests": {karpv1.AMD_GPU_RESOURCE: "8"}},
        "nodeClassRef": {"group": "kaito.sh", "kind": "KaitoNodeClass", "name": "default"},
    }
    return nc


async def wait_until(fn, timeout=30.0, interval=0.05):
    deadline = asyncio.get_event_loop().time() + timeout
    while True:
        val = await fn()
        if val:
            return val
        if asyncio.get_event_loop().time() > deadline:
            raise TimeoutError(getattr(fn, "__name__", "condition"))
        await asyncio.sleep(interval)


def test_full_provision_cycle_over_http_transport():
    async def main():
        server = InMemoryAPIServer()
        rest = RESTServerHandle(server)
        port = await rest.start()
        kube = HTTPClient(f"http://127.0.0.1:{port}")

        # cloud + kubelet actors on their own direct client (separate
        # process in reality)
        actor_client = InMemoryClient(server)
        catalog = InstanceTypeProvider()
        pools = FakeAgentPools()
        aks = AKSSimulator(
            actor_client, pools, ready_latency=0.05, plugin_latency=0.05,
            gpu_count_for=catalog.gpu_count,
        )
        instances = InstanceProvider(
            pools, kube, catalog, "rg", "cluster", node_wait_interval=0.02
        )
        cloud = AzureCloudProvider(instances, catalog)

        options = Options()  # defaults: no leader election, gates default
        manager = build_manager(kube, options, cloud)
        await manager.start(serve_http=False)
        try:
            # -- provision to Initialized over the wire
            await kube.create(spec_nodeclaim("wire1"))

            async def initialized():
                try:
                    nc = await kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "wire1")
                except NotFoundError:
                    return None
                return nc if karpv1.is_initialized(nc) else None

            nc = await wait_until(initialized)
            node_name = nc["status"]["nodeName"]
            node = await kube.get("v1", "Node", node_name)
            assert ko.node_is_ready(node)
            assert ko.qty(ko.node_allocatable(node)[karpv1.AMD_GPU_RESOURCE]) == ko.qty("8")
            assert ko.has_finalizer(nc, karpv1.TERMINATION_FINALIZER)

            # -- wire-level semantics checks with the production client
            with pytest.raises(InvalidError):
                await kube.list(
                    "storage.k8s.io/v1", "VolumeAttachment",
                    field_selector="spec.nodeName=x",
                )
            stale = ko.deep_copy(nc)
            await kube.patch(
                karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "wire1",
                {"metadata": {"labels": {**ko.labels_of(nc), "x": "1"}}},
            )
            with pytest.raises(ConflictError):
                await kube.patch(
                    karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "wire1",
                    {
                        "metadata": {
                            "resourceVersion": stale["metadata"]["resourceVersion"],
                            "labels": {"y": "2"},
                        }
                    },
                )

            # -- teardown over the wire: NodeClaim delete drains everything
            await kube.delete(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "wire1")

            async def all_gone():
                try:
                    await kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "wire1")
                    return None
                except NotFoundError:
                    pass
                try:
                    await kube.get("v1", "Node", node_name)
                    return None
                except NotFoundError:
                    pass
                return "wire1" not in pools.pools or None

            await wait_until(all_gone)
        finally:
            await manager.stop()
            await kube.close()
            await rest.stop()

    run(main(), timeout=120)


def test_watch_gone_triggers_relist_over_http():
    """A watch opened at an expired resourceVersion gets the real
    apiserver's 200 + ERROR(410) framing; the production client raises
    GoneError and the informer relists."""

    async def main():
        from gpu_provisioner_amd.kube.informer import Informer

        server = InMemoryAPIServer()
        rest = RESTServerHandle(server)
        port = await rest.start()
        kube = HTTPClient(f"http://127.0.0.1:{port}")
        actor = InMemoryClient(server)
        try:
            # age the history far past the ring so rv=1 is provably expired
            for i in range(5000):
                await actor.create(
                    {"apiVersion": "v1", "kind": "Node", "metadata": {"name": f"n{i}"}}
                )
                if i % 2 == 0:
                    await actor.delete("v1", "Node", f"n{i}")
            from gpu_provisioner_amd.kube.client import GoneError

            with pytest.raises(GoneError):
                async for _ in kube.watch("v1", "Node", resource_version="1"):
                    break
            # the informer recovers by relisting (production loop)
            inf = Informer(kube, "v1", "Node")
            inf.start()
            await asyncio.wait_for(inf.wait_for_sync(), 10)
            assert len(inf.list()) == 2500
            await inf.stop()
        finally:
            await kube.close()
            await rest.stop()

    run(main(), timeout=120)


def test_eviction_subresource_over_http():
    async def main():
        server = InMemoryAPIServer()
        rest = RESTServerHandle(server)
        port = await rest.start()
        kube = HTTPClient(f"http://127.0.0.1:{port}")
        actor = InMemoryClient(server)
        try:
            await actor.create(
                {"apiVersion": "v1", "kind": "Pod",
                 "metadata": {"name": "p1", "namespace": "default"}, "spec": {}}
            )
            await kube.evict(
                {"apiVersion": "v1", "kind": "Pod",
                 "metadata": {"name": "p1", "namespace": "default"}},
                grace_period_seconds=0,
            )
            assert server.evictions == [("default", "p1")]
            with pytest.raises(NotFoundError):
                await kube.get("v1", "Pod", "p1", "default")
        finally:
            await kube.close()
            await rest.stop()

    run(main(), timeout=60)


def test_watch_bookmarks_advance_informer_rv_over_http():
    """Real apiservers interleave BOOKMARK events; the production client
    yields them and the informer advances its resume resourceVersion
    without treating them as object events."""

    async def main():
        from gpu_provisioner_amd.kube.client import BOOKMARK
        from gpu_provisioner_amd.kube.informer import Informer

        server = InMemoryAPIServer()
        rest = RESTServerHandle(server)
        port = await rest.start()
        kube = HTTPClient(f"http://127.0.0.1:{port}")
        actor = InMemoryClient(server)
        try:
            inf = Informer(kube, "v1", "Node")
            events = []
            inf.add_handler(lambda et, obj: events.append(et))
            inf.start()
            await asyncio.wait_for(inf.wait_for_sync(), 10)
            rv_before = int(inf._rv or 0)
            # 30 events → at least one server-emitted bookmark in the stream
            for i in range(30):
                await actor.create(
                    {"apiVersion": "v1", "kind": "Node", "metadata": {"name": f"bm{i}"}}
                )
            await wait_until(lambda: _done(len(inf.list()) == 30))
            assert int(inf._rv) > rv_before
            # bookmarks are NOT delivered to handlers as object events
            assert BOOKMARK not in events
            assert events.count("ADDED") == 30
            # raw client surface: the bookmark event itself is observable
            seen_bookmark = False
            count = 0
            async for et, obj in kube.watch("v1", "Node", resource_version="0"):
                count += 1
                if et == BOOKMARK:
                    seen_bookmark = True
                    assert obj["metadata"]["resourceVersion"]
                    break
                if count > 40:
                    break
            assert seen_bookmark
            await inf.stop()
        finally:
            await kube.close()
            await rest.stop()

    async def _done(x):
        return x or None

    run(main(), timeout=60)
