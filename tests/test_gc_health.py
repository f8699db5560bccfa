"""Garbage-collection (both directions) and node auto-repair tests.
Behavioral spec: reference pkg/controllers/instance/garbagecollection/ and
vendor/.../nodeclaim/garbagecollection/ (§3.4), node/health (§3.5)."""
import asyncio


from gpu_provisioner_amd.apis import v1 as karpv1
from gpu_provisioner_amd.fake.harness import Harness
from gpu_provisioner_amd.kube import objects as ko
from tests.conftest import run


def make_harness(gc_interval=0.2, adoption_age=0.2, **kw) -> Harness:
    return Harness(**kw).add_all_controllers(
        gc_interval=gc_interval, adoption_age=adoption_age
    )


def test_instance_gc_deletes_leaked_pool_and_node():
    """A pool whose NodeClaim vanished (deleted mid-create crash) is adopted
    and deleted after the 30s-equivalent age floor, along with its leaked
    Node (reference instance GC :51-124)."""

    async def main():
        h = make_harness(gc_interval=0.2, adoption_age=0.3)
        await h.start()
        try:
            # manufacture a leak: create a pool directly (as if the controller
            # crashed after the ARM call but before NodeClaim status patch)
            prov = h.instances
            nc = h.make_nodeclaim("leak1")
            pool = prov.new_agent_pool_object(nc, "Standard_ND128isr_MI355X_v6")
            poller = await h.agent_pools.begin_create_or_update("rg", "cluster", "leak1", pool)
            await poller.result()
            assert "leak1" in h.agent_pools.pools
            node_name = h.aks.node_name("leak1")
            await h.kube.get("v1", "Node", node_name)  # node materialized
            # within the age floor it is NOT collected
            await asyncio.sleep(0.2)
            assert "leak1" in h.agent_pools.pools

            async def pool_gone():
                return "leak1" not in h.agent_pools.pools or None

            await h.wait_for(pool_gone, timeout=10)
            await h.wait_gone("v1", "Node", node_name, timeout=10)
        finally:
            await h.stop()

    run(main())


def test_instance_gc_spares_live_nodeclaims():
    async def main():
        h = make_harness(gc_interval=0.1, adoption_age=0.0)
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("live1"))
            await h.wait_initialized("live1")
            await asyncio.sleep(0.5)  # several GC sweeps
            assert "live1" in h.agent_pools.pools
            nc = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "live1")
            assert not ko.is_deleting(nc)
        finally:
            await h.stop()

    run(main())


def test_nodeclaim_gc_deletes_claims_whose_instance_vanished():
    """Registered NodeClaim whose pool was deleted out-of-band (cloud-side
    drift) is GC'd once its node stops reporting Ready (reference vendored GC
    :60-118)."""

    async def main():
        h = make_harness(gc_interval=0.2)
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("drift1"))
            done = await h.wait_initialized("drift1")
            node_name = done["status"]["nodeName"]
            # cloud-side out-of-band deletion: pool vanishes AND kubelet dies
            h.agent_pools.pools.pop("drift1")
            await h.kube.patch(
                "v1", "Node", node_name,
                {"status": {"conditions": [
                    {"type": "Ready", "status": "Unknown", "reason": "NodeStatusUnknown"}
                ]}},
                subresource="status",
            )
            await h.wait_gone(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "drift1", timeout=15)
        finally:
            await h.stop()

    run(main())


def test_nodeclaim_gc_trusts_ready_node_over_cloud_list():
    """If the node is still Ready, a transiently incomplete cloud List must
    NOT kill the claim (the guard at reference :77-98)."""

    async def main():
        h = make_harness(gc_interval=0.1)
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("keep1"))
            await h.wait_initialized("keep1")
            # pool vanishes from list() but kubelet still Ready
            saved = h.agent_pools.pools.pop("keep1")
            await asyncio.sleep(0.5)
            nc = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "keep1")
            assert not ko.is_deleting(nc)
            h.agent_pools.pools["keep1"] = saved
        finally:
            await h.stop()

    run(main())


def test_health_repair_replaces_unhealthy_node():
    """NodeReady=False past the toleration window → NodeClaim force-deleted
    with a termination-timestamp annotation (reference §3.5)."""

    async def main():
        h = make_harness()
        # shrink the 10-min toleration for test speed
        for p in h.cloud.repair_policies():
            p.toleration_seconds = 0.3
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("sick1"))
            done = await h.wait_initialized("sick1")
            node_name = done["status"]["nodeName"]
            await h.kube.patch(
                "v1", "Node", node_name,
                {"status": {"conditions": [
                    {"type": "Ready", "status": "False", "reason": "KubeletNotReady"}
                ]}},
                subresource="status",
            )
            await h.wait_gone(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "sick1", timeout=15)
            await h.wait_gone("v1", "Node", node_name, timeout=15)
            assert "sick1" not in h.agent_pools.pools
        finally:
            await h.stop()

    run(main())


def test_health_tolerates_brief_unreadiness():
    async def main():
        h = make_harness()
        for p in h.cloud.repair_policies():
            p.toleration_seconds = 5.0
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("flap1"))
            done = await h.wait_initialized("flap1")
            node_name = done["status"]["nodeName"]
            await h.kube.patch(
                "v1", "Node", node_name,
                {"status": {"conditions": [
                    {"type": "Ready", "status": "False", "reason": "Flap"}
                ]}},
                subresource="status",
            )
            await asyncio.sleep(0.3)
            # recovers before toleration elapses
            await h.kube.patch(
                "v1", "Node", node_name,
                {"status": {"conditions": [
                    {"type": "Ready", "status": "True", "reason": "KubeletReady"}
                ]}},
                subresource="status",
            )
            await asyncio.sleep(0.5)
            nc = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "flap1")
            assert not ko.is_deleting(nc)
        finally:
            await h.stop()

    run(main())
