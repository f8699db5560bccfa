"""The MI355X GPU-health repair loop (net-new vs the reference's
NodeReady-only repair): the nodeagent publishes AMDGPUHealthy on its Node,
the cloud provider's RepairPolicies watch it with a 5-min toleration, and
node.health replaces GPU-sick nodes through the normal termination path."""
import asyncio

from gpu_provisioner_amd.apis import v1 as karpv1
from gpu_provisioner_amd.fake.harness import Harness
from gpu_provisioner_amd.kube import objects as ko
from gpu_provisioner_amd.nodeagent import (
    GPUReport,
    NodeReport,
    node_condition_from_report,
    patch_node_condition,
)
from tests.conftest import run


def healthy_report() -> NodeReport:
    g = GPUReport(index=0, arch="gfx950", cus=256, healthy=True)
    return NodeReport(healthy=True, gpu_count=8, gpus=[g])


def sick_report() -> NodeReport:
    g = GPUReport(index=0, problems=["MFMA bf16 selftest failed: elem 3 got 0 want 192"])
    return NodeReport(
        healthy=False,
        gpu_count=8,
        gpus=[g],
        problems=["gpu0: MFMA bf16 selftest failed: elem 3 got 0 want 192"],
    )


def test_condition_shapes():
    c = node_condition_from_report(healthy_report())
    assert c["type"] == karpv1.AMD_GPU_HEALTHY_CONDITION_TYPE
    assert c["status"] == "True" and c["reason"] == "AllChecksPassed"
    c = node_condition_from_report(sick_report())
    assert c["status"] == "False" and c["reason"] == "GPUUnhealthy"
    assert "MFMA" in c["message"]


def test_repair_policies_include_gpu_health():
    h = Harness()
    types = {(p.condition_type, p.condition_status) for p in h.cloud.repair_policies()}
    assert (karpv1.AMD_GPU_HEALTHY_CONDITION_TYPE, ko.CONDITION_FALSE) in types
    assert ("Ready", ko.CONDITION_FALSE) in types
    assert ("Ready", ko.CONDITION_UNKNOWN) in types


def test_patch_node_condition_merges_not_replaces():
    async def main():
        h = Harness().add_all_controllers(gc_interval=60.0, with_health=False)
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("cond1"))
            nc = await h.wait_initialized("cond1")
            node_name = nc["status"]["nodeName"]
            await patch_node_condition(h.kube, node_name, healthy_report())
            node = await h.kube.get("v1", "Node", node_name)
            # existing conditions (Ready) survive the merge
            assert ko.node_is_ready(node)
            assert ko.condition_is_true(node, karpv1.AMD_GPU_HEALTHY_CONDITION_TYPE)
            # flip to sick: transition recorded
            await patch_node_condition(h.kube, node_name, sick_report())
            node = await h.kube.get("v1", "Node", node_name)
            cond = ko.get_condition(node, karpv1.AMD_GPU_HEALTHY_CONDITION_TYPE)
            assert cond["status"] == ko.CONDITION_FALSE
            assert ko.node_is_ready(node)
        finally:
            await h.stop()

    run(main())


def test_gpu_sick_node_is_replaced_after_toleration():
    async def main():
        h = Harness(gpu_repair_toleration=0.2).add_all_controllers(gc_interval=60.0)
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("sick1"))
            nc = await h.wait_initialized("sick1")
            node_name = nc["status"]["nodeName"]
            await patch_node_condition(h.kube, node_name, sick_report())
            # toleration elapses → NodeClaim force-terminated, pool torn down
            await h.wait_gone(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "sick1", timeout=20)

            async def pool_gone():
                return "sick1" not in h.agent_pools.pools or None

            await h.wait_for(pool_gone, timeout=20)
        finally:
            await h.stop()

    run(main())


def test_gpu_recovery_within_toleration_cancels_repair():
    async def main():
        h = Harness(gpu_repair_toleration=1.5).add_all_controllers(gc_interval=60.0)
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("flap1"))
            nc = await h.wait_initialized("flap1")
            node_name = nc["status"]["nodeName"]
            await patch_node_condition(h.kube, node_name, sick_report())
            await asyncio.sleep(0.3)
            await patch_node_condition(h.kube, node_name, healthy_report())
            await asyncio.sleep(1.8)  # past the original toleration window
            got = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "flap1")
            assert not ko.is_deleting(got)
            assert "flap1" in h.agent_pools.pools
        finally:
            await h.stop()

    run(main())


def test_patch_node_condition_retries_conflicts_with_fresh_read():
    """the status patch carries the read's
    resourceVersion; on a conflicting concurrent write it re-reads and
    retries, so a kubelet condition written in between is never clobbered
    or resurrected."""

    async def main():
        from gpu_provisioner_amd.kube.client import ConflictError

        h = Harness().add_all_controllers(gc_interval=60.0, with_health=False)
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("cond2"))
            nc = await h.wait_initialized("cond2")
            node_name = nc["status"]["nodeName"]

            # kubelet writes a condition the agent's earlier read has not
            # seen; the agent's FIRST patch attempt conflicts (injected),
            # forcing the retry path against a fresh read
            node = await h.kube.get("v1", "Node", node_name)
            ko.set_condition(node, "KubeletFresh", "True", "Heartbeat", "")
            await h.kube.update_status(node)

            fired = {"n": 0}

            def conflict_once(verb, gvk, payload):
                if verb == "patch" and gvk == ("v1", "Node") and fired["n"] == 0:
                    fired["n"] += 1
                    return ConflictError("injected concurrent write")
                return None

            h.server.reactors.append(conflict_once)
            await patch_node_condition(h.kube, node_name, healthy_report())
            h.server.reactors.clear()
            assert fired["n"] == 1  # the retry actually happened

            after = await h.kube.get("v1", "Node", node_name)
            assert ko.condition_is_true(after, karpv1.AMD_GPU_HEALTHY_CONDITION_TYPE)
            # the kubelet's concurrent condition survived the merge
            assert ko.condition_is_true(after, "KubeletFresh")
            assert ko.node_is_ready(after)
        finally:
            await h.stop()

    run(main())
