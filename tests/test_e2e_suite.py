"""The e2e suite — the reference's 8 ginkgo scenarios
(test/e2e/suites/suite_test.go) plus scale/fault tiers.

The 8 reference scenarios (spec1-spec8) are DUAL-BACKEND: each spec drives
only the tests/e2e_env.py environment surface, so the identical functions
run against the in-process controller topology + AKS simulator (default) or
a LIVE cluster (`E2E_LIVE=1` + KUBECONFIG, controller deployed via the
chart). Cloud-internal assertions (agent-pool bodies, create counters) are
checked where the backend can observe them and skipped on live, exactly as
the reference's ginkgo suite judges outcomes by the kube-visible surface.

The scale/perf/fault specs (spec9+) intentionally stay in-process only:
they script simulator latency, fault reactors and zero-latency churn that a
live cluster cannot reproduce deterministically.
"""
import asyncio
import os

import pytest

from gpu_provisioner_amd.apis import v1 as karpv1
from gpu_provisioner_amd.apis import v1alpha1
from gpu_provisioner_amd.fake.harness import Harness
from gpu_provisioner_amd.kube import objects as ko
from tests.conftest import run
from tests.e2e_env import make_env, spec_nodeclaim

LIVE = os.environ.get("E2E_LIVE", "") == "1"


def run_spec(spec, timeout: float = 120.0, **env_kw):
    """Run one dual-backend spec body against the selected environment."""

    async def main():
        env = make_env(**env_kw)
        await env.start()
        try:
            await spec(env)
        finally:
            await env.stop()

    # live expectations run on the reference's 10-min SLO; give the spec
    # room for several of them plus cleanup
    run(main(), timeout=3600.0 if LIVE else timeout)


# --------------------------------------------------------------- spec 1-8
# (reference test/e2e/suites/suite_test.go — same scenarios, same order)


def test_spec1_provision_via_workspace_label():
    async def spec(env):
        await env.create(env.nodeclaim("ws1", {karpv1.KAITO_WORKSPACE_LABEL_KEY: "llm"}))
        nc = await env.wait_initialized("ws1")
        node = await env.kube.get("v1", "Node", nc["status"]["nodeName"])
        assert ko.node_is_ready(node)
        assert ko.qty(ko.node_allocatable(node)[karpv1.AMD_GPU_RESOURCE]) == ko.qty("8")
        assert ko.labels_of(node)[karpv1.KAITO_WORKSPACE_LABEL_KEY] == "llm"

    run_spec(spec)


def test_spec2_provision_via_ragengine_label():
    async def spec(env):
        await env.create(env.nodeclaim("rag1", {karpv1.KAITO_RAGENGINE_LABEL_KEY: "rag"}))
        nc = await env.wait_initialized("rag1")
        assert karpv1.is_initialized(nc)

    run_spec(spec)


def test_spec3_terminate_via_nodeclaim_delete():
    async def spec(env):
        await env.create(env.nodeclaim("del1", {karpv1.KAITO_WORKSPACE_LABEL_KEY: "x"}))
        nc = await env.wait_initialized("del1")
        node_name = nc["status"]["nodeName"]
        await env.kube.delete(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "del1")
        await env.wait_gone(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "del1")
        await env.wait_gone("v1", "Node", node_name)
        await env.expect_pool_gone("del1")

    run_spec(spec)


def test_spec4_terminate_via_node_delete():
    async def spec(env):
        await env.create(env.nodeclaim("del2", {karpv1.KAITO_WORKSPACE_LABEL_KEY: "x"}))
        nc = await env.wait_initialized("del2")
        node_name = nc["status"]["nodeName"]
        await env.kube.delete("v1", "Node", node_name)
        await env.wait_gone("v1", "Node", node_name)
        await env.wait_gone(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "del2")
        await env.expect_pool_gone("del2")

    run_spec(spec)


def test_spec5_provision_via_kaitonodeclass_ref():
    """Managed purely by NodeClassRef GroupKind, no kaito labels."""

    async def spec(env):
        try:
            await env.create(v1alpha1.new_kaitonodeclass("default"))
        except Exception:
            pass  # live cluster may already carry the class
        await env.create(
            env.nodeclaim(
                "ncref1",
                labels={"app": "custom"},
                node_class=v1alpha1.node_class_ref("default"),
            )
        )
        nc = await env.wait_initialized("ncref1")
        assert karpv1.is_initialized(nc)

    run_spec(spec)


def test_spec6_negative_foreign_nodeclass_ignored():
    """AKSNodeClass ref + no kaito labels → no finalizer, no instance, no
    node (reference suite_test.go:387-450)."""

    async def spec(env):
        before = env.create_calls()
        await env.create(
            env.nodeclaim(
                "foreign1",
                labels={"app": "other"},
                node_class={
                    "group": "karpenter.azure.com",
                    "kind": "AKSNodeClass",
                    "name": "d",
                },
            )
        )
        await asyncio.sleep(15.0 if env.is_live else 0.5)
        nc = await env.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "foreign1")
        assert not ko.has_finalizer(nc, karpv1.TERMINATION_FINALIZER)
        assert not karpv1.is_launched(nc)
        after = env.create_calls()
        if before is not None and after is not None:
            assert after == before
        # cleanup (no finalizer → immediate)
        await env.kube.delete(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "foreign1")

    run_spec(spec)


def test_spec7_azurelinux_annotation_sets_os_image():
    async def spec(env):
        await env.create(
            env.nodeclaim(
                "azlinux1",
                labels={karpv1.KAITO_WORKSPACE_LABEL_KEY: "x"},
                annotations={karpv1.NODE_IMAGE_FAMILY_ANNOTATION_KEY: "AzureLinux"},
            )
        )
        nc = await env.wait_initialized("azlinux1")
        props = env.pool_properties("azlinux1")
        if props is not None:
            assert props["osSKU"] == "AzureLinux"
        node = await env.kube.get("v1", "Node", nc["status"]["nodeName"])
        os_image = node["status"]["nodeInfo"]["osImage"]
        assert any(
            s in os_image for s in ("AzureLinux", "Azure Linux", "CBL-Mariner", "Mariner")
        ), os_image

    run_spec(spec)


def test_spec8_delete_while_provisioning():
    """Deleting a NodeClaim mid-create must still tear everything down (the
    GC-covered crash window, reference delete-trigger spec + GC readme)."""

    async def spec(env):
        await env.create(env.nodeclaim("mid1", {karpv1.KAITO_WORKSPACE_LABEL_KEY: "x"}))

        async def create_started():
            # backend-observable create counter, else the controller's
            # finalizer ack — both mean "the create path is in flight"
            calls = env.create_calls()
            if calls is not None:
                return True if calls > 0 else None
            nc = await env.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "mid1")
            return True if ko.has_finalizer(nc, karpv1.TERMINATION_FINALIZER) else None

        await env.eventually(create_started, desc="create started")
        # delete while the agent-pool LRO is still running
        await env.kube.delete(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "mid1")
        await env.wait_gone(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "mid1")
        await env.expect_pool_gone("mid1")
        # no leaked nodes either
        for node in await env.kube.list("v1", "Node"):
            assert ko.labels_of(node).get(karpv1.AGENTPOOL_LABEL_KEY) != "mid1"

    run_spec(
        spec,
        create_latency=0.3,
        ready_latency=0.05,
        controllers={"gc_interval": 0.3, "adoption_age": 0.1},
    )


# ---------------------------------------------------- in-process-only tiers
# (scripted simulator latency/faults — not reproducible on a live cluster)

pytestmark_scale = pytest.mark.skipif(
    LIVE, reason="scale/fault tiers script the simulator; live runs cover spec1-8"
)


def env_harness(**kw) -> Harness:
    return Harness(ready_latency=0.05, plugin_latency=0.05, **kw).add_all_controllers(
        gc_interval=60.0
    )


@pytestmark_scale
def test_spec10_32_concurrent_churn_with_drift():
    """BASELINE config #5: 32 concurrent NodeClaims with drift detection +
    delete/disruption reconcile under churn. One third of the fleet is
    mutated out-of-band (drift → detected), one third deleted mid-life
    (churn), the rest must stay Initialized and un-drifted throughout."""

    async def main():
        h = Harness(node_wait_interval=0.002).add_all_controllers(
            lifecycle_workers=64, gc_interval=60.0, drift_interval=0.05,
            drift_replace=True,
        )
        await h.start()
        try:
            names = [f"churn{i:02d}" for i in range(32)]
            await asyncio.gather(
                *(
                    h.kube.create(
                        spec_nodeclaim(n, {karpv1.KAITO_WORKSPACE_LABEL_KEY: "fleet"})
                    )
                    for n in names
                )
            )
            await asyncio.gather(*(h.wait_initialized(n, timeout=30) for n in names))

            drifting, deleting, steady = names[:10], names[10:20], names[20:]
            for n in drifting:
                h.agent_pools.pools[n]["properties"]["osSKU"] = "AzureLinux"
            await asyncio.gather(
                *(h.kube.delete(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, n) for n in deleting)
            )
            # drifted claims are detected and (DriftReplace on) replaced:
            # the claim is deleted and its pool torn down, concurrently with
            # the explicit churn deletes
            await asyncio.gather(
                *(
                    h.wait_gone(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, n, timeout=30)
                    for n in drifting + deleting
                )
            )

            async def pools_gone():
                return (
                    all(n not in h.agent_pools.pools for n in drifting + deleting) or None
                )

            await h.wait_for(pools_gone, timeout=30)
            # the steady fleet survived the churn untouched
            for n in steady:
                nc = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, n)
                assert karpv1.is_initialized(nc)
                assert not ko.is_deleting(nc)
                assert not ko.condition_is_true(nc, karpv1.COND_DRIFTED)
                assert n in h.agent_pools.pools
        finally:
            await h.stop()

    run(main())


@pytestmark_scale
def test_spec11_concurrent_provisioning_pipelines():
    """8 concurrent NodeClaims against a cloud with real latency (0.3s LRO +
    0.05s node-ready) must provision in ~one latency budget, not 8 serial
    ones — the lifecycle controller's worker pool and the per-claim LRO waits
    must overlap (BASELINE config #3 shape)."""

    async def main():
        h = Harness(
            create_latency=0.3, ready_latency=0.05, node_wait_interval=0.01
        ).add_all_controllers(gc_interval=60.0)
        await h.start()
        try:
            import time

            names = [f"par{i}" for i in range(8)]
            t0 = time.monotonic()
            await asyncio.gather(
                *(
                    h.kube.create(
                        spec_nodeclaim(n, {karpv1.KAITO_WORKSPACE_LABEL_KEY: "w"})
                    )
                    for n in names
                )
            )
            await asyncio.gather(*(h.wait_initialized(n, timeout=20) for n in names))
            elapsed = time.monotonic() - t0
            # serial would be >= 8 * 0.35s = 2.8s; pipelined stays well under
            assert elapsed < 2.0, f"provisioning serialized: {elapsed:.2f}s for 8 claims"
        finally:
            await h.stop()

    run(main())


@pytestmark_scale
def test_spec9_workload_pod_binds_to_provisioned_node():
    """BASELINE config #4: a workload pod requesting amd.com/gpu schedules
    onto the provisioned MI355X node (binding simulated at the apiserver)."""

    async def main():
        h = env_harness()
        await h.start()
        try:
            await h.kube.create(spec_nodeclaim("bind1", {karpv1.KAITO_WORKSPACE_LABEL_KEY: "x"}))
            nc = await h.wait_initialized("bind1")
            node_name = nc["status"]["nodeName"]
            node = await h.kube.get("v1", "Node", node_name)
            # the scheduler's feasibility view: allocatable has 8 amd.com/gpu
            assert ko.qty(ko.node_allocatable(node)[karpv1.AMD_GPU_RESOURCE]) == ko.qty("8")
            pod = {
                "apiVersion": "v1",
                "kind": "Pod",
                "metadata": {"name": "train", "namespace": "default"},
                "spec": {
                    "nodeName": node_name,
                    "containers": [
                        {"name": "c", "resources": {"requests": {karpv1.AMD_GPU_RESOURCE: "8"}}}
                    ],
                },
                "status": {"phase": "Running"},
            }
            await h.kube.create(pod)
            got = await h.kube.get("v1", "Pod", "train", "default")
            assert got["spec"]["nodeName"] == node_name
        finally:
            await h.stop()

    run(main())


@pytestmark_scale
def test_spec12_128_concurrent_churn_completes():
    """Regression for the teardown livelock: 128 concurrent NodeClaims
    through a full churn cycle must complete — sub-10ms requeue backstops
    plus unconditional condition patches once turned finalize into a
    406k-reconciles-per-35s event storm that starved all progress."""

    async def main():
        import time

        h = Harness(node_wait_interval=0.01).add_all_controllers(
            lifecycle_workers=256, termination_requeue=0.02, drain_requeue=0.02,
            instance_poll=0.02, gc_interval=60.0, with_health=False,
        )
        await h.start()
        try:
            names = [f"big{i:03d}" for i in range(128)]
            t0 = time.monotonic()
            await asyncio.gather(
                *(
                    h.kube.create(
                        spec_nodeclaim(n, {karpv1.KAITO_WORKSPACE_LABEL_KEY: "w"})
                    )
                    for n in names
                )
            )
            await asyncio.gather(*(h.wait_initialized(n, timeout=60) for n in names))
            await asyncio.gather(
                *(
                    h.kube.delete(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, n)
                    for n in names
                )
            )
            await asyncio.gather(
                *(
                    h.wait_gone(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, n, timeout=60)
                    for n in names
                )
            )
            assert not h.agent_pools.pools
            assert time.monotonic() - t0 < 90
        finally:
            await h.stop()

    run(main(), timeout=200)


@pytestmark_scale
def test_spec13_512_claim_burst():
    """Burst scale: 512 concurrent NodeClaims (64 full MI355X hosts worth)
    provision and tear down without queue/worker degradation."""

    async def main():
        import time

        h = Harness(node_wait_interval=0.01).add_all_controllers(
            lifecycle_workers=256, termination_workers=128,
            termination_requeue=0.02, drain_requeue=0.02, instance_poll=0.02,
            gc_interval=60.0, with_health=False,
        )
        await h.start()
        try:
            names = [f"burst{i:03d}" for i in range(512)]
            t0 = time.monotonic()
            await asyncio.gather(
                *(
                    h.kube.create(
                        spec_nodeclaim(n, {karpv1.KAITO_WORKSPACE_LABEL_KEY: "w"})
                    )
                    for n in names
                )
            )
            await asyncio.gather(*(h.wait_initialized(n, timeout=120) for n in names))
            await asyncio.gather(
                *(
                    h.kube.delete(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, n)
                    for n in names
                )
            )
            await asyncio.gather(
                *(
                    h.wait_gone(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, n, timeout=120)
                    for n in names
                )
            )
            assert not h.agent_pools.pools
            assert time.monotonic() - t0 < 60
        finally:
            await h.stop()

    run(main(), timeout=240)


@pytestmark_scale
def test_metrics_wired_through_churn():
    """The karpenter_* metric series must actually move when the lifecycle
    acts — guards against silent metric rot."""

    async def main():
        from gpu_provisioner_amd.metrics import registry as m

        def counter_val(c, **labels):
            return c.labels(**labels)._value.get()

        labels = dict(
            nodepool="kaito", capacity_type="on-demand",
            instance_type="Standard_ND128isr_MI355X_v6",
        )
        created0 = counter_val(m.NODECLAIMS_CREATED, **labels)
        terminated0 = counter_val(m.NODECLAIMS_TERMINATED, **labels)

        h = env_harness()
        await h.start()
        try:
            await h.kube.create(spec_nodeclaim("met1", {karpv1.KAITO_WORKSPACE_LABEL_KEY: "w"}))
            await h.wait_initialized("met1")
            await h.kube.delete(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "met1")
            await h.wait_gone(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "met1")
        finally:
            await h.stop()

        assert counter_val(m.NODECLAIMS_CREATED, **labels) == created0 + 1
        assert counter_val(m.NODECLAIMS_TERMINATED, **labels) == terminated0 + 1
        # launch duration observed at least once
        h1 = m.LAUNCH_DURATION.labels(nodepool="kaito")
        assert sum(b.get() for b in h1._buckets) >= 1

    run(main())


@pytestmark_scale
def test_spec14_spot_zone_pinned_claim():
    """The spot example's shape end-to-end: spot capacity type + zone
    pinning land on the pool (Spot priority, availabilityZones) and the
    cheapest eligible SKU is chosen."""

    async def main():
        h = env_harness()
        await h.start()
        try:
            nc = karpv1.new_nodeclaim(
                "spot1", labels={karpv1.KAITO_WORKSPACE_LABEL_KEY: "batch"}
            )
            nc["metadata"]["annotations"] = {karpv1.DO_NOT_DISRUPT_ANNOTATION_KEY: "true"}
            nc["spec"] = {
                "nodeClassRef": {"group": "kaito.sh", "kind": "KaitoNodeClass", "name": "default"},
                "requirements": [
                    {"key": karpv1.INSTANCE_TYPE_LABEL_KEY, "operator": "In",
                     "values": ["Standard_ND64is_MI355X_v6", "Standard_ND32is_MI355X_v6"]},
                    {"key": karpv1.CAPACITY_TYPE_LABEL_KEY, "operator": "In", "values": ["spot"]},
                    {"key": karpv1.ZONE_LABEL_KEY, "operator": "In",
                     "values": ["eastus2-1", "eastus2-2"]},
                ],
                "resources": {"requests": {karpv1.AMD_GPU_RESOURCE: "2"}},
            }
            await h.kube.create(nc)
            got = await h.wait_initialized("spot1")
            props = h.agent_pools.pools["spot1"]["properties"]
            assert props["vmSize"] == "Standard_ND32is_MI355X_v6"  # cheapest eligible
            assert props["scaleSetPriority"] == "Spot"
            assert props["scaleSetEvictionPolicy"] == "Delete"
            assert props["availabilityZones"] == ["1", "2"]
            assert got["status"]["allocatable"][karpv1.AMD_GPU_RESOURCE] == "2"
            labels = ko.labels_of(
                await h.kube.get("v1", "Node", got["status"]["nodeName"])
            )
            assert labels[karpv1.CAPACITY_TYPE_LABEL_KEY] == "spot"
        finally:
            await h.stop()

    run(main())


@pytestmark_scale
def test_spec15_2048_claim_burst():
    """Fleet scale: 2048 concurrent NodeClaims (256 full MI355X hosts,
    16,384 GPUs) provision and tear down in one burst — queue, informer
    and index structures must stay O(change), not O(fleet)."""

    async def main():
        import time

        h = Harness(node_wait_interval=0.01).add_all_controllers(
            lifecycle_workers=512, termination_workers=256,
            termination_requeue=0.02, drain_requeue=0.02, instance_poll=0.02,
            gc_interval=120.0, with_health=False,
        )
        await h.start()
        try:
            names = [f"burst{i:04d}" for i in range(2048)]
            t0 = time.monotonic()
            await asyncio.gather(
                *(
                    h.kube.create(
                        spec_nodeclaim(n, {karpv1.KAITO_WORKSPACE_LABEL_KEY: "w"})
                    )
                    for n in names
                )
            )
            await asyncio.gather(*(h.wait_initialized(n, timeout=240) for n in names))
            await asyncio.gather(
                *(
                    h.kube.delete(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, n)
                    for n in names
                )
            )
            await asyncio.gather(
                *(
                    h.wait_gone(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, n, timeout=240)
                    for n in names
                )
            )
            assert not h.agent_pools.pools
            # measured ~4.5s on an 8-core container; generous CI bound
            assert time.monotonic() - t0 < 120
        finally:
            await h.stop()

    run(main(), timeout=500)
