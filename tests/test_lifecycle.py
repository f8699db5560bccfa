"""End-to-end lifecycle tests against the in-process stack: NodeClaim →
agent pool → node → Ready → Initialized → delete. Mirrors the behavioral
scenarios of the reference's lifecycle controller
(vendor/sigs.k8s.io/karpenter/pkg/controllers/nodeclaim/lifecycle/) and its
e2e suite provision/terminate specs (test/e2e/suites/suite_test.go)."""
import asyncio


from gpu_provisioner_amd.apis import v1 as karpv1
from gpu_provisioner_amd.controllers.lifecycle.controller import LifecycleController
from gpu_provisioner_amd.fake.harness import Harness
from gpu_provisioner_amd.kube import objects as ko
from gpu_provisioner_amd.providers.instance.armapi import ARMError
from tests.conftest import run


def make_harness(**kw) -> Harness:
    # lifecycle always runs with termination + eviction (registration puts the
    # termination finalizer on Nodes; the termination controller removes it)
    return Harness(**kw).add_all_controllers(
        lifecycle_workers=16, termination_requeue=0.05, with_health=False
    )


def test_provision_to_initialized_full_path():
    """The minimum e2e slice: kaito-labeled NodeClaim reaches Initialized with
    amd.com/gpu registered; node carries synced labels/owner-ref."""

    async def main():
        h = make_harness(ready_latency=0.05, plugin_latency=0.05)
        await h.start()
        try:
            nc = await h.kube.create(h.make_nodeclaim("gpu1"))
            done = await h.wait_initialized("gpu1")
            # conditions
            assert karpv1.is_launched(done)
            assert karpv1.is_registered(done)
            assert karpv1.is_initialized(done)
            # status populated
            assert done["status"]["providerID"].startswith("azure:///subscriptions/")
            assert done["status"]["nodeName"].startswith("aks-gpu1-")
            assert done["status"]["allocatable"][karpv1.AMD_GPU_RESOURCE] == "8"
            # finalizer present
            assert ko.has_finalizer(done, karpv1.TERMINATION_FINALIZER)
            # node got labels + owner ref + initialized label
            node = await h.kube.get("v1", "Node", done["status"]["nodeName"])
            labels = ko.labels_of(node)
            assert labels[karpv1.KAITO_WORKSPACE_LABEL_KEY] == "ws"
            assert labels[karpv1.NODEPOOL_LABEL_KEY] == "kaito"
            assert labels[karpv1.NODE_REGISTERED_LABEL_KEY] == "true"
            assert labels[karpv1.NODE_INITIALIZED_LABEL_KEY] == "true"
            assert labels[karpv1.AMD_GPU_PRODUCT_LABEL_KEY] == "AMD-Instinct-MI355X"
            owners = node["metadata"]["ownerReferences"]
            assert owners[0]["kind"] == "NodeClaim" and owners[0]["name"] == "gpu1"
            # agent pool shape
            pool = h.agent_pools.pools["gpu1"]
            props = pool["properties"]
            assert props["count"] == 1
            assert props["vmSize"] == "Standard_ND128isr_MI355X_v6"
            # stable ARM profile: ROCm bootstrap via kubeletConfig/labels,
            # no gpuProfile (not in the pinned 2024-09-01 schema — armschema.py)
            assert "gpuProfile" not in props
            assert props["kubeletConfig"]["topologyManagerPolicy"] == "single-numa-node"
        finally:
            await h.stop()

    run(main())


def test_initialization_gated_on_amd_gpu_allocatable():
    """NodeClaim must NOT be Initialized before the device plugin registers
    amd.com/gpu (the reference gate, initialization.go:119-133)."""

    async def main():
        h = make_harness(ready_latency=0.0, plugin_latency=0.4)
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("gpu2"))

            async def registered():
                nc = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "gpu2")
                return nc if karpv1.is_registered(nc) else None

            nc = await h.wait_for(registered)
            assert not karpv1.is_initialized(nc)
            cond = ko.get_condition(nc, karpv1.COND_INITIALIZED)
            if cond is not None:
                assert cond["status"] == "False"
            done = await h.wait_initialized("gpu2")
            assert done["status"]["capacity"][karpv1.AMD_GPU_RESOURCE] == "8"
        finally:
            await h.stop()

    run(main())


def test_delete_nodeclaim_tears_down_node_and_pool():
    """Deprovision path (reference §3.3): NodeClaim delete → node deleted →
    agent pool deleted → finalizer removed → NodeClaim gone."""

    async def main():
        h = make_harness()
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("gpu3"))
            done = await h.wait_initialized("gpu3")
            node_name = done["status"]["nodeName"]
            await h.kube.delete(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "gpu3")
            await h.wait_gone(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "gpu3")
            await h.wait_gone("v1", "Node", node_name)
            assert "gpu3" not in h.agent_pools.pools
        finally:
            await h.stop()

    run(main())


def test_unmanaged_nodeclaim_ignored():
    """A NodeClaim without kaito labels or a KaitoNodeClass ref must get no
    finalizer and no instance (reference e2e negative spec, suite_test.go:387-450)."""

    async def main():
        h = make_harness()
        await h.start()
        try:
            nc = karpv1.new_nodeclaim("other", labels={"app": "x"})
            nc["spec"] = {
                "requirements": [
                    {"key": karpv1.INSTANCE_TYPE_LABEL_KEY, "operator": "In",
                     "values": ["Standard_ND128isr_MI355X_v6"]}
                ],
                "nodeClassRef": {"group": "karpenter.azure.com", "kind": "AKSNodeClass", "name": "x"},
            }
            await h.kube.create(nc)
            await asyncio.sleep(0.3)
            got = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "other")
            assert not ko.has_finalizer(got, karpv1.TERMINATION_FINALIZER)
            assert not karpv1.is_launched(got)
            assert h.agent_pools.create_calls == 0
        finally:
            await h.stop()

    run(main())


def test_insufficient_capacity_deletes_nodeclaim():
    """InsufficientCapacity from the cloud deletes the NodeClaim so the owner
    retries (reference launch.go:79-124)."""

    async def main():
        h = make_harness()
        h.agent_pools.create_error.set(
            ARMError(409, "SkuNotAvailable", "MI355X capacity exhausted in region"), max_calls=1
        )
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("gpu4"))
            await h.wait_gone(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "gpu4")
            assert "gpu4" not in h.agent_pools.pools
        finally:
            await h.stop()

    run(main())


def test_create_error_retries_and_recovers():
    """A transient ARM failure surfaces as Launched=False then recovers via
    rate-limited requeue."""

    async def main():
        h = make_harness()
        h.agent_pools.create_error.set(ARMError(500, "InternalServerError", "boom"), max_calls=1)
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("gpu5"))
            done = await h.wait_initialized("gpu5")
            assert karpv1.is_initialized(done)
            assert h.agent_pools.create_calls >= 2
        finally:
            await h.stop()

    run(main())


def test_create_in_progress_is_adopted():
    """A create that crashed mid-LRO is adopted, not failed
    (reference instance.go:106-110)."""

    async def main():
        h = make_harness()
        await h.start()
        try:
            # pre-create the pool as if a previous incarnation started it
            poller = await h.agent_pools.begin_create_or_update(
                "rg", "cluster",
                "gpu6",
                {"properties": {"vmSize": "Standard_ND128isr_MI355X_v6", "nodeLabels": {}}},
            )
            await poller.result()
            # subsequent create attempts hit "in progress"
            h.agent_pools.create_error.set(
                ARMError(409, "AgentPoolOperationInProgress", "operation in progress"), max_calls=10
            )
            await h.kube.create(h.make_nodeclaim("gpu6"))
            done = await h.wait_initialized("gpu6")
            assert karpv1.is_initialized(done)
        finally:
            await h.stop()

    run(main())


def test_launch_idempotent_one_create_call():
    """Concurrent/repeated reconciles must produce exactly one ARM create
    (the UID idempotency cache, reference launch.go:38-76)."""

    async def main():
        h = make_harness(ready_latency=0.1, plugin_latency=0.1)
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("gpu7"))
            await h.wait_initialized("gpu7")
            assert h.agent_pools.create_calls == 1
        finally:
            await h.stop()

    run(main())


def test_spot_capacity_type_and_disk_request():
    async def main():
        h = make_harness()
        await h.start()
        try:
            nc = h.make_nodeclaim("gpu8")
            nc["spec"]["requirements"].append(
                {"key": karpv1.CAPACITY_TYPE_LABEL_KEY, "operator": "In", "values": ["spot"]}
            )
            nc["spec"]["resources"]["requests"]["ephemeral-storage"] = "256Gi"
            await h.kube.create(nc)
            await h.wait_initialized("gpu8")
            props = h.agent_pools.pools["gpu8"]["properties"]
            assert props["scaleSetPriority"] == "Spot"
            assert props["osDiskSizeGB"] == 256
        finally:
            await h.stop()

    run(main())


def test_azurelinux_image_family_annotation():
    """kaito.sh/node-image-family annotation drives OSSKU
    (reference instance.go:364,415-441 + e2e AzureLinux spec)."""

    async def main():
        h = make_harness()
        await h.start()
        try:
            nc = h.make_nodeclaim("gpu9")
            nc["metadata"]["annotations"] = {karpv1.NODE_IMAGE_FAMILY_ANNOTATION_KEY: "AzureLinux"}
            await h.kube.create(nc)
            await h.wait_initialized("gpu9")
            assert h.agent_pools.pools["gpu9"]["properties"]["osSKU"] == "AzureLinux"
        finally:
            await h.stop()

    run(main())


def test_invalid_agent_pool_name_rejected():
    async def main():
        h = make_harness()
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("Bad-Name-Far-Too-Long-For-AKS".lower()[:20]))
            await asyncio.sleep(0.3)
            assert h.agent_pools.create_calls == 0
        finally:
            await h.stop()

    run(main())


def test_registration_liveness_disabled_by_default():
    """The reference ships liveness disabled (controller.go:154); ours is a
    deliberate, gated choice — no TTL unless RegistrationLiveness is on."""
    from gpu_provisioner_amd.fake.harness import Harness

    h = Harness().add_all_controllers()
    assert h.lifecycle.registration_ttl is None


def test_registration_liveness_deletes_stuck_nodeclaim():
    """With the gate on, a NodeClaim that launches but never Registers (node
    never appears) is deleted once the TTL elapses."""
    import gpu_provisioner_amd.controllers.lifecycle.controller as lc
    from gpu_provisioner_amd.fake.harness import Harness

    async def main():
        # AKS sim that never brings the node up: infinite ready latency
        h = Harness(ready_latency=3600.0, node_wait_interval=0.01)
        h.instances.node_wait_attempts = 2  # launch fails fast on node wait
        h.add_all_controllers(gc_interval=120.0, with_drift=False)
        h.lifecycle.registration_ttl = 0.5
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("stuck1"))
            # never reaches Registered; liveness deletes it after the TTL
            await h.wait_gone(
                karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "stuck1", timeout=30
            )
        finally:
            await h.stop()

    run(main())


def test_registration_liveness_gate_wiring():
    from gpu_provisioner_amd.controllers.lifecycle.controller import (
        REGISTRATION_TTL_SECONDS,
        LifecycleController,
    )
    from gpu_provisioner_amd.fake.harness import Harness
    from gpu_provisioner_amd.main import build_manager
    from gpu_provisioner_amd.operator.options import Options

    h = Harness()
    opts = Options.from_env_and_args(
        ["--feature-gates", "RegistrationLiveness=true"], {}
    )
    mgr = build_manager(h.kube, opts, h.cloud.inner)
    lc = [c for c in mgr.controllers if isinstance(c, LifecycleController)][0]
    assert lc.registration_ttl == REGISTRATION_TTL_SECONDS
    mgr2 = build_manager(h.kube, Options.from_env_and_args([], {}), h.cloud.inner)
    lc2 = [c for c in mgr2.controllers if isinstance(c, LifecycleController)][0]
    assert lc2.registration_ttl is None


def test_lro_allocation_failure_deletes_claim_and_gc_cleans_failed_pool():
    """Azure surfaces most allocation failures DURING the create LRO (the
    PUT is accepted, the LRO ends Failed). Launch must map it to
    InsufficientCapacity and delete the claim; the leaked Failed pool is
    then instance-GC'd."""
    from gpu_provisioner_amd.fake.harness import Harness
    from gpu_provisioner_amd.providers.instance.armapi import ARMError

    async def main():
        h = Harness(node_wait_interval=0.01).add_all_controllers(
            gc_interval=0.3, adoption_age=0.1
        )
        h.agent_pools.create_lro_error.set(
            ARMError(200, "AllocationFailed", "zone exhausted mid-provision"),
            max_calls=1,
        )
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("lrofail1"))
            # launch maps LRO failure → InsufficientCapacity → claim deleted
            await h.wait_gone(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "lrofail1", timeout=30)

            async def failed_pool_collected():
                return "lrofail1" not in h.agent_pools.pools or None

            await h.wait_for(failed_pool_collected, timeout=30)
        finally:
            await h.stop()

    run(main())
