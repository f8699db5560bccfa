"""Drift detection + replacement (net-new vs the reference's IsDrifted stub,
reference pkg/cloudprovider/cloudprovider.go:94-97): the cloud provider
compares the live agent pool against the NodeClaim's declared shape, the
nodeclaim.drift singleton maintains the Drifted condition, and the
DriftReplace gate turns detection into replacement."""
import asyncio


from gpu_provisioner_amd.apis import v1 as karpv1
from gpu_provisioner_amd.cloudprovider.azure import (
    DRIFT_INSTANCE_TYPE,
    DRIFT_NODE_IMAGE,
    DRIFT_SKU_RETIRED,
)
from gpu_provisioner_amd.fake.harness import Harness
from gpu_provisioner_amd.kube import objects as ko
from tests.conftest import run

VM = "Standard_ND128isr_MI355X_v6"


def drift_env(**kw) -> Harness:
    return Harness().add_all_controllers(gc_interval=60.0, drift_interval=0.1, **kw)


async def provisioned(h: Harness, name: str, **kw) -> dict:
    await h.kube.create(h.make_nodeclaim(name, **kw))
    return await h.wait_initialized(name)


# --------------------------------------------------------- is_drifted (unit)


def test_is_drifted_fresh_claim_is_not_drifted():
    async def main():
        h = Harness().add_all_controllers(gc_interval=60.0, with_drift=False)
        await h.start()
        try:
            nc = await provisioned(h, "fresh1")
            assert await h.cloud.is_drifted(nc) == ""
        finally:
            await h.stop()

    run(main())


def test_is_drifted_node_image_pool_mutated():
    """Pool osSKU mutated out-of-band → NodeImageDrift."""

    async def main():
        h = Harness().add_all_controllers(gc_interval=60.0, with_drift=False)
        await h.start()
        try:
            nc = await provisioned(h, "img1")
            h.agent_pools.pools["img1"]["properties"]["osSKU"] = "AzureLinux"
            assert await h.cloud.is_drifted(nc) == DRIFT_NODE_IMAGE
        finally:
            await h.stop()

    run(main())


def test_is_drifted_node_image_annotation_changed():
    """kaito.sh/node-image-family annotation changed after provisioning →
    the Ubuntu pool no longer matches → NodeImageDrift."""

    async def main():
        h = Harness().add_all_controllers(gc_interval=60.0, with_drift=False)
        await h.start()
        try:
            nc = await provisioned(h, "img2")
            ko.meta(nc).setdefault("annotations", {})[
                karpv1.NODE_IMAGE_FAMILY_ANNOTATION_KEY
            ] = "AzureLinux"
            assert await h.cloud.is_drifted(nc) == DRIFT_NODE_IMAGE
        finally:
            await h.stop()

    run(main())


def test_is_drifted_instance_type():
    async def main():
        h = Harness().add_all_controllers(gc_interval=60.0, with_drift=False)
        await h.start()
        try:
            nc = await provisioned(h, "type1")
            nc["spec"]["requirements"] = [
                {
                    "key": karpv1.INSTANCE_TYPE_LABEL_KEY,
                    "operator": "In",
                    "values": ["Standard_ND64is_MI355X_v6"],
                }
            ]
            assert await h.cloud.is_drifted(nc) == DRIFT_INSTANCE_TYPE
        finally:
            await h.stop()

    run(main())


def test_is_drifted_sku_retired():
    async def main():
        h = Harness().add_all_controllers(gc_interval=60.0, with_drift=False)
        await h.start()
        try:
            nc = await provisioned(h, "retired1")
            # drop the instance-type requirement so only catalog membership is
            # checked, then retire the SKU from the catalog
            nc["spec"]["requirements"] = []
            del h.catalog._types[VM]
            assert await h.cloud.is_drifted(nc) == DRIFT_SKU_RETIRED
        finally:
            await h.stop()

    run(main())


def test_is_drifted_vanished_instance_is_not_drift():
    """A vanished pool is GC's problem, not drift."""

    async def main():
        h = Harness().add_all_controllers(gc_interval=60.0, with_drift=False)
        await h.start()
        try:
            nc = await provisioned(h, "gone1")
            del h.agent_pools.pools["gone1"]
            assert await h.cloud.is_drifted(nc) == ""
        finally:
            await h.stop()

    run(main())


def test_is_drifted_no_provider_id():
    async def main():
        h = Harness().add_all_controllers(gc_interval=60.0, with_drift=False)
        await h.start()
        try:
            assert await h.cloud.is_drifted(h.make_nodeclaim("nopid")) == ""
        finally:
            await h.stop()

    run(main())


# ------------------------------------------------------- drift controller


def test_drift_controller_sets_and_clears_condition():
    async def main():
        h = drift_env()
        await h.start()
        try:
            await provisioned(h, "dc1")
            h.agent_pools.pools["dc1"]["properties"]["osSKU"] = "AzureLinux"

            async def drifted():
                nc = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "dc1")
                c = ko.get_condition(nc, karpv1.COND_DRIFTED)
                return c if c and c.get("status") == ko.CONDITION_TRUE else None

            cond = await h.wait_for(drifted)
            assert cond["reason"] == DRIFT_NODE_IMAGE

            # pool reconverges → condition flips to False (history retained)
            h.agent_pools.pools["dc1"]["properties"]["osSKU"] = "Ubuntu"

            async def cleared():
                nc = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "dc1")
                c = ko.get_condition(nc, karpv1.COND_DRIFTED)
                return c if c and c.get("status") == ko.CONDITION_FALSE else None

            assert await h.wait_for(cleared)
        finally:
            await h.stop()

    run(main())


def test_drift_controller_never_drifted_keeps_status_untouched():
    async def main():
        h = drift_env()
        await h.start()
        try:
            await provisioned(h, "clean1")
            await asyncio.sleep(0.3)  # several drift sweeps
            nc = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "clean1")
            assert ko.get_condition(nc, karpv1.COND_DRIFTED) is None
        finally:
            await h.stop()

    run(main())


def test_drift_replace_deletes_drifted_claim():
    async def main():
        h = drift_env(drift_replace=True)
        await h.start()
        try:
            await provisioned(h, "rep1")
            h.agent_pools.pools["rep1"]["properties"]["osSKU"] = "AzureLinux"
            await h.wait_gone(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "rep1", timeout=20)

            async def pool_gone():
                return "rep1" not in h.agent_pools.pools or None

            await h.wait_for(pool_gone, timeout=20)
        finally:
            await h.stop()

    run(main())


def test_drift_replace_honors_do_not_disrupt():
    async def main():
        h = drift_env(drift_replace=True)
        await h.start()
        try:
            nc = h.make_nodeclaim("keep1")
            nc["metadata"]["annotations"] = {karpv1.DO_NOT_DISRUPT_ANNOTATION_KEY: "true"}
            await h.kube.create(nc)
            await h.wait_initialized("keep1")
            h.agent_pools.pools["keep1"]["properties"]["osSKU"] = "AzureLinux"

            async def drifted():
                got = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "keep1")
                return ko.condition_is_true(got, karpv1.COND_DRIFTED) or None

            await h.wait_for(drifted)
            await asyncio.sleep(0.3)  # more sweeps: still not deleted
            got = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "keep1")
            assert not ko.is_deleting(got)
        finally:
            await h.stop()

    run(main())


# ------------------------------------------------------------ feature gates


def test_feature_gates_parse_drift():
    from gpu_provisioner_amd.operator.options import FeatureGates

    g = FeatureGates.parse("NodeRepair=true,Drift=true,DriftReplace=true")
    assert g.drift and g.drift_replace and g.node_repair
    g = FeatureGates.parse("Drift=false")
    assert not g.drift and not g.drift_replace
    g = FeatureGates.parse("")
    assert g.drift and not g.drift_replace  # defaults


def test_build_manager_registers_drift_controller():
    from gpu_provisioner_amd.controllers.drift.controller import DriftController
    from gpu_provisioner_amd.main import build_manager
    from gpu_provisioner_amd.operator.options import Options

    async def main():
        h = Harness()
        options = Options.from_env_and_args([], {})
        mgr = build_manager(h.kube, options, h.cloud.inner if hasattr(h.cloud, "inner") else h.cloud)
        kinds = [type(c).__name__ for c in mgr.controllers]
        assert "DriftController" in kinds
        drift = [c for c in mgr.controllers if isinstance(c, DriftController)][0]
        assert drift.replace is False  # DriftReplace defaults off

        options2 = Options.from_env_and_args(["--feature-gates", "Drift=false"], {})
        mgr2 = build_manager(h.kube, options2, h.cloud.inner if hasattr(h.cloud, "inner") else h.cloud)
        assert "DriftController" not in [type(c).__name__ for c in mgr2.controllers]

    run(main())


def test_is_drifted_node_requirements_mutated():
    """RequirementsDrift: a node label mutated out from under the claim's
    requirements; keys absent from the node are NOT judged (platform may
    simply not stamp them)."""
    from gpu_provisioner_amd.cloudprovider.azure import DRIFT_REQUIREMENTS

    async def main():
        h = Harness().add_all_controllers(gc_interval=60.0, with_drift=False)
        await h.start()
        try:
            nc = await provisioned(h, "reqd1")
            node_name = nc["status"]["nodeName"]
            node = await h.kube.get("v1", "Node", node_name)
            # out-of-band: someone rewrote the instance-type label
            await h.kube.patch(
                "v1", "Node", node_name,
                {"metadata": {"labels": {
                    **node["metadata"]["labels"],
                    karpv1.INSTANCE_TYPE_LABEL_KEY: "Standard_D4s_v5",
                }}},
            )
            # drift reads through the informer cache: wait for the watch
            # to deliver the mutation (production sweeps every 2 min)
            async def informer_caught_up():
                cached = h.nodes.get(node_name)
                return (
                    cached
                    and ko.labels_of(cached).get(karpv1.INSTANCE_TYPE_LABEL_KEY)
                    == "Standard_D4s_v5"
                ) or None

            await h.wait_for(informer_caught_up)
            assert await h.cloud.is_drifted(nc) == DRIFT_REQUIREMENTS
            # a requirement on a key the node does not carry is not drift
            nc2 = await provisioned(h, "reqd2")
            nc2["spec"]["requirements"].append(
                {"key": "topology.kubernetes.io/zone", "operator": "In", "values": ["eastus2-1"]}
            )
            assert await h.cloud.is_drifted(nc2) == ""
        finally:
            await h.stop()

    run(main())
