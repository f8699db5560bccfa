"""Requirements algebra (reference vendor/.../pkg/scheduling/), MI355X
catalog/offerings model, metrics decorator, and event recorder dedupe."""
import asyncio

import pytest

from gpu_provisioner_amd.apis import v1 as karpv1
from gpu_provisioner_amd.events.recorder import EventRecorder
from gpu_provisioner_amd.fake.apiserver import InMemoryAPIServer, InMemoryClient
from gpu_provisioner_amd.kube import objects as ko
from gpu_provisioner_amd.providers.instancetype.catalog import InstanceTypeProvider
from gpu_provisioner_amd.scheduling.requirements import (
    Requirement,
    Requirements,
)
from tests.conftest import run

# ------------------------------------------------------------- requirements


def test_requirement_operators():
    r_in = Requirement("k", "In", ["a", "b"])
    assert r_in.has("a") and not r_in.has("c")
    r_notin = Requirement("k", "NotIn", ["a"])
    assert not r_notin.has("a") and r_notin.has("z")
    r_exists = Requirement("k", "Exists")
    assert r_exists.has("anything")
    r_dne = Requirement("k", "DoesNotExist")
    assert not r_dne.has("anything") and r_dne.is_empty()
    r_gt = Requirement("k", "Gt", ["5"])
    assert r_gt.has("6") and not r_gt.has("5") and not r_gt.has("x")
    r_lt = Requirement("k", "Lt", ["5"])
    assert r_lt.has("4") and not r_lt.has("5")


def test_requirement_intersection():
    a = Requirement("k", "In", ["a", "b", "c"])
    b = Requirement("k", "In", ["b", "c", "d"])
    assert sorted(a.intersect(b).values) == ["b", "c"]
    c = Requirement("k", "NotIn", ["b"])
    assert sorted(a.intersect(c).values) == ["a", "c"]
    d = Requirement("k", "Exists")
    assert sorted(a.intersect(d).values) == ["a", "b", "c"]
    # numeric window
    gt = Requirement("k", "Gt", ["2"])
    lt = Requirement("k", "Lt", ["10"])
    window = gt.intersect(lt)
    assert window.has("5") and not window.has("2") and not window.has("10")
    nums = Requirement("k", "In", ["1", "5", "20"])
    assert sorted(nums.intersect(window).values) == ["5"]
    # operator round-trip through dict form
    assert Requirement.from_dict(a.to_dict()).has("a")
    assert a.to_dict()["operator"] == "In"


def test_requirements_from_nodeclaim_and_compat():
    nc = karpv1.new_nodeclaim("x", labels={"zone": "1"})
    nc["spec"]["requirements"] = [
        {"key": karpv1.INSTANCE_TYPE_LABEL_KEY, "operator": "In",
         "values": ["Standard_ND128isr_MI355X_v6"]},
        {"key": "amd.com/gpu.count", "operator": "Gt", "values": ["4"]},
        {"key": "forbidden", "operator": "DoesNotExist"},
    ]
    reqs = Requirements.from_nodeclaim(nc)
    assert reqs.values_of(karpv1.INSTANCE_TYPE_LABEL_KEY) == ["Standard_ND128isr_MI355X_v6"]
    assert reqs.compatible(
        {"zone": "1", karpv1.INSTANCE_TYPE_LABEL_KEY: "Standard_ND128isr_MI355X_v6",
         "amd.com/gpu.count": "8"}
    )
    assert not reqs.compatible(
        {"zone": "2", karpv1.INSTANCE_TYPE_LABEL_KEY: "Standard_ND128isr_MI355X_v6",
         "amd.com/gpu.count": "8"}
    )
    assert not reqs.compatible(
        {"zone": "1", karpv1.INSTANCE_TYPE_LABEL_KEY: "Standard_ND128isr_MI355X_v6",
         "amd.com/gpu.count": "2"}
    )
    assert not reqs.compatible(
        {"zone": "1", karpv1.INSTANCE_TYPE_LABEL_KEY: "Standard_ND128isr_MI355X_v6",
         "amd.com/gpu.count": "8", "forbidden": "yes"}
    )
    # single-valued labels rendering
    labels = reqs.labels()
    assert labels["zone"] == "1"
    assert labels[karpv1.INSTANCE_TYPE_LABEL_KEY] == "Standard_ND128isr_MI355X_v6"
    assert "amd.com/gpu.count" not in labels


def test_requirements_add_intersects_same_key():
    reqs = Requirements()
    reqs.add(Requirement("k", "In", ["a", "b"]))
    reqs.add(Requirement("k", "NotIn", ["a"]))
    assert reqs.values_of("k") == ["b"]


# ------------------------------------------------------------------ catalog


def test_catalog_mi355x_topology():
    cat = InstanceTypeProvider("westus3")
    it = cat.get("Standard_ND128isr_MI355X_v6")
    assert it is not None
    assert it.capacity[karpv1.AMD_GPU_RESOURCE] == "8"
    assert it.capacity["memory"] == "2048Gi"
    assert it.requirements[karpv1.AMD_GPU_VRAM_LABEL_KEY] == "288G"
    assert it.requirements["amd.com/compute-arch"] == "gfx950"
    assert it.requirements[karpv1.XGMI_TOPOLOGY_LABEL_KEY] == "8x-7l-153g"
    # allocatable precompute: capacity - overhead
    alloc = it.allocatable()
    assert ko.qty(alloc["cpu"]) == ko.qty(it.capacity["cpu"]) - ko.qty("240m")
    assert ko.qty(alloc["memory"]) == ko.qty("2036Gi")
    assert alloc[karpv1.AMD_GPU_RESOURCE] == "8"  # GPUs have no overhead
    # zoned offerings, spot cheaper than on-demand
    zones = {o.zone for o in it.offerings}
    assert zones == {"westus3-1", "westus3-2", "westus3-3"}
    spot = it.cheapest_offering("spot")
    od = it.cheapest_offering("on-demand")
    assert spot.price < od.price


def test_catalog_gpu_detection():
    cat = InstanceTypeProvider()
    assert cat.is_gpu_sku("Standard_ND128isr_MI355X_v6")
    assert cat.is_gpu_sku("Standard_ND96isr_MI300X_v5")
    assert cat.gpu_count("Standard_ND64is_MI355X_v6") == 4
    assert not cat.is_gpu_sku("Standard_D4s_v5")
    assert cat.gpu_count("Standard_D4s_v5") == 0
    # AMD-family fallback for SKUs outside the catalog
    assert cat.is_gpu_sku("Standard_ND256is_MI455X_v7")
    # NVIDIA ND/NC SKUs outside the catalog must NOT be treated as AMD GPU
    # SKUs — a bare Standard_ND prefix match would stamp a ROCm gpuProfile
    # on an ND A100/H100 pool
    assert not cat.is_gpu_sku("Standard_ND96asr_v4")  # NVIDIA A100
    assert not cat.is_gpu_sku("Standard_ND96isr_H100_v5")  # NVIDIA H100
    assert not cat.is_gpu_sku("Standard_NC24ads_A100_v4")
    assert cat.gpu_count("Standard_ND96asr_v4") == 0


def test_mi300x_previous_generation_in_catalog():
    cat = InstanceTypeProvider()
    it = cat.get("Standard_ND96isr_MI300X_v5")
    assert it.requirements[karpv1.AMD_GPU_PRODUCT_LABEL_KEY] == "AMD-Instinct-MI300X"
    assert it.requirements[karpv1.AMD_GPU_VRAM_LABEL_KEY] == "192G"
    assert it.requirements["amd.com/compute-arch"] == "gfx942"


# ------------------------------------------------------- metrics decorator


def test_metrics_decorator_counts_errors_and_duration():
    from gpu_provisioner_amd.cloudprovider.decorator import MetricsDecorator, current_controller
    from gpu_provisioner_amd.cloudprovider.types import (
        CloudProvider,
        NodeClaimNotFoundError,
    )
    from gpu_provisioner_amd.metrics.registry import CLOUDPROVIDER_ERRORS

    class Boom(CloudProvider):
        async def create(self, nc):
            raise NodeClaimNotFoundError("x")

        async def delete(self, nc): ...
        async def get(self, pid): return {}
        async def list(self): return []
        async def get_instance_types(self, np=None): return []
        async def is_drifted(self, nc): return ""
        def repair_policies(self): return []
        def name(self): return "test-provider"
        def get_supported_node_classes(self): return []

    async def main():
        current_controller.set("test.controller")
        wrapped = MetricsDecorator(Boom())
        with pytest.raises(NodeClaimNotFoundError):
            await wrapped.create({})
        count = CLOUDPROVIDER_ERRORS.labels(
            controller="test.controller",
            method="Create",
            provider="test-provider",
            error_type="NodeClaimNotFoundError",
        )._value.get()
        assert count >= 1
        assert await wrapped.list() == []

    run(main())


# ----------------------------------------------------------- event recorder


def test_event_recorder_dedupes():
    async def main():
        kube = InMemoryClient(InMemoryAPIServer())
        rec = EventRecorder(kube)
        obj = {"apiVersion": "karpenter.sh/v1", "kind": "NodeClaim",
               "metadata": {"name": "a", "uid": "u1"}}
        for _ in range(5):
            rec.publish(obj, "Launched", "instance launched")
        rec.publish(obj, "Registered", "node registered")
        await asyncio.sleep(0.05)
        events = await kube.list("v1", "Event", namespace="default")
        reasons = sorted(e["reason"] for e in events)
        assert reasons == ["Launched", "Registered"]  # dedupe collapsed 5→1

    run(main())


def test_catalog_sku_file_override(tmp_path):
    """GPU_PROV_SKU_FILE-style override: add a SKU, replace a built-in,
    remove a built-in — deployments track real SKU lists without code
    changes (the built-in MI355X names are acknowledged extrapolations)."""
    import yaml

    f = tmp_path / "skus.yaml"
    f.write_text(yaml.safe_dump([
        {"name": "Standard_ND96isr_MI400X_v7", "vcpu": 96, "memory_gib": 2300,
         "gpus": 8, "max_os_disk_gib": 4096, "price": 90.0,
         "vram_gb": 432, "arch": "gfx1000x"},
        {"name": "Standard_ND128isr_MI355X_v6", "vcpu": 128, "memory_gib": 2048,
         "gpus": 8, "max_os_disk_gib": 4096, "price": 60.0},  # price corrected
        {"name": "Standard_ND32is_MI355X_v6", "remove": True},
    ]))
    cat = InstanceTypeProvider(sku_file=str(f))
    assert cat.get("Standard_ND96isr_MI400X_v7") is not None
    assert cat.gpu_count("Standard_ND96isr_MI400X_v7") == 8
    assert cat.get("Standard_ND32is_MI355X_v6") is None
    flagship = cat.get("Standard_ND128isr_MI355X_v6")
    assert min(o.price for o in flagship.offerings if o.capacity_type == "on-demand") == 60.0
    # malformed entries fail loudly
    bad = tmp_path / "bad.yaml"
    bad.write_text(yaml.safe_dump([{"name": "X", "vcpu": 1}]))
    with pytest.raises(ValueError, match="missing fields"):
        InstanceTypeProvider(sku_file=str(bad))


def test_shipped_sku_override_example_loads():
    """examples/sku-catalog-override.yaml must keep loading through the
    real GPU_PROV_SKU_FILE code path (price override, next-gen add,
    removal)."""
    import os

    path = os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "examples", "sku-catalog-override.yaml",
    )
    cat = InstanceTypeProvider(sku_file=path)
    assert cat.get("Standard_ND96isr_MI400X_v7") is not None
    assert cat.gpu_count("Standard_ND96isr_MI400X_v7") == 8
    assert cat.get("Standard_ND32is_MI355X_v6") is None


def test_event_recorder_dedupe_window_capped():
    """The dedupe window is bounded (r02): high-churn fleets key entries by
    object uid, and an uncapped rate x TTL window measured tens of MB."""
    from gpu_provisioner_amd.events import recorder as rec_mod

    async def main():
        server = InMemoryAPIServer()
        kube = InMemoryClient(server)
        r = rec_mod.EventRecorder(kube)
        for i in range(rec_mod.DEDUPE_MAX_ENTRIES + 500):
            obj = {"apiVersion": "v1", "kind": "Node",
                   "metadata": {"name": f"n{i}", "uid": f"u{i}"}}
            r.publish(obj, "Reason", "msg")
        assert len(r._seen) == rec_mod.DEDUPE_MAX_ENTRIES
        # dedupe still works for a key inside the window
        before = len(r._pending)
        r.publish({"apiVersion": "v1", "kind": "Node",
                   "metadata": {"name": "nlast", "uid": f"u{rec_mod.DEDUPE_MAX_ENTRIES+499}"}},
                  "Reason", "msg")
        # same (uid, reason, type) as the most recent publish → deduped
        assert len(r._seen) == rec_mod.DEDUPE_MAX_ENTRIES
        await asyncio.gather(*list(r._pending), return_exceptions=True)

    run(main())
