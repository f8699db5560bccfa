"""Table-driven instance-provider unit tests, mirroring the reference's
pkg/providers/instance/instance_test.go (TestNewAgentPoolObject, TestGet,
TestFromAgentPoolToInstance, TestDelete, TestList, TestCreateSuccess/
Failure, TestDetermineOSSKU) and pkg/utils/utils_test.go providerID parsing."""
import pytest

from gpu_provisioner_amd.apis import v1 as karpv1
from gpu_provisioner_amd.cloudprovider.types import (
    CreateError,
    InsufficientCapacityError,
    NodeClaimNotFoundError,
)
from gpu_provisioner_amd.fake.agentpools import AKSSimulator, FakeAgentPools
from gpu_provisioner_amd.fake.apiserver import InMemoryAPIServer, InMemoryClient
from gpu_provisioner_amd.kube import objects as ko
from gpu_provisioner_amd.providers.instance import bootstrap
from gpu_provisioner_amd.providers.instance.armapi import (
    ARMError,
    taint_from_string,
    taint_to_string,
)
from gpu_provisioner_amd.providers.instance.provider import (
    CREATION_TIMESTAMP_LABEL,
    InstanceProvider,
)
from gpu_provisioner_amd.providers.instancetype.catalog import InstanceTypeProvider
from gpu_provisioner_amd.utils.utils import (
    build_provider_id,
    parse_agent_pool_name_from_id,
)
from tests.conftest import run

VM = "Standard_ND128isr_MI355X_v6"


def make_provider(**kw):
    server = InMemoryAPIServer()
    kube = InMemoryClient(server)
    pools = FakeAgentPools(**{k: v for k, v in kw.items() if k.endswith("latency")})
    catalog = InstanceTypeProvider()
    aks = AKSSimulator(kube, pools, gpu_count_for=catalog.gpu_count)
    provider = InstanceProvider(
        pools, kube, catalog, "rg", "cluster", node_wait_interval=0.01
    )
    return provider, pools, kube, aks


def nodeclaim(name="gpu1", vm=VM, **spec):
    nc = karpv1.new_nodeclaim(name, labels={karpv1.KAITO_WORKSPACE_LABEL_KEY: "ws"})
    nc["metadata"]["uid"] = "uid-" + name
    nc["spec"] = {
        "requirements": [
            {"key": karpv1.INSTANCE_TYPE_LABEL_KEY, "operator": "In", "values": [vm]}
        ],
        **spec,
    }
    return nc


# -------------------------------------------------------- providerID parsing


def test_parse_agent_pool_name_from_provider_id():
    pid = build_provider_id("sub", "MC_rg_c_loc", "gpu1", "35723984")
    assert pid.startswith("azure:///subscriptions/sub/")
    assert parse_agent_pool_name_from_id(pid) == "gpu1"
    assert parse_agent_pool_name_from_id("azure:///bogus") is None
    assert parse_agent_pool_name_from_id("") is None
    # VMSS name that doesn't follow aks-<pool>-<hash>-vmss
    bad = pid.replace("aks-gpu1-35723984-vmss", "custom-vmss-name")
    assert parse_agent_pool_name_from_id(bad) is None


# ---------------------------------------------------------- taint round-trip


def test_taint_string_round_trip():
    t = {"key": "amd.com/gpu", "value": "present", "effect": "NoSchedule"}
    s = taint_to_string(t)
    assert s == "amd.com/gpu=present:NoSchedule"
    assert taint_from_string(s) == t
    t2 = {"key": "dedicated", "effect": "NoExecute"}
    assert taint_from_string(taint_to_string(t2)) == t2


# ------------------------------------------------------- newAgentPoolObject


def test_new_agent_pool_object_shape():
    provider, *_ = make_provider()
    nc = nodeclaim(
        taints=[{"key": "sku", "value": "gpu", "effect": "NoSchedule"}],
        resources={"requests": {"ephemeral-storage": "512Gi", karpv1.AMD_GPU_RESOURCE: "8"}},
    )
    pool = provider.new_agent_pool_object(nc, VM)
    props = pool["properties"]
    assert pool["name"] == "gpu1"
    assert props["count"] == 1
    assert props["vmSize"] == VM
    assert props["osDiskSizeGB"] == 512
    assert props["osSKU"] == "Ubuntu"
    assert props["nodeTaints"] == ["sku=gpu:NoSchedule"]
    labels = props["nodeLabels"]
    assert labels[karpv1.NODEPOOL_LABEL_KEY] == "kaito"
    assert labels[karpv1.KAITO_WORKSPACE_LABEL_KEY] == "ws"
    assert CREATION_TIMESTAMP_LABEL in labels
    # MI355X bootstrap
    assert labels[karpv1.AMD_GPU_COUNT_LABEL_KEY] == "8"
    assert labels[karpv1.AMD_GPU_VRAM_LABEL_KEY] == "288G"
    assert labels[karpv1.XGMI_TOPOLOGY_LABEL_KEY].startswith("8x-7l-")
    # stable ARM profile (the default): no gpuProfile property — it is not
    # in the pinned 2024-09-01 agentPools schema (armschema.py); the ROCm
    # stack comes from the chart's DaemonSets
    assert "gpuProfile" not in props
    assert props["kubeletConfig"]["topologyManagerPolicy"] == "single-numa-node"


def test_new_agent_pool_object_cpu_sku_has_no_gpu_profile():
    provider, *_ = make_provider()
    nc = nodeclaim(vm="Standard_D4s_v5")
    pool = provider.new_agent_pool_object(nc, "Standard_D4s_v5")
    assert "gpuProfile" not in pool["properties"]
    assert karpv1.AMD_GPU_COUNT_LABEL_KEY not in pool["properties"]["nodeLabels"]


def test_new_agent_pool_object_nvidia_sku_gets_no_rocm_profile():
    """Out-of-catalog NVIDIA ND/NC SKUs must never receive the ROCm
    bootstrap (gpuProfile/kubeletConfig/linuxOSConfig) — VERDICT r01 #4."""
    provider, *_ = make_provider()
    for sku in ("Standard_ND96asr_v4", "Standard_ND96isr_H100_v5", "Standard_NC24ads_A100_v4"):
        pool = provider.new_agent_pool_object(nodeclaim(vm=sku), sku)
        props = pool["properties"]
        assert "gpuProfile" not in props, sku
        assert "kubeletConfig" not in props, sku
        assert karpv1.AMD_GPU_COUNT_LABEL_KEY not in props["nodeLabels"], sku


@pytest.mark.parametrize(
    "annotation,expected",
    [
        ("", "Ubuntu"),
        ("Ubuntu", "Ubuntu"),
        ("ubuntu2204", "Ubuntu"),
        ("AzureLinux", "AzureLinux"),
        ("azurelinux", "AzureLinux"),
        ("Mariner", "AzureLinux"),
        ("Windows2049", "Ubuntu"),  # unknown → default
    ],
)
def test_determine_os_sku(annotation, expected):
    assert bootstrap.determine_os_sku(annotation) == expected


def test_vm_size_picks_cheapest_known_sku():
    provider, *_ = make_provider()
    nc = nodeclaim()
    nc["spec"]["requirements"][0]["values"] = [
        "Standard_ND128isr_MI355X_v6",
        "Standard_ND32is_MI355X_v6",  # cheapest
        "Standard_ND64is_MI355X_v6",
    ]
    assert provider._pick_vm_size(nc) == "Standard_ND32is_MI355X_v6"


def test_vm_size_missing_requirement_rejected():
    provider, *_ = make_provider()
    nc = nodeclaim()
    nc["spec"]["requirements"] = []
    with pytest.raises(CreateError, match="instance-type"):
        provider._pick_vm_size(nc)


# ------------------------------------------------------------ create / get


def test_create_success_returns_instance_with_provider_id():
    async def main():
        provider, pools, kube, aks = make_provider()
        inst = await provider.create(nodeclaim())
        assert inst.name == "gpu1"
        assert inst.type == VM
        assert inst.state == "Succeeded"
        assert parse_agent_pool_name_from_id(inst.id) == "gpu1"
        assert inst.capacity_type == "on-demand"

    run(main())


def test_create_name_regex_rejected():
    async def main():
        provider, *_ = make_provider()
        for bad in ("Gpu1", "1gpu", "waytoolongofaname", "has-dash"):
            with pytest.raises(CreateError):
                await provider.create(nodeclaim(bad))

    run(main())


def test_create_maps_capacity_errors():
    async def main():
        provider, pools, *_ = make_provider()
        pools.create_error.set(ARMError(409, "QuotaExceeded", "quota"), max_calls=1)
        with pytest.raises(InsufficientCapacityError):
            await provider.create(nodeclaim())

    run(main())


def test_create_node_wait_timeout():
    async def main():
        provider, pools, kube, aks = make_provider()
        # break the simulator so no node ever appears
        pools.on_pool_ready = None
        provider.node_wait_attempts = 3
        with pytest.raises(CreateError, match="providerID"):
            await provider.create(nodeclaim())

    run(main())


def test_get_parses_pool_and_maps_not_found():
    async def main():
        provider, pools, kube, aks = make_provider()
        created = await provider.create(nodeclaim())
        inst = await provider.get(created.id)
        assert inst.name == "gpu1"
        with pytest.raises(NodeClaimNotFoundError):
            await provider.get(build_provider_id("s", "r", "nope", "beef"))
        with pytest.raises(NodeClaimNotFoundError):
            await provider.get("azure:///unparseable")

    run(main())


# --------------------------------------------------------------- list/delete


def test_list_filters_to_kaito_nodeclaim_pools():
    async def main():
        provider, pools, kube, aks = make_provider()
        await provider.create(nodeclaim("mine1"))
        # foreign pool (no kaito labels): must be invisible
        poller = await pools.begin_create_or_update(
            "rg", "cluster", "foreign", {"properties": {"vmSize": VM, "nodeLabels": {}}}
        )
        await poller.result()
        # kaito-labeled but no creation timestamp: also invisible
        poller = await pools.begin_create_or_update(
            "rg", "cluster", "half",
            {"properties": {"vmSize": VM,
                            "nodeLabels": {karpv1.NODEPOOL_LABEL_KEY: "kaito"}}},
        )
        await poller.result()
        instances = await provider.list()
        assert [i.name for i in instances] == ["mine1"]
        assert parse_agent_pool_name_from_id(instances[0].id) == "mine1"

    run(main())


def test_delete_skips_if_already_deleting_and_maps_not_found():
    async def main():
        provider, pools, kube, aks = make_provider()
        await provider.create(nodeclaim())
        pools.pools["gpu1"]["properties"]["provisioningState"] = "Deleting"
        deletes_before = pools.delete_calls
        await provider.delete("gpu1")  # skip: already deleting
        assert pools.delete_calls == deletes_before
        pools.pools["gpu1"]["properties"]["provisioningState"] = "Succeeded"
        await provider.delete("gpu1")
        assert pools.delete_calls == deletes_before + 1
        with pytest.raises(NodeClaimNotFoundError):
            await provider.delete("gpu1")

    run(main())


def test_instance_conversion_detects_deleting_state():
    from gpu_provisioner_amd.cloudprovider.azure import AzureCloudProvider

    provider, pools, kube, aks = make_provider()
    cloud = AzureCloudProvider(provider, provider.catalog)
    inst = provider._to_instance(
        {"name": "x", "properties": {"vmSize": VM, "provisioningState": "Deleting",
                                     "nodeLabels": {}}}
    )
    nc = cloud.instance_to_nodeclaim(inst)
    assert ko.is_deleting(nc)
    # allocatable precompute from the catalog
    assert nc["status"]["capacity"][karpv1.AMD_GPU_RESOURCE] == "8"
    assert ko.qty(nc["status"]["allocatable"]["cpu"]) < ko.qty(nc["status"]["capacity"]["cpu"])


def test_vm_size_respects_zone_and_capacity_type_requirements():
    """Offering eligibility: the cheapest SKU overall loses to one whose
    offerings actually satisfy the claim's zone/capacity-type requirements."""
    from gpu_provisioner_amd.cloudprovider.types import InsufficientCapacityError

    provider, _, _, _ = make_provider()
    nc = {
        "apiVersion": karpv1.API_VERSION,
        "kind": karpv1.KIND_NODECLAIM,
        "metadata": {"name": "zreq"},
        "spec": {
            "requirements": [
                {
                    "key": karpv1.INSTANCE_TYPE_LABEL_KEY,
                    "operator": "In",
                    "values": ["Standard_ND32is_MI355X_v6", "Standard_ND128isr_MI355X_v6"],
                },
                {"key": karpv1.ZONE_LABEL_KEY, "operator": "In", "values": ["eastus2-2"]},
            ]
        },
    }
    # both SKUs offer eastus2-2: cheapest (ND32) wins
    assert provider._pick_vm_size(nc) == "Standard_ND32is_MI355X_v6"

    # restrict to a zone no offering has: InsufficientCapacity, not a pick
    nc["spec"]["requirements"][1]["values"] = ["nowhere-9"]
    with pytest.raises(InsufficientCapacityError):
        provider._pick_vm_size(nc)

    # spot-only requirement still eligible (catalog offers spot everywhere)
    nc["spec"]["requirements"][1] = {
        "key": karpv1.CAPACITY_TYPE_LABEL_KEY,
        "operator": "In",
        "values": [karpv1.CAPACITY_TYPE_SPOT],
    }
    assert provider._pick_vm_size(nc) == "Standard_ND32is_MI355X_v6"


def test_zone_requirement_sets_availability_zones():
    provider, _, _, _ = make_provider()
    nc = nodeclaim("zone1")
    nc["spec"]["requirements"].append(
        {"key": karpv1.ZONE_LABEL_KEY, "operator": "In", "values": ["eastus2-1", "eastus2-3"]}
    )
    pool = provider.new_agent_pool_object(nc, VM)
    assert pool["properties"]["availabilityZones"] == ["1", "3"]
    # no zone requirement -> field absent (AKS default zone spread)
    pool2 = provider.new_agent_pool_object(nodeclaim("zone2"), VM)
    assert "availabilityZones" not in pool2["properties"]


def test_min_values_gates_instance_type_flexibility():
    """karpenter minValues: the claim demands >= N orderable instance-type
    options; fewer in the catalog -> InsufficientCapacity (claim released)."""
    from gpu_provisioner_amd.cloudprovider.types import InsufficientCapacityError

    provider, _, _, _ = make_provider()
    nc = nodeclaim("minv1")
    nc["spec"]["requirements"] = [
        {
            "key": karpv1.INSTANCE_TYPE_LABEL_KEY,
            "operator": "In",
            "values": [VM, "Standard_FAKE_SKU_v9"],
            "minValues": 2,
        }
    ]
    with pytest.raises(InsufficientCapacityError):
        provider._pick_vm_size(nc)
    # satisfied when enough orderable SKUs exist
    nc["spec"]["requirements"][0]["values"] = [VM, "Standard_ND64is_MI355X_v6"]
    assert provider._pick_vm_size(nc) in (VM, "Standard_ND64is_MI355X_v6")
