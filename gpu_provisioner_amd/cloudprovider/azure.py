"""Azure CloudProvider adapter: karpenter contract over the instance provider.

Behavioral spec: reference pkg/cloudprovider/cloudprovider.go — Create :51-61,
List :63-74, Get :76-87, Delete-by-NodeClaim-name :89-92, IsDrifted(empty)
:94-97, RepairPolicies (NodeReady False/Unknown tolerated 10 min) :103-116,
Name()="azure" :119-121, GetSupportedNodeClasses :123-125, and
instanceToNodeClaim :127-173 including deleting-state detection. Differences:
GetInstanceTypes returns the real MI355X catalog instead of an empty list
(reference stub :99-101).
"""
from __future__ import annotations

from typing import Optional

from ..apis import v1 as karpv1
from ..apis import v1alpha1
from ..kube import objects as ko
from ..providers.instance import bootstrap
from ..providers.instance.provider import InstanceProvider
from ..providers.instancetype.catalog import InstanceTypeProvider
from .types import CloudProvider, Instance, RepairPolicy

NODE_REPAIR_TOLERATION_SECONDS = 600.0  # 10 min
# GPU-sick nodes (nodeagent's AMDGPUHealthy=False: failed HBM/MFMA/LDS/xGMI
# self-tests) are repaired faster than NodeReady flaps: the kubelet is fine,
# the evidence is direct, and a degraded 8x MI355X host is expensive to keep
GPU_REPAIR_TOLERATION_SECONDS = 300.0  # 5 min

# drift reasons (net-new: the reference's IsDrifted is a stub returning "")
DRIFT_INSTANCE_TYPE = "InstanceTypeDrift"
DRIFT_NODE_IMAGE = "NodeImageDrift"
DRIFT_SKU_RETIRED = "SKURetiredDrift"
DRIFT_REQUIREMENTS = "RequirementsDrift"


class AzureCloudProvider(CloudProvider):
    def __init__(
        self,
        instances: InstanceProvider,
        catalog: InstanceTypeProvider,
        *,
        repair_toleration: float = NODE_REPAIR_TOLERATION_SECONDS,
        gpu_repair_toleration: float = GPU_REPAIR_TOLERATION_SECONDS,
    ):
        self.instances = instances
        self.catalog = catalog
        # drift-sweep instance view: ONE paged agent-pool LIST per sweep
        # instead of an ARM GET per claim — a 10k-claim fleet would
        # otherwise issue ~83 GETs/s against ARM every 2 min. The drift
        # controller invalidates at the start of each sweep; the TTL is a
        # staleness backstop for ad-hoc is_drifted calls.
        self._drift_cache: tuple = (0.0, None)  # (fetched_at_monotonic, {pool: Instance})
        self.drift_cache_ttl = 30.0
        self._repair_policies = [
            RepairPolicy("Ready", ko.CONDITION_FALSE, repair_toleration),
            RepairPolicy("Ready", ko.CONDITION_UNKNOWN, repair_toleration),
            # net-new: on-node GPU evidence from the mi355x-nodeagent
            RepairPolicy(
                karpv1.AMD_GPU_HEALTHY_CONDITION_TYPE,
                ko.CONDITION_FALSE,
                gpu_repair_toleration,
            ),
        ]

    async def create(self, nodeclaim: dict) -> dict:
        instance = await self.instances.create(nodeclaim)
        return self.instance_to_nodeclaim(instance)

    async def delete(self, nodeclaim: dict) -> None:
        # agent pools have no per-VM handle: deletion is by pool name, which
        # equals the NodeClaim name (reference cloudprovider.go:89-92)
        await self.instances.delete(ko.name_of(nodeclaim))

    async def get(self, provider_id: str) -> dict:
        instance = await self.instances.get(provider_id)
        return self.instance_to_nodeclaim(instance)

    async def list(self) -> list:
        return [self.instance_to_nodeclaim(i) for i in await self.instances.list()]

    async def get_instance_types(self, nodepool: Optional[dict] = None) -> list:
        return self.catalog.list()

    async def is_drifted(self, nodeclaim: dict) -> str:
        """Compare the live agent pool against the NodeClaim's declared shape.

        Net improvement over the reference (stub, cloudprovider.go:94-97):
          * InstanceTypeDrift — the pool's VM size is no longer among the
            NodeClaim's node.kubernetes.io/instance-type requirement values;
          * NodeImageDrift — the pool's osSKU disagrees with what the
            kaito.sh/node-image-family annotation selects today;
          * SKURetiredDrift — the pool runs an AMD GPU SKU that has been
            retired from the MI355X catalog (no longer orderable).
        A vanished instance is NOT drift — that is the GC controllers' domain.
        """
        pid = karpv1.provider_id_of(nodeclaim)
        if not pid:
            return ""
        instance = await self._instance_for_drift(nodeclaim, pid)
        if instance is None:
            return ""  # vanished instance is the GC controllers' domain
        wanted = karpv1.requirement_values(nodeclaim, karpv1.INSTANCE_TYPE_LABEL_KEY)
        if wanted and instance.type and instance.type not in wanted:
            return DRIFT_INSTANCE_TYPE
        want_sku = bootstrap.determine_os_sku(
            ko.annotations_of(nodeclaim).get(karpv1.NODE_IMAGE_FAMILY_ANNOTATION_KEY, "")
        )
        if instance.os_sku and instance.os_sku != want_sku:
            return DRIFT_NODE_IMAGE
        if (
            instance.type
            and self.catalog.get(instance.type) is None
            and self.catalog.is_gpu_sku(instance.type)
        ):
            return DRIFT_SKU_RETIRED
        # RequirementsDrift: the backing Node's labels mutated out from
        # under the claim's requirements (upstream karpenter's
        # NodeRequirementDrift). Only keys PRESENT on the node are judged —
        # absence may just mean the platform doesn't stamp that label.
        node = await self._node_for(pid)
        if node is not None:
            from ..scheduling.requirements import Requirements

            labels = ko.labels_of(node)
            for req in Requirements.from_nodeclaim(nodeclaim):
                if req.key in labels and not req.has(labels[req.key]):
                    return DRIFT_REQUIREMENTS
        return ""

    def invalidate_drift_cache(self) -> None:
        """Called by the drift controller at the start of every sweep so
        each sweep judges a fresh cloud snapshot."""
        self._drift_cache = (0.0, None)

    async def _instance_for_drift(self, nodeclaim: dict, provider_id: str):
        """Instance view for the drift sweep, served from a short-TTL
        snapshot of the paged agent-pool LIST (one ARM call per window,
        O(1) per claim)."""
        import time as _time

        from ..utils.utils import parse_agent_pool_name_from_id

        now = _time.monotonic()
        fetched_at, cache = self._drift_cache
        if cache is None or now - fetched_at > self.drift_cache_ttl:
            cache = {i.name: i for i in await self.instances.list()}
            self._drift_cache = (now, cache)
        pool = parse_agent_pool_name_from_id(provider_id) or ko.name_of(nodeclaim)
        return cache.get(pool)

    async def _node_for(self, provider_id: str):
        inf = getattr(self.instances, "nodes_informer", None)
        if inf is not None and inf.has_synced and inf.has_index("providerID"):
            nodes = inf.by_index("providerID", provider_id)
            return nodes[0] if nodes else None
        for node in await self.instances.kube.list("v1", "Node"):
            if ko.provider_id_of(node) == provider_id:
                return node
        return None

    def repair_policies(self) -> list:
        return self._repair_policies

    def name(self) -> str:
        return "azure"

    def get_supported_node_classes(self) -> list:
        return [(v1alpha1.GROUP, v1alpha1.KIND_KAITONODECLASS)]

    # -- conversion (reference cloudprovider.go:127-173) ---------------------

    def instance_to_nodeclaim(self, instance: Instance) -> dict:
        labels = dict(instance.labels)
        if instance.type:
            labels[karpv1.INSTANCE_TYPE_LABEL_KEY] = instance.type
        labels[karpv1.CAPACITY_TYPE_LABEL_KEY] = instance.capacity_type
        nodeclaim: dict = {
            "apiVersion": karpv1.API_VERSION,
            "kind": karpv1.KIND_NODECLAIM,
            "metadata": {"name": instance.name, "labels": labels},
            "spec": {},
            "status": {"providerID": instance.id, "imageID": instance.image_id},
        }
        it = self.catalog.get(instance.type)
        if it is not None:
            nodeclaim["status"]["capacity"] = dict(it.capacity)
            nodeclaim["status"]["allocatable"] = it.allocatable()
        if "deleting" in (instance.state or "").lower():
            # surfacing in-flight deletion lets GC skip pools that are already
            # going away (reference cloudprovider.go:155-166)
            ko.meta(nodeclaim)["deletionTimestamp"] = ko.fmt_time(ko.now())
        return nodeclaim
