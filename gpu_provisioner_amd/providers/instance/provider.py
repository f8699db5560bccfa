"""Instance provider: NodeClaim → single-VM AKS agent pool (MI355X).

The Azure-specific core, re-designed from the behavioral spec of reference
pkg/providers/instance/instance.go (Create :76-151, Get :153-165, List
:167-176, Delete :178-187, newAgentPoolObject :321-369, node lookup :371-385)
and armutils.go (create :28-40, delete-with-state-check :42-76). Differences
by design:
  * GPU knowledge comes from the MI355X catalog (providers/instancetype)
    instead of a `Standard_N` prefix match;
  * agent pools carry the ROCm bootstrap (gpuProfile, kubelet/sysctl, GPU
    topology node labels — providers/instance/bootstrap.py);
  * capacity (amd.com/gpu) is returned on the Instance so launch can
    pre-populate NodeClaim status.capacity before the node exists.
"""
from __future__ import annotations

import asyncio
import logging
import random
import re
import time
from typing import Optional

from ...apis import v1 as karpv1
from ...cloudprovider.types import (
    CreateError,
    Instance,
    InsufficientCapacityError,
    NodeClaimNotFoundError,
)
from ...kube import objects as ko
from ...kube.client import KubeClient
from ...utils.utils import parse_agent_pool_name_from_id
from ..instancetype.catalog import InstanceTypeProvider
from . import bootstrap
from .armapi import (
    AgentPoolsAPI,
    ARMError,
    is_arm_not_found,
    is_create_in_progress,
    pool_labels,
    pool_name,
    pool_state,
    pool_vm_size,
    taint_to_string,
)

log = logging.getLogger(__name__)

# agent-pool naming constraint (reference instance.go:50,80-84)
AGENT_POOL_NAME_RE = re.compile(r"^[a-z][a-z0-9]{0,11}$")

# GC bookkeeping label persisted on the agent pool + node so leak adoption
# survives controller restarts (reference instance.go:340-342)
CREATION_TIMESTAMP_LABEL = "kaito.sh/creation-timestamp"

NODE_WAIT_ATTEMPTS = 30
NODE_WAIT_INTERVAL = 1.0
NODE_WAIT_JITTER = 0.1

_ARM_THROTTLE_CODES = {"TooManyRequests", "SubscriptionRequestsThrottled"}
_CAPACITY_CODES = {
    "SkuNotAvailable",
    "AllocationFailed",
    "OverconstrainedAllocationRequest",
    "ZonalAllocationFailed",
    "QuotaExceeded",
    "OperationNotAllowed.QuotaExceeded",
}


class InstanceProvider:
    def __init__(
        self,
        agent_pools: AgentPoolsAPI,
        kube: KubeClient,
        catalog: InstanceTypeProvider,
        resource_group: str,
        cluster_name: str,
        *,
        node_wait_attempts: int = NODE_WAIT_ATTEMPTS,
        node_wait_interval: float = NODE_WAIT_INTERVAL,
        arm_profile=None,  # armschema.ArmApiProfile; None → ARM_API_PROFILE env / stable
    ):
        from .armschema import profile_from_env

        self.agent_pools = agent_pools
        self.kube = kube
        self.catalog = catalog
        self.resource_group = resource_group
        self.cluster_name = cluster_name
        self.node_wait_attempts = node_wait_attempts
        self.node_wait_interval = node_wait_interval
        self.arm_profile = arm_profile if arm_profile is not None else profile_from_env()
        # optional Node informer for cache-backed lookups (the reference
        # reads through controller-runtime's cached client; an apiserver
        # LIST per wait attempt is O(cluster) and shows at 128 concurrent)
        self.nodes_informer = None

    def set_nodes_informer(self, informer) -> None:
        informer.add_index(
            "agentpool",
            lambda o: [
                v
                for v in (
                    (o.get("metadata", {}).get("labels") or {}).get(karpv1.AGENTPOOL_LABEL_KEY),
                    (o.get("metadata", {}).get("labels") or {}).get(
                        karpv1.AZURE_AGENTPOOL_LABEL_KEY
                    ),
                )
                if v
            ]
            or None,
        )
        self.nodes_informer = informer

    # ------------------------------------------------------------------ create

    async def create(self, nodeclaim: dict) -> Instance:
        name = ko.name_of(nodeclaim)
        if not AGENT_POOL_NAME_RE.match(name):
            raise CreateError(
                f"agent pool name {name!r} must match {AGENT_POOL_NAME_RE.pattern}",
                condition_reason="InvalidName",
            )
        vm_size = self._pick_vm_size(nodeclaim)
        from .armschema import SchemaViolation

        try:
            pool = self.new_agent_pool_object(nodeclaim, vm_size)
        except SchemaViolation as e:
            raise CreateError(str(e), condition_reason="InvalidRequest") from e
        try:
            poller = await self.agent_pools.begin_create_or_update(
                self.resource_group, self.cluster_name, name, pool
            )
            created = await poller.result()
        except ARMError as e:
            if is_create_in_progress(e):
                # an earlier create (before a crash) is still running — adopt it
                log.info("agent pool %s create already in progress; adopting", name)
                created = await self.agent_pools.get(self.resource_group, self.cluster_name, name)
            elif e.code in _CAPACITY_CODES:
                raise InsufficientCapacityError(f"{vm_size}: {e.message or e.code}") from e
            else:
                raise CreateError(f"creating agent pool {name}: {e}") from e
        provider_id, node = await self._wait_for_node(name)
        if not provider_id:
            raise CreateError(
                f"node for agent pool {name} did not register a providerID within "
                f"{self.node_wait_attempts * self.node_wait_interval:.0f}s",
                condition_reason="NodeRegistrationTimeout",
            )
        return self._to_instance(created, provider_id=provider_id)

    def _pick_vm_size(self, nodeclaim: dict) -> str:
        values = karpv1.requirement_values(nodeclaim, karpv1.INSTANCE_TYPE_LABEL_KEY)
        if not values:
            raise CreateError(
                f"NodeClaim {ko.name_of(nodeclaim)} has no "
                f"{karpv1.INSTANCE_TYPE_LABEL_KEY} requirement",
                condition_reason="NoInstanceType",
            )
        # karpenter semantics: among the allowed SKUs, pick the one with the
        # cheapest offering COMPATIBLE with the claim's zone / capacity-type
        # requirements (not just the globally cheapest); SKUs outside the
        # catalog fall back to the first allowed value
        from ...scheduling.requirements import Requirements

        reqs = Requirements.from_nodeclaim(nodeclaim)
        it_req = reqs.get(karpv1.INSTANCE_TYPE_LABEL_KEY)
        if it_req is not None and it_req.min_values is not None:
            # karpenter minValues semantics: the scheduler promised at least
            # N instance-type options; fewer orderable ones than that means
            # the request's flexibility contract cannot be met
            orderable = [v for v in values if self.catalog.get(v) is not None]
            if len(orderable) < it_req.min_values:
                raise InsufficientCapacityError(
                    f"instance-type requirement needs minValues="
                    f"{it_req.min_values} orderable types, only {len(orderable)} "
                    f"of {values} are in the catalog"
                )
        zone_req = reqs.get(karpv1.ZONE_LABEL_KEY)
        ct_req = reqs.get(karpv1.CAPACITY_TYPE_LABEL_KEY)

        def eligible_price(it) -> Optional[float]:
            prices = [
                o.price
                for o in it.offerings
                if o.available
                and (zone_req is None or zone_req.has(o.zone))
                and (ct_req is None or ct_req.has(o.capacity_type))
            ]
            return min(prices) if prices else None

        priced = []
        for v in values:
            it = self.catalog.get(v)
            if it is None:
                continue
            p = eligible_price(it)
            if p is not None:
                priced.append((p, v))
        if priced:
            return min(priced)[1]
        if any(self.catalog.get(v) is not None for v in values):
            # SKUs known but no offering satisfies the zone/capacity-type
            # requirements — surface as capacity exhaustion so the claim is
            # released for the owner to retry (launch error taxonomy)
            raise InsufficientCapacityError(
                f"no offering of {values} satisfies the zone/capacity-type requirements"
            )
        return values[0]

    def new_agent_pool_object(self, nodeclaim: dict, vm_size: str) -> dict:
        """Reference newAgentPoolObject (instance.go:321-369), MI355X-native."""
        labels = dict(ko.labels_of(nodeclaim))
        labels[karpv1.NODEPOOL_LABEL_KEY] = karpv1.KAITO_NODEPOOL_NAME
        labels[CREATION_TIMESTAMP_LABEL] = str(int(time.time() * 1000))  # unix ms
        labels.update(bootstrap.gpu_node_labels(vm_size, self.catalog))
        taints = [taint_to_string(t) for t in nodeclaim.get("spec", {}).get("taints") or []]
        capacity_type = (
            karpv1.requirement_values(nodeclaim, karpv1.CAPACITY_TYPE_LABEL_KEY)
            or [karpv1.CAPACITY_TYPE_ON_DEMAND]
        )[0]
        props: dict = {
            "count": 1,
            "vmSize": vm_size,
            "osType": "Linux",
            "osSKU": bootstrap.determine_os_sku(
                ko.annotations_of(nodeclaim).get(karpv1.NODE_IMAGE_FAMILY_ANNOTATION_KEY, "")
            ),
            "mode": "User",
            "nodeLabels": labels,
            "nodeTaints": taints,
            "tags": {"managed-by": "gpu-provisioner-amd"},
        }
        if capacity_type == karpv1.CAPACITY_TYPE_SPOT:
            props["scaleSetPriority"] = "Spot"
            props["scaleSetEvictionPolicy"] = "Delete"
        # zone requirement → AKS availabilityZones (zone values are
        # "<region>-<n>"; the pool API takes the bare zone numbers)
        zones = [
            z.rsplit("-", 1)[-1]
            for z in karpv1.requirement_values(nodeclaim, karpv1.ZONE_LABEL_KEY)
            if z.rsplit("-", 1)[-1].isdigit()
        ]
        if zones:
            props["availabilityZones"] = sorted(set(zones))
        disk = (
            nodeclaim.get("spec", {})
            .get("resources", {})
            .get("requests", {})
            .get("ephemeral-storage")
        )
        if disk:
            props["osDiskSizeGB"] = max(1, int(ko.qty(disk).value / 2**30))
        if self.catalog.is_gpu_sku(vm_size):
            gpus = self.catalog.gpu_count(vm_size)
            gpu_profile = bootstrap.rocm_gpu_profile(self.arm_profile)
            if gpu_profile is not None:
                props["gpuProfile"] = gpu_profile
            props["kubeletConfig"] = bootstrap.rocm_kubelet_config()
            props["linuxOSConfig"] = bootstrap.rocm_linux_os_config(gpus)
        pool = {"name": ko.name_of(nodeclaim), "properties": props}
        # contract check against the pinned api-version (armschema.py): a
        # field outside the schema must fail HERE, in our own tests, not be
        # silently dropped by ARM in production
        from .armschema import validate_agent_pool

        validate_agent_pool(pool, self.arm_profile)
        return pool

    async def _wait_for_node(self, pool: str) -> tuple:
        """Wait for the Node object + providerID (reference instance.go:123-149,
        getNodesByName :371-385 — lookup via the two agentpool labels).

        With a Node informer the wait is EVENT-driven: a temporary handler
        resolves as soon as a matching node (with providerID) lands in the
        cache — add_handler's cache replay covers nodes that already exist.
        Without one, fall back to the reference's poll loop."""
        inf = self.nodes_informer
        if inf is not None and inf.has_synced:
            def pred(event_type: str, obj):
                if event_type in ("DELETED", "ABSENT") or obj is None:
                    return None
                pid = ko.provider_id_of(obj)
                return (pid, obj) if pid else None

            try:
                # keyed on the agentpool index: O(1) dispatch per node event
                return await inf.wait_until(
                    pred,
                    index="agentpool",
                    value=pool,
                    timeout=self.node_wait_attempts * self.node_wait_interval,
                )
            except asyncio.TimeoutError:
                return "", None
        for _ in range(self.node_wait_attempts):
            node = await self._node_for_pool(pool)
            if node is not None:
                pid = ko.provider_id_of(node)
                if pid:
                    return pid, node
            await asyncio.sleep(self.node_wait_interval * (1 + random.uniform(0, NODE_WAIT_JITTER)))
        return "", None

    async def _node_for_pool(self, pool: str) -> Optional[dict]:
        inf = self.nodes_informer
        if inf is not None and inf.has_synced:
            nodes = inf.by_index("agentpool", pool)
            return nodes[0] if nodes else None
        for selector in (
            f"{karpv1.AGENTPOOL_LABEL_KEY}={pool}",
            f"{karpv1.AZURE_AGENTPOOL_LABEL_KEY}={pool}",
        ):
            nodes = await self.kube.list("v1", "Node", label_selector=selector)
            if nodes:
                return nodes[0]
        return None

    # --------------------------------------------------------------------- get

    async def get(self, provider_id: str) -> Instance:
        pool = parse_agent_pool_name_from_id(provider_id)
        if not pool:
            raise NodeClaimNotFoundError(f"cannot parse agent pool from providerID {provider_id!r}")
        try:
            ap = await self.agent_pools.get(self.resource_group, self.cluster_name, pool)
        except ARMError as e:
            if is_arm_not_found(e):
                raise NodeClaimNotFoundError(f"agent pool {pool} not found") from e
            raise
        return self._to_instance(ap, provider_id=provider_id)

    # -------------------------------------------------------------------- list

    async def list(self) -> list:
        """Agent pools that are kaito-owned and nodeclaim-created (reference
        instance.go:167-176, predicates :387-413)."""
        out = []
        async for ap in self.agent_pools.list(self.resource_group, self.cluster_name):
            labels = pool_labels(ap)
            if labels.get(karpv1.NODEPOOL_LABEL_KEY) != karpv1.KAITO_NODEPOOL_NAME:
                continue
            if CREATION_TIMESTAMP_LABEL not in labels:
                continue
            inst = self._to_instance(ap)
            node = await self._node_for_pool(pool_name(ap))
            if node is not None:
                inst.id = ko.provider_id_of(node)
            out.append(inst)
        return out

    # ------------------------------------------------------------------ delete

    async def delete(self, pool: str) -> None:
        """Skip if already Deleting; NotFound → NodeClaimNotFoundError
        (reference instance.go:178-187 + armutils.go:42-76)."""
        try:
            ap = await self.agent_pools.get(self.resource_group, self.cluster_name, pool)
        except ARMError as e:
            if is_arm_not_found(e):
                raise NodeClaimNotFoundError(f"agent pool {pool} not found") from e
            raise
        if pool_state(ap) == "Deleting":
            return
        try:
            poller = await self.agent_pools.begin_delete(
                self.resource_group, self.cluster_name, pool
            )
            await poller.result()
        except ARMError as e:
            if is_arm_not_found(e):
                raise NodeClaimNotFoundError(f"agent pool {pool} not found") from e
            raise

    # --------------------------------------------------------------- conversion

    def _to_instance(self, ap: dict, provider_id: str = "") -> Instance:
        props = ap.get("properties", {})
        labels = pool_labels(ap)
        return Instance(
            name=pool_name(ap),
            id=provider_id,
            type=pool_vm_size(ap),
            state=pool_state(ap),
            image_id=props.get("nodeImageVersion", ""),
            capacity_type=(
                karpv1.CAPACITY_TYPE_SPOT
                if props.get("scaleSetPriority") == "Spot"
                else karpv1.CAPACITY_TYPE_ON_DEMAND
            ),
            labels=dict(labels),
            tags=dict(props.get("tags") or {}),
            created_at=labels.get(CREATION_TIMESTAMP_LABEL, ""),
            os_sku=props.get("osSKU", ""),
        )
