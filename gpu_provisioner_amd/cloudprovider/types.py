"""CloudProvider contract, typed error taxonomy, InstanceType/Offerings model.

Behavioral spec: reference vendor/sigs.k8s.io/karpenter/pkg/cloudprovider/
types.go — interface :72-100, InstanceType/Offerings with allocatable
precompute :102-219, error taxonomy :476-584. The controllers branch on these
error types (e.g. launch deletes the NodeClaim on InsufficientCapacity), so
the taxonomy is load-bearing.
"""
from __future__ import annotations

import abc
from dataclasses import dataclass, field
from typing import Optional

from ..kube.objects import qty


# ---------------------------------------------------------------------------
# Error taxonomy
# ---------------------------------------------------------------------------


class CloudProviderError(Exception):
    pass


class NodeClaimNotFoundError(CloudProviderError):
    """The instance backing a NodeClaim no longer exists (types.go:476-508)."""


class InsufficientCapacityError(CloudProviderError):
    """The cloud cannot satisfy the request right now (types.go:510-535).
    Launch reacts by deleting the NodeClaim so the workload owner retries."""


class NodeClassNotReadyError(CloudProviderError):
    """The referenced NodeClass is not ready (types.go:537-562)."""


class CreateError(CloudProviderError):
    """Create failed for a reason recorded on the Launched condition
    (types.go:564-584)."""

    def __init__(self, message: str, condition_reason: str = "LaunchFailed"):
        super().__init__(message)
        self.condition_reason = condition_reason


def is_nodeclaim_not_found(err: BaseException) -> bool:
    return isinstance(err, NodeClaimNotFoundError)


def is_insufficient_capacity(err: BaseException) -> bool:
    return isinstance(err, InsufficientCapacityError)


def error_type_of(err: BaseException) -> str:
    """Label value for the cloudprovider error counter (metrics decorator)."""
    for cls in (
        NodeClaimNotFoundError,
        InsufficientCapacityError,
        NodeClassNotReadyError,
        CreateError,
    ):
        if isinstance(err, cls):
            return cls.__name__
    return err.__class__.__name__


# ---------------------------------------------------------------------------
# InstanceType / Offerings model
# ---------------------------------------------------------------------------


@dataclass
class Offering:
    capacity_type: str  # on-demand | spot
    zone: str
    price: float
    available: bool = True


@dataclass
class InstanceType:
    """An orderable VM SKU with its resource envelope and topology metadata.

    The reference stubbed GetInstanceTypes to an empty list
    (pkg/cloudprovider/cloudprovider.go:99-101); here the MI355X catalog is a
    first-class model (providers/instancetype) with GPU/HBM/xGMI capacity, and
    allocatable = capacity - overhead precomputed as karpenter does
    (types.go:102-219).
    """

    name: str
    capacity: dict = field(default_factory=dict)  # resource -> Quantity-str
    overhead: dict = field(default_factory=dict)  # kube-reserved+system-reserved
    requirements: dict = field(default_factory=dict)  # label key -> value
    offerings: list = field(default_factory=list)

    def allocatable(self) -> dict:
        out = {}
        for res, cap in self.capacity.items():
            ovh = self.overhead.get(res)
            if ovh:
                out[res] = str(qty(cap) - qty(ovh))
            else:
                out[res] = str(cap)
        return out

    def cheapest_offering(self, capacity_type: str = "") -> Optional[Offering]:
        cands = [
            o
            for o in self.offerings
            if o.available and (not capacity_type or o.capacity_type == capacity_type)
        ]
        return min(cands, key=lambda o: o.price) if cands else None


# ---------------------------------------------------------------------------
# Instance — the cloud-side record of a launched machine
# (reference pkg/providers/instance/types.go:19-29)
# ---------------------------------------------------------------------------


@dataclass
class Instance:
    name: str  # agent-pool name
    id: str = ""  # providerID (azure:///...vmss/virtualMachines/0)
    type: str = ""  # VM size
    state: str = ""  # provisioning state: Creating/Succeeded/Deleting/Failed
    image_id: str = ""
    capacity_type: str = "on-demand"
    labels: dict = field(default_factory=dict)
    tags: dict = field(default_factory=dict)
    created_at: str = ""
    os_sku: str = ""  # pool osSKU (Ubuntu/AzureLinux) — drift detection input


@dataclass
class RepairPolicy:
    """A node condition the provider tolerates for `toleration_seconds` before
    repair (reference pkg/cloudprovider/cloudprovider.go:103-116)."""

    condition_type: str
    condition_status: str
    toleration_seconds: float


# ---------------------------------------------------------------------------
# CloudProvider interface (types.go:72-100)
# ---------------------------------------------------------------------------


class CloudProvider(abc.ABC):
    @abc.abstractmethod
    async def create(self, nodeclaim: dict) -> dict:
        """Launch an instance for the NodeClaim; returns a NodeClaim dict with
        status (providerID, imageID, capacity, allocatable) and instance labels
        populated."""

    @abc.abstractmethod
    async def delete(self, nodeclaim: dict) -> None:
        """Terminate the instance. Raises NodeClaimNotFoundError once gone."""

    @abc.abstractmethod
    async def get(self, provider_id: str) -> dict:
        """NodeClaim-shaped record for a providerID. Raises NodeClaimNotFoundError."""

    @abc.abstractmethod
    async def list(self) -> list:
        """All NodeClaim-shaped records this provider manages."""

    @abc.abstractmethod
    async def get_instance_types(self, nodepool: Optional[dict] = None) -> list:
        """Orderable InstanceTypes (the MI355X catalog)."""

    @abc.abstractmethod
    async def is_drifted(self, nodeclaim: dict) -> str:
        """Drift reason or ''. The reference stubs this to always-empty
        (cloudprovider.go:94-97); here it compares the live agent pool against
        the NodeClaim's declared shape (see azure.py)."""

    @abc.abstractmethod
    def repair_policies(self) -> list:
        """RepairPolicy list driving the node.health controller."""

    @abc.abstractmethod
    def name(self) -> str:
        ...

    @abc.abstractmethod
    def get_supported_node_classes(self) -> list:
        """[(group, kind)] of supported NodeClass GroupKinds."""
