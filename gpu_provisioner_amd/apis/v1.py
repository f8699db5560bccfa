"""karpenter.sh/v1 NodeClaim API surface: constants, condition types, typed helpers.

Behavioral spec: reference vendor/sigs.k8s.io/karpenter/pkg/apis/v1/
(nodeclaim.go:27-157 spec, nodeclaim_status.go:26-79 status+conditions,
labels.go:42-61 label/annotation/finalizer registry). Objects themselves are
wire-format dicts (see kube/objects.py); this module holds the domain
vocabulary and NodeClaim-specific accessors.
"""
from __future__ import annotations

from typing import Optional

from ..kube import objects as ko

# -- group/version -----------------------------------------------------------

GROUP = "karpenter.sh"
API_VERSION = "karpenter.sh/v1"
KIND_NODECLAIM = "NodeClaim"

# -- labels / annotations / finalizers (labels.go:42-61) ---------------------

NODEPOOL_LABEL_KEY = "karpenter.sh/nodepool"
CAPACITY_TYPE_LABEL_KEY = "karpenter.sh/capacity-type"
NODE_INITIALIZED_LABEL_KEY = "karpenter.sh/initialized"
NODE_REGISTERED_LABEL_KEY = "karpenter.sh/registered"
TERMINATION_FINALIZER = "karpenter.sh/termination"
TERMINATION_TIMESTAMP_ANNOTATION_KEY = "karpenter.sh/nodeclaim-termination-timestamp"
DISRUPTED_TAINT_KEY = "karpenter.sh/disrupted"
UNREGISTERED_TAINT_KEY = "karpenter.sh/unregistered"
DO_NOT_DISRUPT_ANNOTATION_KEY = "karpenter.sh/do-not-disrupt"

CAPACITY_TYPE_ON_DEMAND = "on-demand"
CAPACITY_TYPE_SPOT = "spot"

# kaito management labels — the gate that scopes every watch/list
# (reference vendor/.../pkg/utils/nodeclaim/nodeclaim.go:40-42, the fork's patch)
KAITO_WORKSPACE_LABEL_KEY = "kaito.sh/workspace"
KAITO_RAGENGINE_LABEL_KEY = "kaito.sh/ragengine"
KAITO_NODEPOOL_NAME = "kaito"

# node image family annotation (reference pkg/providers/instance/instance.go:364,415-441)
NODE_IMAGE_FAMILY_ANNOTATION_KEY = "kaito.sh/node-image-family"

# well-known kube keys
INSTANCE_TYPE_LABEL_KEY = "node.kubernetes.io/instance-type"
ZONE_LABEL_KEY = "topology.kubernetes.io/zone"
ARCH_LABEL_KEY = "kubernetes.io/arch"
OS_LABEL_KEY = "kubernetes.io/os"
HOSTNAME_LABEL_KEY = "kubernetes.io/hostname"
AZURE_AGENTPOOL_LABEL_KEY = "kubernetes.azure.com/agentpool"
AGENTPOOL_LABEL_KEY = "agentpool"
EXCLUDE_FROM_LB_LABEL_KEY = "node.kubernetes.io/exclude-from-external-load-balancers"

# AMD GPU surface (net-new vs the reference's nvidia.com/gpu)
AMD_GPU_RESOURCE = "amd.com/gpu"
# Node condition published by the mi355x-nodeagent DaemonSet (HBM/MFMA/LDS/
# xGMI self-tests); the repair policies watch it like NodeReady
AMD_GPU_HEALTHY_CONDITION_TYPE = "AMDGPUHealthy"
AMD_GPU_PRODUCT_LABEL_KEY = "amd.com/gpu.product"
AMD_GPU_VRAM_LABEL_KEY = "amd.com/gpu.vram"
AMD_GPU_COUNT_LABEL_KEY = "amd.com/gpu.count"
XGMI_TOPOLOGY_LABEL_KEY = "amd.com/xgmi-topology"

# ephemeral startup taints tolerated during initialization
# (reference vendor/.../lifecycle/initialization.go + karpenter KnownEphemeralTaints)
KNOWN_EPHEMERAL_TAINTS = (
    {"key": "node.kubernetes.io/not-ready", "effect": "NoSchedule"},
    {"key": "node.kubernetes.io/not-ready", "effect": "NoExecute"},
    {"key": "node.kubernetes.io/unreachable", "effect": "NoSchedule"},
    {"key": "node.kubernetes.io/unreachable", "effect": "NoExecute"},
    {"key": "node.cloudprovider.kubernetes.io/uninitialized", "effect": "NoSchedule"},
    {"key": UNREGISTERED_TAINT_KEY, "effect": "NoExecute"},
)

# -- condition types (nodeclaim_status.go:26-35) ------------------------------

COND_LAUNCHED = "Launched"
COND_REGISTERED = "Registered"
COND_INITIALIZED = "Initialized"
COND_DRAINED = "Drained"
COND_VOLUMES_DETACHED = "VolumesDetached"
COND_INSTANCE_TERMINATING = "InstanceTerminating"
COND_DRIFTED = "Drifted"
COND_READY = "Ready"

# -- NodeClaim accessors ------------------------------------------------------


def new_nodeclaim(name: str, labels: Optional[dict] = None, **spec) -> dict:
    obj = {
        "apiVersion": API_VERSION,
        "kind": KIND_NODECLAIM,
        "metadata": {"name": name},
        "spec": dict(spec) if spec else {},
        "status": {},
    }
    if labels:
        obj["metadata"]["labels"] = dict(labels)
    return obj


def requirements_of(nodeclaim: dict) -> list:
    return nodeclaim.get("spec", {}).get("requirements") or []


def provider_id_of(nodeclaim: dict) -> str:
    return nodeclaim.get("status", {}).get("providerID", "")


def node_name_of(nodeclaim: dict) -> str:
    return nodeclaim.get("status", {}).get("nodeName", "")


def is_launched(nodeclaim: dict) -> bool:
    return ko.condition_is_true(nodeclaim, COND_LAUNCHED)


def is_registered(nodeclaim: dict) -> bool:
    return ko.condition_is_true(nodeclaim, COND_REGISTERED)


def is_initialized(nodeclaim: dict) -> bool:
    return ko.condition_is_true(nodeclaim, COND_INITIALIZED)


def termination_grace_period_of(nodeclaim: dict) -> Optional[str]:
    return nodeclaim.get("spec", {}).get("terminationGracePeriod")


def node_class_ref_of(nodeclaim: dict) -> dict:
    return nodeclaim.get("spec", {}).get("nodeClassRef") or {}


def is_managed(nodeclaim: dict) -> bool:
    """The kaito management gate: a NodeClaim is ours if its NodeClassRef
    GroupKind matches a supported NodeClass OR it carries a kaito.sh/workspace
    or kaito.sh/ragengine label (reference vendor/sigs.k8s.io/karpenter/pkg/
    utils/nodeclaim/nodeclaim.go:40-74 — the fork's defining patch)."""
    labels = ko.labels_of(nodeclaim)
    if KAITO_WORKSPACE_LABEL_KEY in labels or KAITO_RAGENGINE_LABEL_KEY in labels:
        return True
    ref = node_class_ref_of(nodeclaim)
    from .v1alpha1 import GROUP as KAITO_GROUP, KIND_KAITONODECLASS

    return ref.get("group") == KAITO_GROUP and ref.get("kind") == KIND_KAITONODECLASS


def node_is_managed(node: dict) -> bool:
    """A Node is managed if it carries the kaito labels (synced at
    registration) or the karpenter.sh/nodepool label."""
    labels = ko.labels_of(node)
    return (
        KAITO_WORKSPACE_LABEL_KEY in labels
        or KAITO_RAGENGINE_LABEL_KEY in labels
        or labels.get(NODEPOOL_LABEL_KEY) == KAITO_NODEPOOL_NAME
    )


def requirement_values(nodeclaim: dict, key: str) -> list:
    """Values of an In requirement with the given key (e.g. the VM size from
    node.kubernetes.io/instance-type — reference instance.go:90-95)."""
    for req in requirements_of(nodeclaim):
        if req.get("key") == key and req.get("operator", "In") == "In":
            return req.get("values") or []
    return []
