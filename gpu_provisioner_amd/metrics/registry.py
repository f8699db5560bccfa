"""Prometheus metrics registry.

Keeps the reference's metric names (karpenter_* and controller-runtime-style
reconcile metrics) so dashboards/alerts written for the reference keep working
(reference vendor/sigs.k8s.io/karpenter/pkg/metrics/metrics.go:32-98,
vendor/.../pkg/cloudprovider/metrics/cloudprovider.go:48-104,
vendor/.../controllers/node/termination/metrics.go).
"""
from __future__ import annotations

from prometheus_client import REGISTRY, Counter, Gauge, Histogram

_BUCKETS_SHORT = (0.001, 0.005, 0.01, 0.05, 0.1, 0.5, 1, 5, 10, 30, 60)
_BUCKETS_LONG = (1, 5, 10, 30, 60, 120, 300, 600, 1200, 3600)


def _get_or_create(cls, name, doc, labelnames=(), **kw):
    """Idempotent metric creation (tests may import twice)."""
    try:
        return cls(name, doc, labelnames, **kw)
    except ValueError:
        collector = REGISTRY._names_to_collectors.get(name)
        if collector is None:
            raise
        return collector


# -- controller runtime ------------------------------------------------------

RECONCILE_DURATION = _get_or_create(
    Histogram,
    "controller_runtime_reconcile_time_seconds",
    "Length of time per reconciliation per controller",
    ("controller",),
    buckets=_BUCKETS_SHORT,
)
RECONCILE_ERRORS = _get_or_create(
    Counter,
    "controller_runtime_reconcile_errors_total",
    "Total number of reconciliation errors per controller",
    ("controller",),
)
RECONCILE_TOTAL = _get_or_create(
    Counter,
    "controller_runtime_reconcile_total",
    "Total number of reconciliations per controller",
    ("controller", "result"),
)
WORKQUEUE_DEPTH = _get_or_create(
    Gauge,
    "workqueue_depth",
    "Current depth of workqueue",
    ("controller",),
)

# -- karpenter nodeclaim/node lifecycle --------------------------------------

NODECLAIMS_CREATED = _get_or_create(
    Counter,
    "karpenter_nodeclaims_created_total",
    "Number of nodeclaims launched",
    ("nodepool", "capacity_type", "instance_type"),
)
NODECLAIMS_TERMINATED = _get_or_create(
    Counter,
    "karpenter_nodeclaims_terminated_total",
    "Number of nodeclaims terminated",
    ("nodepool", "capacity_type", "instance_type"),
)
NODECLAIMS_DISRUPTED = _get_or_create(
    Counter,
    "karpenter_nodeclaims_disrupted_total",
    "Number of nodeclaims disrupted",
    ("reason", "nodepool"),
)
NODECLAIMS_DRIFTED = _get_or_create(
    Counter,
    "karpenter_nodeclaims_drifted_total",
    "Number of nodeclaims detected as drifted from their spec",
    ("reason", "nodepool"),
)
NODES_CREATED = _get_or_create(
    Counter,
    "karpenter_nodes_created_total",
    "Number of nodes created",
    ("nodepool",),
)
NODES_TERMINATED = _get_or_create(
    Counter,
    "karpenter_nodes_terminated_total",
    "Number of nodes terminated",
    ("nodepool",),
)
NODES_DRAINED = _get_or_create(
    Counter,
    "karpenter_nodes_drained_total",
    "Number of nodes drained",
    ("nodepool",),
)
NODECLAIM_TERMINATION_DURATION = _get_or_create(
    Histogram,
    "karpenter_nodeclaims_termination_duration_seconds",
    "Duration of NodeClaim termination, from deletion to finalizer removal",
    ("nodepool",),
    buckets=_BUCKETS_LONG,
)
NODE_TERMINATION_DURATION = _get_or_create(
    Histogram,
    "karpenter_nodes_termination_duration_seconds",
    "Duration of Node termination, from deletion to finalizer removal",
    ("nodepool",),
    buckets=_BUCKETS_LONG,
)
LAUNCH_DURATION = _get_or_create(
    Histogram,
    "karpenter_nodeclaims_launch_duration_seconds",
    "Duration from NodeClaim creation to Launched condition",
    ("nodepool",),
    buckets=_BUCKETS_LONG,
)
REGISTRATION_DURATION = _get_or_create(
    Histogram,
    "karpenter_nodeclaims_registration_duration_seconds",
    "Duration from NodeClaim creation to Registered condition",
    ("nodepool",),
    buckets=_BUCKETS_LONG,
)
INITIALIZATION_DURATION = _get_or_create(
    Histogram,
    "karpenter_nodeclaims_initialization_duration_seconds",
    "Duration from NodeClaim creation to Initialized condition (amd.com/gpu registered)",
    ("nodepool",),
    buckets=_BUCKETS_LONG,
)

# -- cloudprovider decorator --------------------------------------------------

CLOUDPROVIDER_DURATION = _get_or_create(
    Histogram,
    "karpenter_cloudprovider_duration_seconds",
    "Duration of cloud provider method calls",
    ("controller", "method", "provider"),
    buckets=_BUCKETS_SHORT + (120, 300, 600),
)
CLOUDPROVIDER_ERRORS = _get_or_create(
    Counter,
    "karpenter_cloudprovider_errors_total",
    "Total number of errors returned from CloudProvider calls",
    ("controller", "method", "provider", "error_type"),
)

# -- ARM client ---------------------------------------------------------------

ARM_REQUEST_DURATION = _get_or_create(
    Histogram,
    "gpu_provisioner_arm_request_duration_seconds",
    "Duration of Azure ARM API requests",
    ("operation", "code"),
    buckets=_BUCKETS_SHORT + (120, 300),
)
ARM_RETRIES = _get_or_create(
    Counter,
    "gpu_provisioner_arm_request_retries_total",
    "Total ARM request retries",
    ("operation",),
)

# -- runtime ------------------------------------------------------------------

GC_PAUSE_SECONDS = _get_or_create(
    Histogram,
    "gpu_provisioner_gc_pause_seconds",
    "Duration of paced CPython garbage-collection passes (operator.gcpacer)",
    ("generation",),
    buckets=(0.001, 0.0025, 0.005, 0.01, 0.025, 0.05, 0.1, 0.25),
)

# -- build info ---------------------------------------------------------------

BUILD_INFO = _get_or_create(
    Gauge,
    "gpu_provisioner_build_info",
    "Build information",
    ("version",),
)
