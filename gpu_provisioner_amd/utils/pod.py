"""Pod classification for drain/eviction.

Spec: reference vendor/sigs.k8s.io/karpenter/pkg/utils/pod/ (drainable/
evictable predicates) and vendor/.../node/termination/terminator/
terminator.go:119-138 (drain grouping: delete expiring pods first, then evict
by group — non-critical non-daemon → non-critical daemon → critical
non-daemon → critical daemon).
"""
from __future__ import annotations

from typing import Optional

from ..kube import objects as ko

SYSTEM_CRITICAL_PRIORITY_CLASSES = ("system-cluster-critical", "system-node-critical")
SYSTEM_CRITICAL_PRIORITY_VALUE = 2_000_000_000


def is_terminal(pod: dict) -> bool:
    return pod.get("status", {}).get("phase") in ("Succeeded", "Failed")


def is_terminating(pod: dict) -> bool:
    return ko.is_deleting(pod)


def is_owned_by_daemonset(pod: dict) -> bool:
    return any(r.get("kind") == "DaemonSet" for r in ko.owner_references_of(pod))


def is_owned_by_node(pod: dict) -> bool:
    """Static/mirror pods are 'owned' by the Node object."""
    return any(r.get("kind") == "Node" for r in ko.owner_references_of(pod))


def is_critical(pod: dict) -> bool:
    if pod.get("spec", {}).get("priorityClassName") in SYSTEM_CRITICAL_PRIORITY_CLASSES:
        return True
    prio = pod.get("spec", {}).get("priority")
    return prio is not None and prio >= SYSTEM_CRITICAL_PRIORITY_VALUE


def is_drainable(pod: dict) -> bool:
    """Pods the drain loop must act on: alive, not static, not already
    terminating past the point of help."""
    if is_terminal(pod):
        return False
    if is_owned_by_node(pod):
        return False  # mirror pods die with the node
    return True


def is_waiting_on(pod: dict) -> bool:
    """Pods that keep the node 'not yet drained': everything drainable except
    daemonset pods (they tolerate the disruption taint and die with the node)."""
    return is_drainable(pod) and not is_owned_by_daemonset(pod)


def eviction_group(pod: dict) -> int:
    """Lower group evicted first: 0 = non-critical non-daemon, 1 = non-critical
    daemon, 2 = critical non-daemon, 3 = critical daemon."""
    crit = is_critical(pod)
    daemon = is_owned_by_daemonset(pod)
    return (2 if crit else 0) + (1 if daemon else 0)


def group_for_eviction(pods: list) -> list:
    """The pods to evict right now: the lowest non-empty eviction group
    (reference terminator.go:119-138 evicts group by group)."""
    groups: dict = {}
    for p in pods:
        if not is_drainable(p) or is_terminating(p):
            continue
        groups.setdefault(eviction_group(p), []).append(p)
    if not groups:
        return []
    return groups[min(groups)]


def clamp_grace_period(pod: dict, seconds_remaining: Optional[float]) -> Optional[int]:
    """Clamp the pod's deletion grace so eviction completes before the node's
    termination deadline (reference terminator.go:140-177). Returns the grace
    to use, or None for the pod default."""
    if seconds_remaining is None:
        return None
    spec_grace = pod.get("spec", {}).get("terminationGracePeriodSeconds")
    limit = max(0, int(seconds_remaining))
    if spec_grace is None:
        return limit
    return min(int(spec_grace), limit)
