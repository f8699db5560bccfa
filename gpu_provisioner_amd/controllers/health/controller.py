"""node.health controller: auto-repair of unhealthy MI355X nodes.

Behavioral spec: reference vendor/sigs.k8s.io/karpenter/pkg/controllers/node/
health/controller.go — watch Node condition transitions (:73-104), match the
cloud provider's RepairPolicies (NodeReady False/Unknown tolerated 10 min per
pkg/cloudprovider/cloudprovider.go:103-116), requeue until the toleration
elapses (:120-125), then annotate the NodeClaim with a termination timestamp
(forcing drain to cut short) and delete it (:152-183). The reference fork
disabled the cluster-health-percentage gate (:130-151) — we keep that
behavior: repairs proceed regardless of fleet health, because a KAITO fleet
is typically small and a wedged 8×MI355X node is expensive to keep.

Enabled when repair_policies() is non-empty and the NodeRepair feature gate
is on (reference controllers.go:109-111).
"""
from __future__ import annotations

import logging
from datetime import timedelta
from typing import Optional

from ...apis import v1 as karpv1
from ...cloudprovider import decorator
from ...cloudprovider.types import CloudProvider
from ...events.recorder import EventRecorder
from ...kube import objects as ko
from ...kube.client import KubeClient, NotFoundError
from ...kube.controller import Controller, Result
from ...kube.informer import Informer

log = logging.getLogger(__name__)


class HealthController:
    NAME = "node.health"

    def __init__(
        self,
        kube: KubeClient,
        cloud: CloudProvider,
        recorder: EventRecorder,
        nodes: Informer,
        nodeclaims: Optional[Informer] = None,
        workers: int = 16,
    ):
        self.kube = kube
        self.cloud = cloud
        self.recorder = recorder
        self.nodes = nodes
        self.nodeclaims = nodeclaims
        if nodeclaims is not None:
            # same providerID index every other controller uses — an O(cluster)
            # NodeClaim list per unhealthy node does not belong on this path
            nodeclaims.add_index(
                "providerID", lambda o: o.get("status", {}).get("providerID") or None
            )
        self.controller = Controller(self.NAME, self.reconcile, workers=workers)
        self._first_seen: dict = {}  # (node, cond_type, status) -> first observed
        nodes.add_handler(self._on_node_event)

    def _on_node_event(self, event_type: str, obj: dict) -> None:
        if event_type == "DELETED":
            # drop first-seen bookkeeping for vanished nodes (slow leak)
            name = ko.name_of(obj)
            for k in [k for k in self._first_seen if k[0] == name]:
                del self._first_seen[k]
            return
        if not karpv1.node_is_managed(obj):
            return
        # fleet-scale pre-filter: kubelet heartbeats dominate node events;
        # a node with no policy-matching condition needs no reconcile (the
        # toleration timer rides on the reconcile's own requeue). A node
        # that LATER degrades fires this handler again with the unhealthy
        # condition present.
        if self._match_policy(obj) is None:
            return
        self.controller.enqueue_nowait(ko.name_of(obj))

    async def reconcile(self, key: str) -> Optional[Result]:
        decorator.current_controller.set(self.NAME)
        # read-only reconcile: the informer cache suffices (no mutation of
        # the node object below) — a GET per heartbeat is O(cluster) load
        node = self.nodes.get(key) if self.nodes.has_synced else None
        if node is None:
            try:
                node = await self.kube.get("v1", "Node", key)
            except NotFoundError:
                return None
        if not karpv1.node_is_managed(node) or ko.is_deleting(node):
            return None
        match = self._match_policy(node)
        if match is None:
            return None
        condition, policy = match
        transitioned = condition.get("lastTransitionTime")
        if transitioned:
            since = ko.parse_time(transitioned)
        else:
            # kubelet (or a simulator) that doesn't stamp lastTransitionTime:
            # fall back to the first time WE observed this condition state
            seen_key = (key, policy.condition_type, policy.condition_status)
            since = self._first_seen.setdefault(seen_key, ko.now())
            # clear stale entries for other statuses of this type
            for k in list(self._first_seen):
                if k[0] == key and k[1] == policy.condition_type and k != seen_key:
                    del self._first_seen[k]
        elapsed = (ko.now() - since).total_seconds()
        if elapsed < policy.toleration_seconds:
            return Result(requeue_after=policy.toleration_seconds - elapsed)

        nodeclaim = await self._nodeclaim_for(node)
        if nodeclaim is None:
            return None
        if ko.is_deleting(nodeclaim):
            return None
        log.warning(
            "node %s unhealthy (%s=%s for %.0fs) — repairing via NodeClaim delete",
            key, policy.condition_type, policy.condition_status, elapsed,
        )
        self.recorder.publish(
            nodeclaim,
            "NodeRepair",
            f"node {key} {policy.condition_type}={policy.condition_status} "
            f"for {elapsed:.0f}s (tolerated {policy.toleration_seconds:.0f}s); replacing",
            "Warning",
        )
        # force-terminate: stamp the termination timestamp so drain/volume
        # waits cut short instead of blocking on a wedged kubelet
        ko.set_annotation(
            nodeclaim,
            karpv1.TERMINATION_TIMESTAMP_ANNOTATION_KEY,
            ko.fmt_time(ko.now() + timedelta(seconds=0)),
        )
        try:
            await self.kube.patch(
                karpv1.API_VERSION,
                karpv1.KIND_NODECLAIM,
                ko.name_of(nodeclaim),
                {"metadata": {"annotations": ko.annotations_of(nodeclaim)}},
            )
            await self.kube.delete(
                karpv1.API_VERSION,
                karpv1.KIND_NODECLAIM,
                ko.name_of(nodeclaim),
                uid_precondition=ko.uid_of(nodeclaim),
            )
        except NotFoundError:
            pass
        return None

    def _match_policy(self, node: dict):
        for policy in self.cloud.repair_policies():
            for cond in node.get("status", {}).get("conditions") or []:
                if (
                    cond.get("type") == policy.condition_type
                    and cond.get("status") == policy.condition_status
                ):
                    return cond, policy
        return None

    async def _nodeclaim_for(self, node: dict) -> Optional[dict]:
        pid = ko.provider_id_of(node)
        if not pid:
            return None
        if self.nodeclaims is not None and self.nodeclaims.has_synced:
            claims = [
                nc
                for nc in self.nodeclaims.by_index("providerID", pid)
                if karpv1.is_managed(nc)
            ]
        else:
            claims = [
                nc
                for nc in await self.kube.list(karpv1.API_VERSION, karpv1.KIND_NODECLAIM)
                if karpv1.provider_id_of(nc) == pid and karpv1.is_managed(nc)
            ]
        # private copy: the caller stamps the termination-timestamp annotation
        # in place, which must never mutate the shared informer-cache object
        return ko.deep_copy(claims[0]) if len(claims) == 1 else None
