"""node.termination controller: graceful node teardown with drain.

Behavioral spec: reference vendor/sigs.k8s.io/karpenter/pkg/controllers/node/
termination/ — controller.go (finalize pipeline :91-189: ensure NodeClaim
deleted, not-ready+instance-gone short circuit, taint, then
awaitDrain → awaitVolumeDetachment → awaitInstanceTermination :197-288,
finalizer removal + duration metrics :349-377) and terminator/terminator.go
(disrupted taint + LB exclusion :55-92, priority-group drain :119-138, grace
clamping :140-177). awaitInstanceTermination calls cloudProvider.Delete until
NodeClaimNotFound and only then removes the node finalizer — that unblocks
the NodeClaim lifecycle finalize, which is waiting for the Node to vanish.
"""
from __future__ import annotations

import logging
from datetime import timedelta
from typing import Optional

from ...apis import v1 as karpv1
from ...cloudprovider import decorator
from ...cloudprovider.types import CloudProvider, NodeClaimNotFoundError
from ...events.recorder import EventRecorder
from ...kube import objects as ko
from ...kube.client import ConflictError, KubeClient, NotFoundError
from ...kube.controller import Controller, Result, linear_scale_reconciles
from ...kube.informer import Informer
from ...metrics.registry import (
    NODES_DRAINED,
    NODES_TERMINATED,
    NODE_TERMINATION_DURATION,
)
from ...utils import pod as podutils

log = logging.getLogger(__name__)

DRAIN_REQUEUE = 1.0
INSTANCE_POLL = 5.0  # reference controller.go:285


def parse_duration(s: str) -> Optional[timedelta]:
    """k8s metav1.Duration strings like '30s', '5m', '1h30m'."""
    if not s:
        return None
    import re

    total = 0.0
    for num, unit in re.findall(r"(\d+(?:\.\d+)?)(h|m|s|ms)", s):
        total += float(num) * {"h": 3600, "m": 60, "s": 1, "ms": 0.001}[unit]
    return timedelta(seconds=total) if total else None


class TerminationController:
    NAME = "node.termination"

    def __init__(
        self,
        kube: KubeClient,
        cloud: CloudProvider,
        recorder: EventRecorder,
        nodes: Informer,
        nodeclaims: Informer,
        pods: Informer,
        volumeattachments: Optional[Informer],
        eviction_queue,
        workers: Optional[int] = None,
        drain_requeue: float = DRAIN_REQUEUE,
        instance_poll: float = INSTANCE_POLL,
    ):
        self.kube = kube
        self.cloud = cloud
        self.recorder = recorder
        self.nodes = nodes
        self.nodeclaims = nodeclaims
        self.pods = pods
        self.volumeattachments = volumeattachments
        self.eviction_queue = eviction_queue
        self.drain_requeue = drain_requeue
        self.instance_poll = instance_poll
        self.controller = Controller(
            self.NAME,
            self.reconcile,
            # reference scales 100-5000 with CPU (termination/controller.go:58-61)
            workers=workers if workers is not None else linear_scale_reconciles(32, 512),
        )
        from collections import OrderedDict
        # read-your-writes floor (see lifecycle): node-cache reads are
        # trusted only at-or-above our own last write's rv
        self._written_rv: OrderedDict = OrderedDict()
        nodeclaims.add_index(
            "providerID", lambda o: o.get("status", {}).get("providerID") or None
        )
        pods.add_index("nodeName", lambda o: o.get("spec", {}).get("nodeName") or None)
        if volumeattachments is not None:
            volumeattachments.add_index(
                "nodeName", lambda o: o.get("spec", {}).get("nodeName") or None
            )
        nodes.add_handler(self._on_node_event)
        # drain progress is event-driven: pod/volume-attachment changes on a
        # deleting node re-trigger its reconcile immediately, so the
        # drain/volume requeue intervals are only backstops — without these
        # handlers the controller busy-polls, which livelocks at high
        # concurrency (measured: 128-claim teardown saturated the loop)
        pods.add_handler(self._on_pod_event)
        if volumeattachments is not None:
            volumeattachments.add_handler(self._on_va_event)

    def _record_write(self, name: str, obj: dict) -> None:
        try:
            rv = int(obj.get("metadata", {}).get("resourceVersion") or 0)
        except (TypeError, ValueError):
            return
        self._written_rv[name] = max(self._written_rv.get(name, 0), rv)
        self._written_rv.move_to_end(name)
        while len(self._written_rv) > 4096:
            self._written_rv.popitem(last=False)

    def _on_node_event(self, event_type: str, obj: dict) -> None:
        if event_type == "DELETED":
            self._written_rv.pop(ko.name_of(obj), None)
        if ko.is_deleting(obj) and ko.has_finalizer(obj, karpv1.TERMINATION_FINALIZER):
            self.controller.enqueue_nowait(ko.name_of(obj))

    def _on_pod_event(self, event_type: str, obj: dict) -> None:
        node_name = obj.get("spec", {}).get("nodeName")
        if node_name:
            self._enqueue_if_terminating(node_name)

    def _on_va_event(self, event_type: str, obj: dict) -> None:
        node_name = obj.get("spec", {}).get("nodeName")
        if node_name:
            self._enqueue_if_terminating(node_name)

    def _enqueue_if_terminating(self, node_name: str) -> None:
        node = self.nodes.get(node_name) if self.nodes.has_synced else None
        if node is not None and ko.is_deleting(node):
            self.controller.enqueue_nowait(node_name)

    # ------------------------------------------------------------------ main

    async def reconcile(self, key: str) -> Optional[Result]:
        decorator.current_controller.set(self.NAME)
        # cached-client read with a read-your-writes floor (see lifecycle)
        node = None
        if self.nodes.has_synced:
            cached = self.nodes.get(key)
            if cached is not None:
                # opaque-string rv: parse failure → fresh apiserver GET below
                try:
                    rv = int(cached.get("metadata", {}).get("resourceVersion") or 0)
                except (TypeError, ValueError):
                    rv = -1
                if rv >= self._written_rv.get(key, 0):
                    node = ko.deep_copy(cached)
            elif key not in self._written_rv:
                return None
        if node is None:
            try:
                node = await self.kube.get("v1", "Node", key)
            except NotFoundError:
                self._written_rv.pop(key, None)
                return None
        if not ko.is_deleting(node):
            return None
        if not ko.has_finalizer(node, karpv1.TERMINATION_FINALIZER):
            return None
        if not karpv1.node_is_managed(node):
            return None

        nodeclaim = await self._nodeclaim_for_node(node)
        if nodeclaim is not None and not ko.is_deleting(nodeclaim):
            try:
                await self.kube.delete(
                    karpv1.API_VERSION, karpv1.KIND_NODECLAIM, ko.name_of(nodeclaim)
                )
            except NotFoundError:
                pass

        # short-circuit: instance gone + kubelet not Ready → nothing to drain
        # (reference controller.go:117-127)
        if not ko.node_is_ready(node):
            pid = ko.provider_id_of(node)
            if pid:
                try:
                    await self.cloud.get(pid)
                except NodeClaimNotFoundError:
                    return await self._remove_finalizer(node)

        termination_deadline = self._node_termination_time(node, nodeclaim)

        # taint + exclude from load balancers (terminator.go:55-92)
        try:
            await self._taint(node)
        except ConflictError:
            return Result(requeue=True)

        for step in (self._await_drain, self._await_volume_detachment, self._await_instance):
            result = await step(nodeclaim, node, termination_deadline)
            if result is not None:
                return result
        return await self._remove_finalizer(node)

    # ----------------------------------------------------------------- steps

    async def _await_drain(
        self, nodeclaim: Optional[dict], node: dict, deadline
    ) -> Optional[Result]:
        pods = await self._pods_on_node(ko.name_of(node))
        remaining = None
        if deadline is not None:
            remaining = (deadline - ko.now()).total_seconds()
        # evict the lowest-priority non-empty group (terminator.go:119-138)
        group = podutils.group_for_eviction(pods)
        for p in group:
            await self.eviction_queue.add(p, podutils.clamp_grace_period(p, remaining))
        waiting = [p for p in pods if podutils.is_waiting_on(p)]
        if waiting:
            if nodeclaim is not None:
                await self._set_nodeclaim_condition(
                    nodeclaim, karpv1.COND_DRAINED, ko.CONDITION_UNKNOWN, "Draining",
                    f"{len(waiting)} pods remaining",
                )
            return Result(requeue_after=self.drain_requeue)
        if nodeclaim is not None:
            changed = await self._set_nodeclaim_condition(
                nodeclaim, karpv1.COND_DRAINED, ko.CONDITION_TRUE, "Drained"
            )
            if changed:
                NODES_DRAINED.labels(
                    nodepool=ko.labels_of(node).get(karpv1.NODEPOOL_LABEL_KEY, "")
                ).inc()
        return None

    async def _await_volume_detachment(
        self, nodeclaim: Optional[dict], node: dict, deadline
    ) -> Optional[Result]:
        vas = await self._volume_attachments(ko.name_of(node))
        pending = [v for v in vas if not ko.is_deleting(v)]
        if not pending:
            if nodeclaim is not None:
                await self._set_nodeclaim_condition(
                    nodeclaim, karpv1.COND_VOLUMES_DETACHED, ko.CONDITION_TRUE, "VolumesDetached"
                )
            return None
        if deadline is None or ko.now() < deadline:
            if nodeclaim is not None:
                await self._set_nodeclaim_condition(
                    nodeclaim,
                    karpv1.COND_VOLUMES_DETACHED,
                    ko.CONDITION_UNKNOWN,
                    "AwaitingVolumeDetachment",
                    f"{len(pending)} volume attachments remaining",
                )
            return Result(requeue_after=self.drain_requeue)
        # TGP elapsed: proceed to instance termination anyway
        if nodeclaim is not None:
            await self._set_nodeclaim_condition(
                nodeclaim,
                karpv1.COND_VOLUMES_DETACHED,
                ko.CONDITION_FALSE,
                "TerminationGracePeriodElapsed",
                "TerminationGracePeriodElapsed",
            )
        return None

    async def _await_instance(
        self, nodeclaim: Optional[dict], node: dict, deadline
    ) -> Optional[Result]:
        if nodeclaim is None:
            return None
        try:
            await self.cloud.delete(nodeclaim)
        except NodeClaimNotFoundError:
            return None  # instance gone — fall through to finalizer removal
        await self._set_nodeclaim_condition(
            nodeclaim, karpv1.COND_INSTANCE_TERMINATING, ko.CONDITION_TRUE, "InstanceTerminating"
        )
        return Result(requeue_after=self.instance_poll)

    async def _remove_finalizer(self, node: dict) -> Optional[Result]:
        if not ko.remove_finalizer(node, karpv1.TERMINATION_FINALIZER):
            return None
        try:
            updated = await self.kube.update(node)
            self._record_write(ko.name_of(node), updated)
        except NotFoundError:
            return None
        except ConflictError:
            return Result(requeue=True)
        nodepool = ko.labels_of(node).get(karpv1.NODEPOOL_LABEL_KEY, "")
        NODES_TERMINATED.labels(nodepool=nodepool).inc()
        deleted_at = ko.deletion_timestamp_of(node)
        if deleted_at is not None:
            NODE_TERMINATION_DURATION.labels(nodepool=nodepool).observe(
                max(0.0, (ko.now() - deleted_at).total_seconds())
            )
        self.recorder.publish(node, "Terminated", "node drained and instance terminated")
        return None

    # --------------------------------------------------------------- helpers

    async def _taint(self, node: dict) -> None:
        desired = {"key": karpv1.DISRUPTED_TAINT_KEY, "effect": "NoSchedule"}
        # optimistic lock on the taint merge (same lost-update race as
        # registration's taint sync) with IN-PLACE conflict retries against
        # a fresh read: surfacing every conflict as a rate-limited requeue
        # funneled fleet-scale teardowns through the controller's token
        # bucket (32k-claim storms collapsed to bucket speed)
        for attempt in range(5):
            taints = ko.node_taints(node)
            labels = ko.labels_of(node)
            needs_taint = not any(
                t.get("key") == desired["key"] and t.get("effect") == desired["effect"]
                for t in taints
            )
            needs_label = labels.get(karpv1.EXCLUDE_FROM_LB_LABEL_KEY) != "karpenter"
            if not needs_taint and not needs_label:
                return
            patch: dict = {
                "metadata": {"resourceVersion": ko.meta(node).get("resourceVersion")},
                "spec": {},
            }
            if needs_taint:
                patch["spec"]["taints"] = ko.merge_taints(taints, [desired])
            if needs_label:
                patch["metadata"]["labels"] = {
                    **labels,
                    karpv1.EXCLUDE_FROM_LB_LABEL_KEY: "karpenter",
                }
            try:
                updated = await self.kube.patch("v1", "Node", ko.name_of(node), patch)
            except ConflictError:
                try:
                    fresh = await self.kube.get("v1", "Node", ko.name_of(node))
                except NotFoundError:
                    return  # node vanished mid-taint: nothing left to mark
                node.clear()
                node.update(fresh)
                continue
            self._record_write(ko.name_of(node), updated)
            # sync the in-hand node so later writes (finalizer removal) carry
            # the post-taint rv instead of conflicting against our own patch
            node.clear()
            node.update(updated)
            return
        raise ConflictError(f"node {ko.name_of(node)}: taint contention persists")

    async def _nodeclaim_for_node(self, node: dict) -> Optional[dict]:
        pid = ko.provider_id_of(node)
        if not pid:
            return None
        if self.nodeclaims.has_synced and self.nodeclaims.has_index("providerID"):
            claims = self.nodeclaims.by_index("providerID", pid)
        else:
            claims = [
                nc
                for nc in await self.kube.list(karpv1.API_VERSION, karpv1.KIND_NODECLAIM)
                if karpv1.provider_id_of(nc) == pid
            ]
        # duplicates → no single source of truth (reference controller.go:101-105)
        return claims[0] if len(claims) == 1 else None

    async def _pods_on_node(self, node_name: str) -> list:
        if self.pods.has_synced:
            return self.pods.by_index("nodeName", node_name)
        return await self.kube.list("v1", "Pod", field_selector=f"spec.nodeName={node_name}")

    async def _volume_attachments(self, node_name: str) -> list:
        if self.volumeattachments is not None and self.volumeattachments.has_synced:
            return self.volumeattachments.by_index("nodeName", node_name)
        # Fallback before informer sync: list ALL and filter client-side.
        # A real kube-apiserver rejects field selectors on VolumeAttachment
        # (only metadata.name/namespace are registered for it — unlike Pod's
        # spec.nodeName) with 400; the previous `spec.nodeName=` +
        # swallow-everything fallback would report zero pending volumes and
        # silently skip the volume-detach wait. The reference uses an indexed
        # cache for exactly this reason (vendor/.../operator.go:250-293).
        return [
            va
            for va in await self.kube.list("storage.k8s.io/v1", "VolumeAttachment")
            if va.get("spec", {}).get("nodeName") == node_name
        ]

    def _node_termination_time(self, node: dict, nodeclaim: Optional[dict]):
        """Deadline after which drain/volume waits are cut short: the
        termination-timestamp annotation (stamped by health repair or at
        deletion from spec.terminationGracePeriod)."""
        if nodeclaim is None:
            return None
        ann = ko.annotations_of(nodeclaim).get(karpv1.TERMINATION_TIMESTAMP_ANNOTATION_KEY)
        if ann:
            try:
                return ko.parse_time(ann)
            except ValueError:
                return None
        tgp = parse_duration(karpv1.termination_grace_period_of(nodeclaim) or "")
        deleted_at = ko.deletion_timestamp_of(nodeclaim)
        if tgp is not None and deleted_at is not None:
            return deleted_at + tgp
        return None

    async def _set_nodeclaim_condition(
        self, nodeclaim: dict, cond: str, status: str, reason: str, message: str = ""
    ) -> bool:
        """Patch one condition on the NodeClaim status with optimistic
        locking (a raced merge patch of the conditions list would clobber
        concurrent lifecycle/drift writes); returns True if it transitioned."""
        for _ in range(5):
            try:
                fresh = await self.kube.get(
                    karpv1.API_VERSION, karpv1.KIND_NODECLAIM, ko.name_of(nodeclaim)
                )
            except NotFoundError:
                return False
            changed = ko.set_condition(fresh, cond, status, reason, message)
            if not changed:
                return False
            try:
                await self.kube.patch(
                    karpv1.API_VERSION,
                    karpv1.KIND_NODECLAIM,
                    ko.name_of(fresh),
                    {
                        "metadata": {
                            "resourceVersion": fresh["metadata"].get("resourceVersion")
                        },
                        "status": {"conditions": fresh["status"]["conditions"]},
                    },
                    subresource="status",
                )
            except ConflictError:
                continue
            except NotFoundError:
                return False
            return True
        return False
