"""NodeClaim lifecycle controller: launch → registration → initialization,
plus termination finalize.

Behavioral spec: reference vendor/sigs.k8s.io/karpenter/pkg/controllers/
nodeclaim/lifecycle/ — controller.go (finalizer add :131-145, sub-reconciler
chain :150-155, finalize :181-271), launch.go (UID idempotency cache :38-76,
error taxonomy handling :79-124, status population :126-140), registration.go
(node lookup + taint/label/owner sync :45-92,117-147), initialization.go
(NodeReady + taints-cleared + extended-resource gate :46-86,119-133 — here
gating on amd.com/gpu, the AMD device plugin's resource).

Design differences from the reference (deliberate):
  * reconcile reads come from the informer cache (controller-runtime's
    cached-client semantics) guarded by a read-your-writes floor — the
    cache is only trusted at-or-above the rv of this controller's own
    last write, else one fresh apiserver GET. That gives cache-speed
    foreign-event reconciles without stale re-runs after our own writes,
    and makes the reference's 1 s post-patch sleep (controller.go:160-173)
    unnecessary — provision p50 improves by ~1 s;
  * the liveness sub-reconciler (delete NodeClaims that never Register
    within a timeout) is IMPLEMENTED but off by default: the reference
    disabled it in-tree (controller.go:154) because agent-pool creates can
    legitimately exceed the upstream 15-min registration timeout. Enable
    with the RegistrationLiveness feature gate; the timeout here is 30 min
    to leave room for ROCm driver installation on first boot.
"""
from __future__ import annotations

import logging
import time
from collections import OrderedDict
from typing import Optional

from ...apis import v1 as karpv1
from ...cloudprovider import decorator
from ...cloudprovider.types import (
    CloudProvider,
    CreateError,
    InsufficientCapacityError,
    NodeClaimNotFoundError,
    NodeClassNotReadyError,
)
from ...events.recorder import EventRecorder
from ...kube import objects as ko
from ...kube.client import ConflictError, KubeClient, NotFoundError
from ...kube.controller import Controller, Result, linear_scale_reconciles
from ...kube.informer import Informer
from ...metrics.registry import (
    INITIALIZATION_DURATION,
    LAUNCH_DURATION,
    NODECLAIMS_CREATED,
    NODECLAIMS_TERMINATED,
    NODECLAIM_TERMINATION_DURATION,
    REGISTRATION_DURATION,
)
from ...scheduling.requirements import Requirements

log = logging.getLogger(__name__)

INSTANCE_TERMINATION_REQUEUE = 5.0  # reference lifecycle/controller.go:241
REGISTRATION_REQUEUE = 1.0
LAUNCH_CACHE_TTL = 60.0
# registration-liveness timeout (upstream karpenter uses 15 min; +15 for the
# ROCm/amdgpu driver install on first boot). Only active when the
# RegistrationLiveness gate is on — the reference ships it disabled.
REGISTRATION_TTL_SECONDS = 30 * 60.0


class LifecycleController:
    NAME = "nodeclaim.lifecycle"

    def __init__(
        self,
        kube: KubeClient,
        cloud: CloudProvider,
        recorder: EventRecorder,
        nodeclaims: Informer,
        nodes: Informer,
        workers: Optional[int] = None,
        termination_requeue: float = INSTANCE_TERMINATION_REQUEUE,
        registration_ttl: Optional[float] = None,  # None = liveness disabled
    ):
        self.termination_requeue = termination_requeue
        self.registration_ttl = registration_ttl
        self.kube = kube
        self.cloud = cloud
        self.recorder = recorder
        self.nodeclaims = nodeclaims
        self.nodes = nodes
        # launch idempotency cache: uid -> (deadline, created nodeclaim-shaped dict)
        self._launch_cache: OrderedDict = OrderedDict()
        # read-your-writes floor for cached-client reads: key -> highest rv
        # THIS controller wrote. A cache older than our own last write must
        # not be trusted — re-running the chain on pre-write state costs a
        # stale pass per write (measured ~20% bench throughput).
        self._written_rv: OrderedDict = OrderedDict()
        # echo-suppression floor for our own NODE writes (registration
        # taint/label sync, initialization label): node name -> rv
        self._node_written_rv: OrderedDict = OrderedDict()
        self.controller = Controller(
            self.NAME,
            self.reconcile,
            # reference scales 1000-5000 with CPU (lifecycle/controller.go:56-58);
            # asyncio workers are cheap coroutines so the same envelope applies
            workers=workers if workers is not None else linear_scale_reconciles(64, 1024),
        )
        nodeclaims.add_index(
            "providerID", lambda o: o.get("status", {}).get("providerID") or None
        )
        nodes.add_index("providerID", lambda o: o.get("spec", {}).get("providerID") or None)
        nodeclaims.add_handler(self._on_nodeclaim_event)
        nodes.add_handler(self._on_node_event)

    # -- event mapping -------------------------------------------------------

    def _record_write(self, name: str, obj: dict) -> None:
        """Remember the rv of our own write (read-your-writes floor)."""
        try:
            rv = int(ko.meta(obj).get("resourceVersion") or 0)
        except (TypeError, ValueError):
            return
        self._written_rv[name] = max(self._written_rv.get(name, 0), rv)
        self._written_rv.move_to_end(name)
        while len(self._written_rv) > 4096:
            self._written_rv.popitem(last=False)

    def _on_nodeclaim_event(self, event_type: str, obj: dict) -> None:
        name = ko.name_of(obj)
        if event_type == "DELETED":
            self._written_rv.pop(name, None)
        elif event_type == "MODIFIED":
            # echo suppression: a MODIFIED event at-or-below the rv of OUR
            # OWN last write is the watch echoing that write back — the
            # reconcile that performed it already ran the full chain after
            # the write, so the pass would be a no-op. Foreign writes carry
            # a higher rv and still enqueue; progress otherwise rides on
            # returned requeues and foreign events. Measured: ~1/3 of all
            # lifecycle reconcile passes were self-echoes.
            floor = self._written_rv.get(name)
            if floor is not None:
                try:
                    if int(ko.meta(obj).get("resourceVersion") or 0) <= floor:
                        return
                except (TypeError, ValueError):
                    pass
        if karpv1.is_managed(obj):
            self.controller.enqueue_nowait(name)

    def _record_node_write(self, node_name: str, obj: dict) -> None:
        try:
            rv = int(ko.meta(obj).get("resourceVersion") or 0)
        except (TypeError, ValueError):
            return
        self._node_written_rv[node_name] = max(self._node_written_rv.get(node_name, 0), rv)
        self._node_written_rv.move_to_end(node_name)
        while len(self._node_written_rv) > 4096:
            self._node_written_rv.popitem(last=False)

    def _on_node_event(self, event_type: str, obj: dict) -> None:
        """Map Node events to their NodeClaim via providerID index
        (reference controller.go:92-108)."""
        node_name = ko.name_of(obj)
        if event_type == "DELETED":
            self._node_written_rv.pop(node_name, None)
        elif event_type == "MODIFIED":
            # echo suppression, same argument as _on_nodeclaim_event: the
            # reconcile that patched this node kept running its chain after
            # the write, so the watch echo adds nothing
            floor = self._node_written_rv.get(node_name)
            if floor is not None:
                try:
                    if int(ko.meta(obj).get("resourceVersion") or 0) <= floor:
                        return
                except (TypeError, ValueError):
                    pass
        pid = ko.provider_id_of(obj)
        if not pid:
            return
        for nc in self.nodeclaims.by_index("providerID", pid):
            if karpv1.is_managed(nc):
                self.controller.enqueue_nowait(ko.name_of(nc))

    # -- reconcile -----------------------------------------------------------

    async def reconcile(self, key: str) -> Optional[Result]:
        decorator.current_controller.set(self.NAME)
        # read from the informer cache (controller-runtime's cached-client
        # default: no apiserver round-trip per reconcile) — unless the cache
        # is behind OUR OWN last write for this key (read-your-writes floor);
        # writes still carry resourceVersion preconditions either way.
        nodeclaim = None
        if self.nodeclaims.has_synced:
            cached = self.nodeclaims.get(key)
            if cached is not None:
                # k8s resourceVersion is an opaque string; an unparseable rv
                # must fall through to a fresh apiserver GET, not blow up the
                # reconcile into a hot error-retry loop
                try:
                    rv = int(ko.meta(cached).get("resourceVersion") or 0)
                except (TypeError, ValueError):
                    rv = -1
                if rv >= self._written_rv.get(key, 0):
                    nodeclaim = ko.deep_copy(cached)
            elif key not in self._written_rv:
                return None  # cache authoritative: never seen / fully deleted
        if nodeclaim is None:
            try:
                nodeclaim = await self.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, key)
            except NotFoundError:
                self._written_rv.pop(key, None)
                return None
        if not karpv1.is_managed(nodeclaim):
            return None
        if ko.is_deleting(nodeclaim):
            return await self.finalize(nodeclaim)

        # add termination finalizer before anything can be launched
        # (reference controller.go:131-145)
        if not ko.has_finalizer(nodeclaim, karpv1.TERMINATION_FINALIZER):
            ko.add_finalizer(nodeclaim, karpv1.TERMINATION_FINALIZER)
            try:
                nodeclaim = await self.kube.update(nodeclaim)
                self._record_write(key, nodeclaim)
            except ConflictError:
                return Result(requeue=True)

        results = []
        for sub in (self.launch, self.registration, self.initialization, self.liveness):
            res = await sub(nodeclaim)
            if res is not None:
                results.append(res)
        # a rate-limited requeue (requeue=True, ~5ms backoff) outranks any
        # timed requeue. Dropping it loses the retry entirely and wedges the
        # claim until an unrelated event arrives — found by the chaos suite
        # (registration's node-patch conflict retry was being discarded).
        if any(r.requeue for r in results):
            return Result(requeue=True)
        requeues = [r.requeue_after for r in results if r.requeue_after is not None]
        if requeues:
            return Result(requeue_after=min(requeues))
        return None

    # -- liveness (registration timeout; gated, reference controller.go:154) --

    async def liveness(self, nodeclaim: dict) -> Optional[Result]:
        if self.registration_ttl is None or karpv1.is_registered(nodeclaim):
            return None
        created = ko.creation_timestamp_of(nodeclaim)
        if created is None:
            return None
        elapsed = (ko.now() - created).total_seconds()
        if elapsed < self.registration_ttl:
            return Result(requeue_after=self.registration_ttl - elapsed)
        log.warning(
            "NodeClaim %s not Registered within %.0fs — deleting (RegistrationLiveness)",
            ko.name_of(nodeclaim), self.registration_ttl,
        )
        self.recorder.publish(
            nodeclaim,
            "RegistrationTimeout",
            f"node failed to register within {self.registration_ttl:.0f}s; deleting",
            "Warning",
        )
        await self._delete_nodeclaim(nodeclaim)
        return None

    # -- launch ---------------------------------------------------------------

    async def launch(self, nodeclaim: dict) -> Optional[Result]:
        if karpv1.is_launched(nodeclaim):
            return None
        uid = ko.uid_of(nodeclaim)
        launched_now = False  # metrics/events only when WE created this pass
        cached = self._launch_cache.get(uid)
        if cached is not None and cached[0] > time.monotonic():
            created = cached[1]
        else:
            try:
                created = await self.cloud.create(nodeclaim)
            except InsufficientCapacityError as e:
                # unrecoverable right now: delete the NodeClaim so the owner
                # (KAITO workspace) can retry another SKU
                # (reference launch.go:79-124)
                log.warning("nodeclaim %s: insufficient capacity: %s", ko.name_of(nodeclaim), e)
                self.recorder.publish(nodeclaim, "InsufficientCapacity", str(e), "Warning")
                await self._delete_nodeclaim(nodeclaim)
                return None
            except NodeClassNotReadyError as e:
                log.warning("nodeclaim %s: node class not ready: %s", ko.name_of(nodeclaim), e)
                self.recorder.publish(nodeclaim, "NodeClassNotReady", str(e), "Warning")
                await self._delete_nodeclaim(nodeclaim)
                return None
            except CreateError as e:
                ko.set_condition(
                    nodeclaim, karpv1.COND_LAUNCHED, ko.CONDITION_FALSE, e.condition_reason, str(e)
                )
                await self._patch_status(nodeclaim)
                raise  # rate-limited retry via the workqueue
            # constant TTL ⇒ insertion order == expiry order: expire from
            # the front in O(1) (a full scan per insert cost 164µs at
            # steady-state churn — the cache holds rate×TTL entries)
            launched_now = True
            self._launch_cache[uid] = (time.monotonic() + LAUNCH_CACHE_TTL, created)
            self._launch_cache.move_to_end(uid)
            nw = time.monotonic()
            while self._launch_cache:
                _, (deadline, _c) = next(iter(self._launch_cache.items()))
                if deadline > nw:
                    break
                self._launch_cache.popitem(last=False)

        # populate from the created instance (reference launch.go:126-140)
        labels = ko.labels_of(created)
        if labels:
            merged = {**labels, **ko.labels_of(nodeclaim)}
            updated = await self.kube.patch(
                karpv1.API_VERSION,
                karpv1.KIND_NODECLAIM,
                ko.name_of(nodeclaim),
                {"metadata": {"labels": merged}},
            )
            ko.meta(nodeclaim)["labels"] = merged
            ko.meta(nodeclaim)["resourceVersion"] = updated["metadata"]["resourceVersion"]
            self._record_write(ko.name_of(nodeclaim), updated)
        status = nodeclaim.setdefault("status", {})
        created_status = created.get("status", {})
        status["providerID"] = created_status.get("providerID", "")
        status["imageID"] = created_status.get("imageID", "")
        for f in ("capacity", "allocatable"):
            if created_status.get(f):
                status[f] = created_status[f]
        ko.set_condition(nodeclaim, karpv1.COND_LAUNCHED, ko.CONDITION_TRUE, "Launched")
        await self._patch_status(nodeclaim)
        # the cache's only job is re-launch dedup while Launched is not yet
        # persisted; once the status write lands (read-your-writes floor
        # guarantees later reconciles see it) the entry — holding the full
        # created-instance tree — is dead weight. Dropping it here turns a
        # rate×TTL memory window (~40k entries at bench churn) into
        # in-flight-only; the TTL sweep stays as the crash-path backstop.
        self._launch_cache.pop(uid, None)
        if launched_now:
            # a stale cached-client read can re-run this sub-reconciler after
            # the claim is already Launched; the UID cache dedupes the cloud
            # create, and this guard dedupes the bookkeeping
            self._observe_since_creation(nodeclaim, LAUNCH_DURATION)
            NODECLAIMS_CREATED.labels(
                nodepool=ko.labels_of(nodeclaim).get(karpv1.NODEPOOL_LABEL_KEY, ""),
                capacity_type=ko.labels_of(nodeclaim).get(karpv1.CAPACITY_TYPE_LABEL_KEY, ""),
                instance_type=ko.labels_of(nodeclaim).get(karpv1.INSTANCE_TYPE_LABEL_KEY, ""),
            ).inc()
        self.recorder.publish(nodeclaim, "Launched", f"instance {status['providerID']} launched")
        return None

    # -- registration ----------------------------------------------------------

    async def registration(self, nodeclaim: dict) -> Optional[Result]:
        if not karpv1.is_launched(nodeclaim):
            return None
        if karpv1.is_registered(nodeclaim):
            return None
        provider_id = karpv1.provider_id_of(nodeclaim)
        if not provider_id:
            return Result(requeue_after=REGISTRATION_REQUEUE)
        node = await self._node_by_provider_id(provider_id)
        if node is None:
            # patch only on TRANSITION: an unconditional write here turns the
            # 1 s requeue into a per-claim event storm (every patch
            # broadcasts → handlers re-enqueue → reconcile → patch …)
            if ko.set_condition(
                nodeclaim,
                karpv1.COND_REGISTERED,
                ko.CONDITION_FALSE,
                "NodeNotFound",
                "node has not registered with the cluster yet",
            ):
                await self._patch_status(nodeclaim)
            return Result(requeue_after=REGISTRATION_REQUEUE)

        # sync taints/labels/owner-ref onto the node (registration.go:117-147).
        # Optimistic lock on the taint merge: without the rv precondition the
        # write races the kubelet's one-shot removal of the not-ready startup
        # taint and resurrects it forever (found by the chaos test). The
        # first attempt uses the informer's node; a conflict means our cached
        # rv is stale (the kubelet just wrote), so retry against a FRESH
        # apiserver read instead of burning a requeue cycle per conflict —
        # informer lag made conflict loops the dominant herd-provision tail.
        for attempt in range(5):
            desired_labels = {
                **Requirements.from_nodeclaim(nodeclaim).labels(),
                **ko.labels_of(nodeclaim),
                karpv1.NODE_REGISTERED_LABEL_KEY: "true",
            }
            node_labels = {**ko.labels_of(node), **desired_labels}
            node_taints = ko.merge_taints(
                ko.node_taints(node), nodeclaim.get("spec", {}).get("taints") or []
            )
            node_finalizers = list(ko.finalizers_of(node))
            if (
                karpv1.TERMINATION_FINALIZER not in node_finalizers
                and not ko.is_deleting(node)
            ):
                # the node carries the termination finalizer so node deletion
                # runs the drain pipeline before the kubelet object vanishes.
                # Never append to a TERMINATING node: a real apiserver
                # rejects new finalizers on deleting objects with 422.
                node_finalizers.append(karpv1.TERMINATION_FINALIZER)
            patch: dict = {
                "metadata": {
                    "labels": node_labels,
                    "finalizers": node_finalizers,
                    "resourceVersion": ko.meta(node).get("resourceVersion"),
                    "ownerReferences": [
                        {
                            "apiVersion": karpv1.API_VERSION,
                            "kind": karpv1.KIND_NODECLAIM,
                            "name": ko.name_of(nodeclaim),
                            "uid": ko.uid_of(nodeclaim),
                            "blockOwnerDeletion": True,
                        }
                    ],
                },
                "spec": {"taints": node_taints or None},
            }
            try:
                updated_node = await self.kube.patch("v1", "Node", ko.name_of(node), patch)
                self._record_node_write(ko.name_of(node), updated_node)
                break
            except ConflictError:
                try:
                    node = await self.kube.get("v1", "Node", ko.name_of(node))
                except NotFoundError:
                    return Result(requeue_after=REGISTRATION_REQUEUE)
        else:
            return Result(requeue=True)  # persistent contention: back off
        status = nodeclaim.setdefault("status", {})
        status["nodeName"] = ko.name_of(node)
        ko.set_condition(nodeclaim, karpv1.COND_REGISTERED, ko.CONDITION_TRUE, "Registered")
        await self._patch_status(nodeclaim)
        self._observe_since_creation(nodeclaim, REGISTRATION_DURATION)
        self.recorder.publish(nodeclaim, "Registered", f"node {ko.name_of(node)} registered")
        return None

    # -- initialization ---------------------------------------------------------

    async def initialization(self, nodeclaim: dict) -> Optional[Result]:
        if not karpv1.is_registered(nodeclaim):
            return None
        if karpv1.is_initialized(nodeclaim):
            return None
        node = await self._node_by_provider_id(karpv1.provider_id_of(nodeclaim))
        if node is None:
            return Result(requeue_after=REGISTRATION_REQUEUE)
        reason = self._initialization_gate(nodeclaim, node)
        if reason:
            if ko.set_condition(
                nodeclaim, karpv1.COND_INITIALIZED, ko.CONDITION_FALSE, "NotInitialized", reason
            ):
                await self._patch_status(nodeclaim)
            return Result(requeue_after=REGISTRATION_REQUEUE)
        updated_node = await self.kube.patch(
            "v1",
            "Node",
            ko.name_of(node),
            {"metadata": {"labels": {karpv1.NODE_INITIALIZED_LABEL_KEY: "true"}}},
        )
        self._record_node_write(ko.name_of(node), updated_node)
        status = nodeclaim.setdefault("status", {})
        # `node` can be the SHARED informer-cache object (_nodes_by_provider_id
        # returns it uncopied); aliasing its capacity/allocatable subtrees into
        # NodeClaim status would let a later in-place status edit silently
        # corrupt the cache for every other reader — copy at the boundary
        status["capacity"] = ko.deep_copy(ko.node_capacity(node))
        status["allocatable"] = ko.deep_copy(ko.node_allocatable(node))
        ko.set_condition(nodeclaim, karpv1.COND_INITIALIZED, ko.CONDITION_TRUE, "Initialized")
        ko.set_condition(nodeclaim, karpv1.COND_READY, ko.CONDITION_TRUE, "Ready")
        await self._patch_status(nodeclaim)
        self._observe_since_creation(nodeclaim, INITIALIZATION_DURATION)
        self.recorder.publish(nodeclaim, "Initialized", f"node {ko.name_of(node)} initialized")
        return None

    def _initialization_gate(self, nodeclaim: dict, node: dict) -> str:
        """'' when initialized; else the blocking reason
        (reference initialization.go:46-86,119-133)."""
        if not ko.node_is_ready(node):
            return "node is not Ready"
        node_taints = ko.node_taints(node)
        startup = nodeclaim.get("spec", {}).get("startupTaints") or []
        for t in startup:
            if any(t.get("key") == nt.get("key") and t.get("effect") == nt.get("effect") for nt in node_taints):
                return f"startup taint {t.get('key')} not yet removed"
        for t in karpv1.KNOWN_EPHEMERAL_TAINTS:
            if any(t["key"] == nt.get("key") and t["effect"] == nt.get("effect") for nt in node_taints):
                return f"ephemeral taint {t['key']} not yet removed"
        # the GPU device-plugin gate: every extended resource the NodeClaim
        # requested (amd.com/gpu) must be registered in node allocatable
        alloc = ko.node_allocatable(node)
        for res in self._requested_extended_resources(nodeclaim):
            q = alloc.get(res)
            if q is None or ko.qty(q).is_zero():
                return f"extended resource {res} not yet registered in allocatable"
        return ""

    @staticmethod
    def _requested_extended_resources(nodeclaim: dict) -> list:
        wanted = set()
        for src in (
            nodeclaim.get("spec", {}).get("resources", {}).get("requests", {}),
            nodeclaim.get("status", {}).get("capacity", {}),
        ):
            for res in src:
                if "/" in res and res != "ephemeral-storage":
                    wanted.add(res)
        return sorted(wanted)

    # -- finalize ----------------------------------------------------------------

    async def finalize(self, nodeclaim: dict) -> Optional[Result]:
        """NodeClaim deletion: delete Nodes, then the cloud instance until
        NotFound, then drop the finalizer (reference controller.go:181-271)."""
        if not ko.has_finalizer(nodeclaim, karpv1.TERMINATION_FINALIZER):
            return None
        provider_id = karpv1.provider_id_of(nodeclaim)
        if karpv1.is_registered(nodeclaim) and provider_id:
            nodes = await self._nodes_by_provider_id(provider_id)
            if nodes:
                for node in nodes:
                    if not ko.is_deleting(node):
                        try:
                            await self.kube.delete("v1", "Node", ko.name_of(node))
                        except NotFoundError:
                            pass
                # node deletion completion is EVENT-driven (the node DELETED
                # event maps back to this claim); this requeue is only a
                # backstop. A short interval here multiplies into a
                # fleet-wide polling storm while termination drains.
                return Result(requeue_after=max(self.termination_requeue, 1.0))
        if ko.set_condition(
            nodeclaim, karpv1.COND_INSTANCE_TERMINATING, ko.CONDITION_TRUE, "InstanceTerminating"
        ):
            await self._patch_status(nodeclaim)
        try:
            await self.cloud.delete(nodeclaim)
            return Result(requeue_after=self.termination_requeue)
        except NodeClaimNotFoundError:
            pass  # instance gone — safe to release the NodeClaim
        ko.remove_finalizer(nodeclaim, karpv1.TERMINATION_FINALIZER)
        try:
            updated = await self.kube.update(nodeclaim)
            self._record_write(ko.name_of(nodeclaim), updated)
        except ConflictError:
            return Result(requeue=True)
        except NotFoundError:
            return None
        NODECLAIMS_TERMINATED.labels(
            nodepool=ko.labels_of(nodeclaim).get(karpv1.NODEPOOL_LABEL_KEY, ""),
            capacity_type=ko.labels_of(nodeclaim).get(karpv1.CAPACITY_TYPE_LABEL_KEY, ""),
            instance_type=ko.labels_of(nodeclaim).get(karpv1.INSTANCE_TYPE_LABEL_KEY, ""),
        ).inc()
        deleted_at = ko.deletion_timestamp_of(nodeclaim)
        if deleted_at is not None:
            NODECLAIM_TERMINATION_DURATION.labels(
                nodepool=ko.labels_of(nodeclaim).get(karpv1.NODEPOOL_LABEL_KEY, "")
            ).observe(max(0.0, (ko.now() - deleted_at).total_seconds()))
        self.recorder.publish(nodeclaim, "Terminated", "instance terminated, finalizer removed")
        return None

    # -- helpers -----------------------------------------------------------------

    async def _node_by_provider_id(self, provider_id: str) -> Optional[dict]:
        nodes = await self._nodes_by_provider_id(provider_id)
        return nodes[0] if nodes else None

    async def _nodes_by_provider_id(self, provider_id: str) -> list:
        if not provider_id:
            return []
        if self.nodes.has_synced:
            # trust the index: node arrival triggers a mapped reconcile, and
            # the registration requeue is the backstop — a full List per
            # reconcile is O(cluster) and dominated the provision profile
            return self.nodes.by_index("providerID", provider_id)
        return [
            n
            for n in await self.kube.list("v1", "Node")
            if ko.provider_id_of(n) == provider_id
        ]

    async def _patch_status(self, nodeclaim: dict) -> None:
        """Status write with optimistic locking: a merge patch of the whole
        conditions list based on a stale read would silently erase conditions
        written concurrently by the drift/termination controllers, so the
        patch carries the in-hand resourceVersion and, on conflict, re-reads
        and re-applies only the fields this controller owns."""
        for _ in range(5):
            try:
                updated = await self.kube.patch(
                    karpv1.API_VERSION,
                    karpv1.KIND_NODECLAIM,
                    ko.name_of(nodeclaim),
                    {
                        "metadata": {
                            "resourceVersion": ko.meta(nodeclaim).get("resourceVersion")
                        },
                        "status": nodeclaim.get("status", {}),
                    },
                    subresource="status",
                )
            except ConflictError:
                try:
                    fresh = await self.kube.get(
                        karpv1.API_VERSION, karpv1.KIND_NODECLAIM, ko.name_of(nodeclaim)
                    )
                except NotFoundError:
                    return
                # graft our status onto the fresh object: scalar fields we
                # own replace, our conditions merge by type (foreign
                # conditions like Drifted/Drained survive)
                ours = nodeclaim.get("status", {})
                fresh_status = fresh.setdefault("status", {})
                for f in ("providerID", "imageID", "nodeName", "capacity", "allocatable"):
                    if f in ours:
                        fresh_status[f] = ours[f]
                for cond in ours.get("conditions") or []:
                    ko.set_condition(
                        fresh,
                        cond.get("type", ""),
                        cond.get("status", ""),
                        cond.get("reason", ""),
                        cond.get("message", ""),
                    )
                nodeclaim["status"] = fresh_status
                ko.meta(nodeclaim)["resourceVersion"] = fresh["metadata"]["resourceVersion"]
                continue
            except NotFoundError:
                return
            # keep the in-hand object's resourceVersion fresh so a later
            # update (e.g. finalizer removal) doesn't conflict with our own write
            ko.meta(nodeclaim)["resourceVersion"] = updated["metadata"]["resourceVersion"]
            self._record_write(ko.name_of(nodeclaim), updated)
            return
        log.warning(
            "NodeClaim %s: status patch abandoned after repeated conflicts",
            ko.name_of(nodeclaim),
        )

    async def _delete_nodeclaim(self, nodeclaim: dict) -> None:
        try:
            await self.kube.delete(
                karpv1.API_VERSION,
                karpv1.KIND_NODECLAIM,
                ko.name_of(nodeclaim),
                uid_precondition=ko.uid_of(nodeclaim),
            )
        except (NotFoundError, ConflictError):
            pass

    def _observe_since_creation(self, nodeclaim: dict, histogram) -> None:
        created = ko.creation_timestamp_of(nodeclaim)
        if created is None:
            return
        histogram.labels(
            nodepool=ko.labels_of(nodeclaim).get(karpv1.NODEPOOL_LABEL_KEY, "")
        ).observe(max(0.0, (ko.now() - created).total_seconds()))
