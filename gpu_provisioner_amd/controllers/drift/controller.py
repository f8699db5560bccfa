"""nodeclaim.drift controller: detect (and optionally replace) drifted nodes.

Net-new capability vs the reference: its `IsDrifted` is a stub returning ""
(reference pkg/cloudprovider/cloudprovider.go:94-97) and every disruption
controller in its vendored karpenter fork is commented out
(vendor/.../pkg/controllers/controllers.go:50-115), so a NodeClaim whose
agent pool no longer matches its spec — e.g. the kaito.sh/node-image-family
annotation changed after provisioning, the pool was mutated out-of-band, or
the SKU was retired from the MI355X catalog — is silently stale forever.

This controller is a singleton sweep (same cadence model as the GC pair):
every `interval` it lists managed, Launched, non-deleting NodeClaims, asks
the cloud provider `is_drifted`, and maintains the `Drifted` condition on
status with the drift reason. When the `DriftReplace` feature gate is on it
additionally deletes drifted NodeClaims (honoring karpenter.sh/do-not-disrupt)
so the owner (KAITO) re-creates a conforming node through the normal
termination path (§3.3) — the MI355X fleet converges instead of rotting.
"""
from __future__ import annotations

import asyncio
import logging
from typing import Optional

from ...apis import v1 as karpv1
from ...cloudprovider import decorator
from ...cloudprovider.types import CloudProvider
from ...events.recorder import EventRecorder
from ...kube import objects as ko
from ...kube.client import ConflictError, KubeClient, NotFoundError
from ...kube.controller import Result, SingletonController
from ...metrics.registry import NODECLAIMS_DISRUPTED, NODECLAIMS_DRIFTED

log = logging.getLogger(__name__)

DRIFT_INTERVAL = 120.0  # same sweep cadence as the GC pair
DRIFT_PARALLELISM = 20


class DriftController:
    NAME = "nodeclaim.drift"

    def __init__(
        self,
        kube: KubeClient,
        cloud: CloudProvider,
        recorder: EventRecorder,
        *,
        interval: float = DRIFT_INTERVAL,
        replace: bool = False,
    ):
        self.kube = kube
        self.cloud = cloud
        self.recorder = recorder
        self.replace = replace
        self.controller = SingletonController(self.NAME, self.reconcile, interval)

    async def reconcile(self, key: str) -> Optional[Result]:
        decorator.current_controller.set(self.NAME)
        # fresh cloud snapshot per sweep (one paged LIST; the provider
        # serves every per-claim is_drifted from it)
        invalidate = getattr(self.cloud, "invalidate_drift_cache", None)
        if invalidate is not None:
            invalidate()
        claims = await self.kube.list(karpv1.API_VERSION, karpv1.KIND_NODECLAIM)
        candidates = [
            nc
            for nc in claims
            if karpv1.is_managed(nc)
            and not ko.is_deleting(nc)
            and karpv1.is_launched(nc)
            and karpv1.provider_id_of(nc)
        ]
        if not candidates:
            return None
        sem = asyncio.Semaphore(DRIFT_PARALLELISM)
        await asyncio.gather(*(self._check_one(nc, sem) for nc in candidates))
        return None

    async def _check_one(self, nodeclaim: dict, sem: asyncio.Semaphore) -> None:
        async with sem:
            name = ko.name_of(nodeclaim)
            reason = await self.cloud.is_drifted(nodeclaim)
            changed = await self._set_drifted_condition(name, reason)
            if not reason:
                return
            if changed:
                log.info("NodeClaim %s drifted: %s", name, reason)
                NODECLAIMS_DRIFTED.labels(
                    reason=reason,
                    nodepool=ko.labels_of(nodeclaim).get(
                        karpv1.NODEPOOL_LABEL_KEY, karpv1.KAITO_NODEPOOL_NAME
                    ),
                ).inc()
                self.recorder.publish(
                    nodeclaim, "Drifted", f"node no longer matches spec: {reason}", "Warning"
                )
            if self.replace:
                await self._replace(nodeclaim, reason)

    async def _set_drifted_condition(self, name: str, reason: str) -> bool:
        """Maintain the Drifted condition with optimistic-concurrency retries.

        Returns True when the condition transitioned (for metrics/events
        dedup). Conditions are written via update_status so a concurrent
        lifecycle write is never clobbered by a merge-patch of the whole list.
        """
        for _ in range(5):
            try:
                nc = await self.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, name)
            except NotFoundError:
                return False
            if reason:
                changed = ko.set_condition(
                    nc, karpv1.COND_DRIFTED, ko.CONDITION_TRUE, reason,
                    f"agent pool no longer matches NodeClaim spec ({reason})",
                )
            else:
                if ko.get_condition(nc, karpv1.COND_DRIFTED) is None:
                    return False  # never drifted: keep status untouched
                changed = ko.set_condition(
                    nc, karpv1.COND_DRIFTED, ko.CONDITION_FALSE, "NotDrifted", ""
                )
            if not changed:
                return False
            try:
                await self.kube.update_status(nc)
                return True
            except ConflictError:
                continue
            except NotFoundError:
                return False
        log.warning("NodeClaim %s: gave up setting Drifted condition after conflicts", name)
        return False

    async def _replace(self, nodeclaim: dict, reason: str) -> None:
        name = ko.name_of(nodeclaim)
        if karpv1.DO_NOT_DISRUPT_ANNOTATION_KEY in ko.annotations_of(nodeclaim):
            log.info("NodeClaim %s drifted (%s) but carries do-not-disrupt; keeping", name, reason)
            return
        log.info("NodeClaim %s drifted (%s) — replacing (DriftReplace gate on)", name, reason)
        NODECLAIMS_DISRUPTED.labels(
            reason="drifted",
            nodepool=ko.labels_of(nodeclaim).get(
                karpv1.NODEPOOL_LABEL_KEY, karpv1.KAITO_NODEPOOL_NAME
            ),
        ).inc()
        self.recorder.publish(
            nodeclaim, "DriftReplace", f"replacing drifted node: {reason}", "Warning"
        )
        try:
            await self.kube.delete(
                karpv1.API_VERSION,
                karpv1.KIND_NODECLAIM,
                name,
                uid_precondition=ko.uid_of(nodeclaim),
            )
        except (NotFoundError, ConflictError):
            # NotFound: already gone. Conflict: uid changed under the same
            # name — the claim was already replaced; either way nothing to
            # do, and the exception must not abort the sweep's gather.
            pass
