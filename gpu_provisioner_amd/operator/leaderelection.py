"""Lease-based leader election (coordination.k8s.io/v1).

Replaces controller-runtime's leaderelection (reference wires it at
vendor/.../pkg/operator/operator.go:157-163, disabled by default per
options.go:117): acquire a Lease, renew on an interval, yield leadership on
renewal failure. Controllers start only while leading.
"""
from __future__ import annotations

import asyncio
import logging
import socket
import uuid
from typing import Callable, Optional

from ..kube import objects as ko
from ..kube.client import ConflictError, KubeClient, NotFoundError

log = logging.getLogger(__name__)

LEASE_DURATION = 15.0
RENEW_INTERVAL = 5.0
RETRY_INTERVAL = 2.0

# renewal outcomes: transient errors must NOT yield leadership while the
# lease is still live (client-go retries until its renewDeadline)
RENEW_OK = "ok"
RENEW_LOST = "lost"  # another holder took the lease — yield immediately
RENEW_ERROR = "error"  # transient API failure — retry until the deadline


class LeaderElector:
    def __init__(
        self,
        kube: KubeClient,
        name: str,
        namespace: str,
        identity: str = "",
        lease_duration: float = LEASE_DURATION,
        renew_interval: float = RENEW_INTERVAL,
        renew_deadline: Optional[float] = None,
    ):
        self.kube = kube
        self.name = name
        self.namespace = namespace
        self.identity = identity or f"{socket.gethostname()}_{uuid.uuid4().hex[:8]}"
        self.lease_duration = lease_duration
        self.renew_interval = renew_interval
        # how long renewals may keep FAILING before leadership is yielded
        # (client-go's renewDeadline shape: under the lease duration so we
        # stop acting before a rival can acquire the expired lease)
        self.renew_deadline = (
            renew_deadline if renew_deadline is not None else lease_duration * 2.0 / 3.0
        )
        self.is_leader = False
        self._task: Optional[asyncio.Task] = None

    async def run(self, on_started_leading: Callable, on_stopped_leading: Callable) -> None:
        """Blocks forever: acquire → lead (callback) → renew loop → on failure
        release and re-acquire."""
        while True:
            acquired = await self._try_acquire()
            if not acquired:
                await asyncio.sleep(RETRY_INTERVAL)
                continue
            self.is_leader = True
            log.info("leader election: acquired lease %s/%s", self.namespace, self.name)
            await on_started_leading()
            try:
                import time as _time

                last_ok = _time.monotonic()
                while True:
                    await asyncio.sleep(self.renew_interval)
                    outcome = await self._renew()
                    if outcome == RENEW_OK:
                        last_ok = _time.monotonic()
                        continue
                    if outcome == RENEW_LOST:
                        break  # another holder: yield immediately
                    # transient failure: keep leading and retrying until the
                    # renew deadline — a single API blip must not stop the
                    # fleet's only active controller
                    if _time.monotonic() - last_ok > self.renew_deadline:
                        log.warning(
                            "leader election: renewals failing for %.0fs (deadline %.0fs)",
                            _time.monotonic() - last_ok, self.renew_deadline,
                        )
                        break
            finally:
                self.is_leader = False
                log.warning("leader election: lost lease %s/%s", self.namespace, self.name)
                await on_stopped_leading()

    def _lease(self) -> dict:
        now = ko.fmt_micro_time(ko.now())
        return {
            "apiVersion": "coordination.k8s.io/v1",
            "kind": "Lease",
            "metadata": {"name": self.name, "namespace": self.namespace},
            "spec": {
                "holderIdentity": self.identity,
                # the API field is integer seconds; a sub-second duration
                # must not truncate to 0 (= every lease instantly expired)
                "leaseDurationSeconds": max(1, int(self.lease_duration)),
                "acquireTime": now,
                "renewTime": now,
            },
        }

    async def _try_acquire(self) -> bool:
        try:
            cur = await self.kube.get("coordination.k8s.io/v1", "Lease", self.name, self.namespace)
        except NotFoundError:
            try:
                await self.kube.create(self._lease())
                return True
            except Exception:
                return False
        spec = cur.get("spec", {})
        holder = spec.get("holderIdentity", "")
        renew = spec.get("renewTime")
        expired = True
        if renew:
            try:
                age = (ko.now() - ko.parse_time(renew)).total_seconds()
                expired = age > spec.get("leaseDurationSeconds", self.lease_duration)
            except ValueError:
                pass
        if holder == self.identity or expired or not holder:
            cur["spec"] = {**spec, **self._lease()["spec"]}
            try:
                await self.kube.update(cur)
                return True
            except (ConflictError, NotFoundError):
                return False
        return False

    async def release(self) -> None:
        """Graceful-shutdown lease hand-off: clear holderIdentity so a
        follower acquires immediately instead of waiting out the lease
        duration. Best-effort (the expiry path still covers crashes)."""
        if not self.is_leader:
            return
        self.is_leader = False
        try:
            cur = await self.kube.get(
                "coordination.k8s.io/v1", "Lease", self.name, self.namespace
            )
            if cur.get("spec", {}).get("holderIdentity") == self.identity:
                cur["spec"]["holderIdentity"] = ""
                await self.kube.update(cur)
                log.info("leader election: released lease %s/%s", self.namespace, self.name)
        except Exception as e:
            log.warning("leader election: lease release failed (expiry will cover): %s", e)

    async def _renew(self) -> str:
        try:
            cur = await self.kube.get("coordination.k8s.io/v1", "Lease", self.name, self.namespace)
        except NotFoundError:
            return RENEW_LOST  # lease deleted out from under us
        except Exception:
            return RENEW_ERROR
        if cur.get("spec", {}).get("holderIdentity") != self.identity:
            return RENEW_LOST
        cur["spec"]["renewTime"] = ko.fmt_micro_time(ko.now())
        try:
            await self.kube.update(cur)
            return RENEW_OK
        except (ConflictError, NotFoundError):
            # concurrent write to OUR lease — re-judge ownership next tick
            return RENEW_ERROR
        except Exception:
            return RENEW_ERROR
