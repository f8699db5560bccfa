"""Eviction queue: rate-limited, deduplicated pod eviction workers.

Spec: reference vendor/sigs.k8s.io/karpenter/pkg/controllers/node/
termination/terminator/eviction.go:93-175 — a shared queue (capacity 10k)
with a dedup set, evicting via the eviction subresource, retrying PDB 429s
with backoff and recording events.
"""
from __future__ import annotations

import logging
from typing import Optional

from ...events.recorder import EventRecorder
from ...kube import objects as ko
from ...kube.client import KubeClient, NotFoundError, TooManyRequestsError
from ...kube.controller import Controller, Result
from ...kube.workqueue import RateLimiter

log = logging.getLogger(__name__)


class EvictionQueue:
    """Enqueue (namespace, name, grace) triples; workers evict with retries."""

    NAME = "eviction.queue"

    def __init__(self, kube: KubeClient, recorder: EventRecorder, workers: int = 32):
        self.kube = kube
        self.recorder = recorder
        self._grace: dict = {}  # (ns, name) -> grace seconds (latest wins)
        self.controller = Controller(
            self.NAME,
            self._reconcile,
            workers=workers,
            # PDB retries: 1s base up to 1 min, generous bucket
            rate_limiter=RateLimiter(base=1.0, cap=60.0, qps=100.0, burst=1000),
        )

    def start(self) -> None:
        self.controller.start()

    async def stop(self) -> None:
        await self.controller.stop()

    async def add(self, pod: dict, grace_period_seconds: Optional[int] = None) -> None:
        key = (ko.namespace_of(pod), ko.name_of(pod))
        self._grace[key] = grace_period_seconds
        await self.controller.queue.add(key)

    async def _reconcile(self, key: tuple) -> Optional[Result]:
        ns, name = key
        grace = self._grace.get(key)
        try:
            pod = await self.kube.get("v1", "Pod", name, ns)
        except NotFoundError:
            self._grace.pop(key, None)
            return None
        try:
            await self.kube.evict(pod, grace)
        except NotFoundError:
            pass
        except TooManyRequestsError as e:
            # PDB blocking: retry with backoff (eviction.go:140-175)
            self.recorder.publish(
                pod, "NotEvicted", f"eviction blocked by PodDisruptionBudget: {e.message}",
                "Warning",
            )
            return Result(requeue=True)
        self.recorder.publish(pod, "Evicted", "evicted in node drain")
        self._grace.pop(key, None)
        return None
