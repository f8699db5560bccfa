"""Metrics decorator for CloudProvider implementations.

Wraps every method with a duration histogram and an error counter labeled by
controller/method/provider/error-type, as the reference does at
cmd/controller/main.go:41 via vendor/.../pkg/cloudprovider/metrics/
cloudprovider.go:48-104.
"""
from __future__ import annotations

import contextvars
import time
from typing import Optional

from ..metrics.registry import CLOUDPROVIDER_DURATION, CLOUDPROVIDER_ERRORS
from .types import CloudProvider, error_type_of

# Which controller is currently calling the cloud provider — set by the
# controller runner so the decorator can label metrics (the reference threads
# this through context, injection.WithControllerName).
current_controller: contextvars.ContextVar = contextvars.ContextVar(
    "current_controller", default=""
)


class MetricsDecorator(CloudProvider):
    def __init__(self, inner: CloudProvider):
        self.inner = inner

    def __getattr__(self, name: str):
        # forward provider-specific extensions (e.g. invalidate_drift_cache)
        # that are outside the metrics-decorated CloudProvider contract.
        # Guard 'inner' itself: before __init__ assigns it (copy/unpickle
        # paths) forwarding would recurse infinitely.
        if name == "inner":
            raise AttributeError(name)
        return getattr(self.inner, name)

    def _observe(self, method: str, start: float, err: Optional[BaseException]) -> None:
        controller = current_controller.get()
        CLOUDPROVIDER_DURATION.labels(
            controller=controller, method=method, provider=self.inner.name()
        ).observe(time.monotonic() - start)
        if err is not None:
            CLOUDPROVIDER_ERRORS.labels(
                controller=controller,
                method=method,
                provider=self.inner.name(),
                error_type=error_type_of(err),
            ).inc()

    async def _call(self, method: str, coro):
        start = time.monotonic()
        try:
            result = await coro
        except BaseException as e:
            self._observe(method, start, e)
            raise
        self._observe(method, start, None)
        return result

    async def create(self, nodeclaim: dict) -> dict:
        return await self._call("Create", self.inner.create(nodeclaim))

    async def delete(self, nodeclaim: dict) -> None:
        return await self._call("Delete", self.inner.delete(nodeclaim))

    async def get(self, provider_id: str) -> dict:
        return await self._call("Get", self.inner.get(provider_id))

    async def list(self) -> list:
        return await self._call("List", self.inner.list())

    async def get_instance_types(self, nodepool: Optional[dict] = None) -> list:
        return await self._call("GetInstanceTypes", self.inner.get_instance_types(nodepool))

    async def is_drifted(self, nodeclaim: dict) -> str:
        return await self._call("IsDrifted", self.inner.is_drifted(nodeclaim))

    def repair_policies(self) -> list:
        return self.inner.repair_policies()

    def name(self) -> str:
        return self.inner.name()

    def get_supported_node_classes(self) -> list:
        return self.inner.get_supported_node_classes()
