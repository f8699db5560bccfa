"""Azure ARM AgentPools API seam: interface, wire shapes, error taxonomy.

The four-method seam every cloud call goes through — the reference's
`AgentPoolsAPI` (pkg/providers/instance/azure_client.go:42-47: BeginCreate-
OrUpdate, BeginDelete, Get, NewListPager). Implementations: fake/agentpools.py
(scripted in-memory AKS) and providers/instance/armclient.py (httpx → ARM).

AgentPool wire shape (containerservice agentPools REST resource):
    {
      "id": ".../agentPools/<name>", "name": "<name>",
      "properties": {
        "count": 1, "vmSize": "...", "osDiskSizeGB": 128, "osType": "Linux",
        "osSKU": "Ubuntu", "provisioningState": "Creating|Succeeded|Deleting|Failed",
        "nodeLabels": {...}, "nodeTaints": ["k=v:Effect"], "tags": {...},
        "scaleSetPriority": "Regular|Spot", "mode": "User",
        "gpuProfile": {...}, "kubeletConfig": {...}, "linuxOSConfig": {...}
      }
    }
"""
from __future__ import annotations

import abc
from typing import AsyncIterator, Optional


class ARMError(Exception):
    """An ARM API error with its HTTP status and service error code."""

    def __init__(self, status: int, code: str, message: str = ""):
        super().__init__(f"{code} ({status}): {message}")
        self.status = status
        self.code = code
        self.message = message


def is_arm_not_found(err: BaseException) -> bool:
    return isinstance(err, ARMError) and err.status == 404


def is_create_in_progress(err: BaseException) -> bool:
    """A create racing a previous (crashed) create — tolerated and adopted
    (reference pkg/providers/instance/instance.go:106-110)."""
    if not isinstance(err, ARMError):
        return False
    return err.status == 409 and err.code in (
        "OperationNotAllowed",
        "AgentPoolOperationInProgress",
        "Conflict",
    ) or "in progress" in (err.message or "").lower()


class LROPoller(abc.ABC):
    """Long-running-operation poller (the ARM create/delete operations take
    minutes; reference drives runtime.Poller via PollUntilDone, armutils.go:35)."""

    @abc.abstractmethod
    def done(self) -> bool:
        ...

    @abc.abstractmethod
    async def poll(self) -> Optional[dict]:
        """One poll round-trip; returns the current resource when available."""

    @abc.abstractmethod
    async def result(self) -> dict:
        """Poll until done; returns the final resource or raises ARMError."""


class AgentPoolsAPI(abc.ABC):
    @abc.abstractmethod
    async def begin_create_or_update(
        self, resource_group: str, cluster_name: str, pool_name: str, agent_pool: dict
    ) -> LROPoller:
        ...

    @abc.abstractmethod
    async def begin_delete(
        self, resource_group: str, cluster_name: str, pool_name: str
    ) -> LROPoller:
        ...

    @abc.abstractmethod
    async def get(self, resource_group: str, cluster_name: str, pool_name: str) -> dict:
        ...

    @abc.abstractmethod
    def list(self, resource_group: str, cluster_name: str) -> AsyncIterator[dict]:
        """Async pager over agent pools."""


# -- helpers over the wire shape ---------------------------------------------


def pool_name(pool: dict) -> str:
    return pool.get("name", "")


def pool_props(pool: dict) -> dict:
    return pool.setdefault("properties", {})


def pool_state(pool: dict) -> str:
    return pool.get("properties", {}).get("provisioningState", "")


def pool_labels(pool: dict) -> dict:
    return pool.get("properties", {}).get("nodeLabels") or {}


def pool_vm_size(pool: dict) -> str:
    return pool.get("properties", {}).get("vmSize", "")


def taint_to_string(taint: dict) -> str:
    """corev1.Taint → AKS node-taint string 'key=value:Effect'."""
    val = taint.get("value", "")
    return f"{taint.get('key')}={val}:{taint.get('effect')}"


def taint_from_string(s: str) -> dict:
    kv, _, effect = s.rpartition(":")
    key, _, value = kv.partition("=")
    t = {"key": key, "effect": effect}
    if value:
        t["value"] = value
    return t
