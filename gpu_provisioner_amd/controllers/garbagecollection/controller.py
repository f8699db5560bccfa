"""The two leak garbage collectors — asymmetric guards, opposite directions.

1. InstanceGCController (cloud→cluster; reference first-party
   pkg/controllers/instance/garbagecollection/controller.go:51-124): every
   2 min, delete cloud agent pools whose NodeClaim no longer exists and that
   are older than 30 s (covers pools leaked when a NodeClaim is deleted
   mid-create — reference readme.md:1-16), then delete their leaked Node
   objects. 20-way parallel.

2. NodeClaimGCController (cluster→cloud; reference vendored
   vendor/.../controllers/nodeclaim/garbagecollection/controller.go:60-118):
   every 2 min, delete Registered NodeClaims whose providerID has vanished
   from cloudProvider.List() and whose Node is not Ready.

Getting either guard wrong deletes healthy capacity: the 30 s age floor keeps
GC #1 from racing an in-flight create; the Registered+NodeReady check keeps
GC #2 from killing claims whose instance List was transiently incomplete.
"""
from __future__ import annotations

import asyncio
import logging
from typing import Optional

from ...apis import v1 as karpv1
from ...cloudprovider import decorator
from ...cloudprovider.types import CloudProvider, NodeClaimNotFoundError
from ...events.recorder import EventRecorder
from ...kube import objects as ko
from ...kube.client import KubeClient, NotFoundError
from ...kube.controller import Result, SingletonController

log = logging.getLogger(__name__)

GC_INTERVAL = 120.0  # reference :123 — every 2 min
ADOPTION_AGE_SECONDS = 30.0  # reference :72-84
GC_PARALLELISM = 20  # reference :91


class InstanceGCController:
    NAME = "instance.garbagecollection"

    def __init__(
        self,
        kube: KubeClient,
        cloud: CloudProvider,
        recorder: EventRecorder,
        interval: float = GC_INTERVAL,
        adoption_age: float = ADOPTION_AGE_SECONDS,
        nodes=None,  # Optional[Informer]: indexed leaked-Node lookups
    ):
        self.kube = kube
        self.cloud = cloud
        self.recorder = recorder
        self.adoption_age = adoption_age
        self.nodes = nodes
        if nodes is not None:
            nodes.add_index(
                "providerID", lambda o: o.get("spec", {}).get("providerID") or None
            )
            # same fn shape as InstanceProvider.set_nodes_informer — add_index
            # is first-registration-wins by name, so both must index BOTH
            # agentpool label keys
            nodes.add_index(
                "agentpool",
                lambda o: [
                    v
                    for v in (
                        (o.get("metadata", {}).get("labels") or {}).get(
                            karpv1.AGENTPOOL_LABEL_KEY
                        ),
                        (o.get("metadata", {}).get("labels") or {}).get(
                            karpv1.AZURE_AGENTPOOL_LABEL_KEY
                        ),
                    )
                    if v
                ]
                or None,
            )
        self.controller = SingletonController(self.NAME, self.reconcile, interval)

    async def reconcile(self, key: str) -> Optional[Result]:
        decorator.current_controller.set(self.NAME)
        instances = await self.cloud.list()  # nodeclaim-shaped records
        claims = await self.kube.list(karpv1.API_VERSION, karpv1.KIND_NODECLAIM)
        live = {ko.name_of(nc) for nc in claims if karpv1.is_managed(nc)}
        now = ko.now()
        leaked = []
        for inst in instances:
            if ko.is_deleting(inst):
                continue  # already being deleted cloud-side
            if ko.name_of(inst) in live:
                continue
            created = self._created_at(inst)
            if created is not None and (now - created).total_seconds() < self.adoption_age:
                continue  # might be an in-flight create we haven't seen yet
            leaked.append(inst)
        if not leaked:
            return None
        sem = asyncio.Semaphore(GC_PARALLELISM)

        async def collect(inst: dict) -> None:
            async with sem:
                name = ko.name_of(inst)
                log.info("GC: deleting leaked instance %s", name)
                try:
                    await self.cloud.delete(inst)
                except NodeClaimNotFoundError:
                    pass
                await self._delete_leaked_nodes(inst)

        await asyncio.gather(*(collect(i) for i in leaked))
        return None

    def _created_at(self, inst: dict):
        # creation-timestamp label persisted on the agent pool (provider.py
        # CREATION_TIMESTAMP_LABEL) — survives controller restarts
        from ...providers.instance.provider import CREATION_TIMESTAMP_LABEL

        raw = ko.labels_of(inst).get(CREATION_TIMESTAMP_LABEL)
        if not raw:
            return None
        try:
            from datetime import datetime, timezone

            return datetime.fromtimestamp(int(raw) / 1000.0, tz=timezone.utc)
        except (ValueError, OSError):
            return None

    async def _delete_leaked_nodes(self, inst: dict) -> None:
        pid = karpv1.provider_id_of(inst)
        name = ko.name_of(inst)
        if self.nodes is not None and self.nodes.has_synced:
            # indexed lookups (shared cache objects → copy before mutating)
            matches: dict = {}
            if pid:
                for n in self.nodes.by_index("providerID", pid):
                    matches[ko.name_of(n)] = n
            for n in self.nodes.by_index("agentpool", name):
                matches.setdefault(ko.name_of(n), n)
            leaked_nodes = [ko.deep_copy(n) for n in matches.values()]
        else:
            leaked_nodes = [
                node
                for node in await self.kube.list("v1", "Node")
                if (ko.provider_id_of(node) == pid and pid)
                or ko.labels_of(node).get(karpv1.AGENTPOOL_LABEL_KEY) == name
            ]
        for node in leaked_nodes:
            try:
                # strip our finalizer first: the instance is already gone,
                # there is nothing left to drain
                if ko.remove_finalizer(node, karpv1.TERMINATION_FINALIZER):
                    await self.kube.update(node)
                await self.kube.delete("v1", "Node", ko.name_of(node))
            except NotFoundError:
                pass


class NodeClaimGCController:
    NAME = "nodeclaim.garbagecollection"

    def __init__(
        self,
        kube: KubeClient,
        cloud: CloudProvider,
        recorder: EventRecorder,
        interval: float = GC_INTERVAL,
        nodes=None,  # Optional[Informer]: indexed node lookups
    ):
        self.kube = kube
        self.cloud = cloud
        self.recorder = recorder
        self.nodes = nodes
        if nodes is not None:
            # providerID index (the same one lifecycle registers) — a full
            # Node list per doomed claim is O(cluster) inside a loop
            nodes.add_index(
                "providerID", lambda o: o.get("spec", {}).get("providerID") or None
            )
        self.controller = SingletonController(self.NAME, self.reconcile, interval)

    async def reconcile(self, key: str) -> Optional[Result]:
        decorator.current_controller.set(self.NAME)
        instances = await self.cloud.list()
        cloud_ids = {karpv1.provider_id_of(i) for i in instances if karpv1.provider_id_of(i)}
        cloud_names = {ko.name_of(i) for i in instances}
        claims = await self.kube.list(karpv1.API_VERSION, karpv1.KIND_NODECLAIM)
        doomed = []
        for nc in claims:
            if not karpv1.is_managed(nc) or ko.is_deleting(nc):
                continue
            if not karpv1.is_registered(nc):
                continue
            pid = karpv1.provider_id_of(nc)
            if (pid and pid in cloud_ids) or ko.name_of(nc) in cloud_names:
                continue
            node = await self._node_for(pid)
            if node is not None and ko.node_is_ready(node):
                # kubelet still reporting Ready: trust the node over a
                # possibly-stale cloud List (reference :77-98)
                continue
            doomed.append(nc)
        if not doomed:
            return None
        sem = asyncio.Semaphore(GC_PARALLELISM)

        async def collect(nc: dict) -> None:
            async with sem:
                log.info("GC: deleting NodeClaim %s (instance vanished)", ko.name_of(nc))
                self.recorder.publish(
                    nc, "InstanceVanished", "cloud instance no longer exists", "Warning"
                )
                try:
                    await self.kube.delete(
                        karpv1.API_VERSION,
                        karpv1.KIND_NODECLAIM,
                        ko.name_of(nc),
                        uid_precondition=ko.uid_of(nc),
                    )
                except NotFoundError:
                    pass

        await asyncio.gather(*(collect(nc) for nc in doomed))
        return None

    async def _node_for(self, provider_id: str) -> Optional[dict]:
        if not provider_id:
            return None
        if self.nodes is not None and self.nodes.has_synced:
            matches = self.nodes.by_index("providerID", provider_id)
            return matches[0] if matches else None  # read-only use
        for node in await self.kube.list("v1", "Node"):
            if ko.provider_id_of(node) == provider_id:
                return node
        return None
