"""In-memory kube-apiserver with real API semantics, plus an InMemoryClient.

The moral equivalent of envtest for this project (the reference tests against
envtest via vendored pkg/test helpers and against testify/gomock fake clients
— reference pkg/fake/k8sClient.go). This fake implements the semantics the
controllers rely on so multi-actor behavior (controller + simulated kubelet +
simulated device plugin) can be tested fully in-process:

  * monotonically increasing resourceVersion, optimistic-concurrency on update
  * generation bump on spec change
  * finalizer-aware deletion (deletionTimestamp set first; object removed when
    the last finalizer is stripped)
  * uid preconditions on delete
  * watch streams (ADDED/MODIFIED/DELETED) with resourceVersion resume and
    410 Gone on expired versions
  * eviction subresource with scriptable PDB-style 429s
  * scriptable per-verb error injection and reactors (test hooks)
"""
from __future__ import annotations

import asyncio
from collections import deque
import itertools
import uuid
from collections import defaultdict
from typing import Any, AsyncIterator, Callable, Optional

from ..kube import objects as ko
from ..kube.client import (
    ADDED,
    DELETED,
    MODIFIED,
    AlreadyExistsError,
    APIError,
    ConflictError,
    GoneError,
    InvalidError,
    KubeClient,
    LabelSelector,
    NotFoundError,
    json_merge_patch,
    match_field_selector,
)

_WATCH_HISTORY = 4096  # events kept for resourceVersion resume
_WATCH_BROKEN = "__WATCH_BROKEN__"  # sentinel: simulated dropped stream

# Field-selector support matrix of a real kube-apiserver. Every resource
# accepts metadata.name/metadata.namespace; a handful register extra
# selectable fields in their registry strategy (upstream
# pkg/registry/core/pod/strategy.go ToSelectableFields et al.). Everything
# else — including VolumeAttachment, Lease, and any CRD that doesn't declare
# spec.selectableFields — gets "field label not supported" with HTTP 400.
# The fake enforces the same matrix so controller code that would break on a
# real apiserver breaks here too (a `spec.nodeName` selector on
# VolumeAttachment once slipped through and silently skipped the
# volume-detach wait).
_SELECTABLE_FIELDS = {
    ("v1", "Pod"): {
        "spec.nodeName",
        "spec.schedulerName",
        "spec.restartPolicy",
        "spec.serviceAccountName",
        "spec.hostNetwork",
        "status.phase",
        "status.podIP",
        "status.nominatedNodeName",
    },
    ("v1", "Node"): {"spec.unschedulable"},
    ("v1", "Namespace"): {"status.phase", "name"},
    ("v1", "Secret"): {"type"},
    ("v1", "Event"): {
        "involvedObject.kind",
        "involvedObject.namespace",
        "involvedObject.name",
        "involvedObject.uid",
        "involvedObject.apiVersion",
        "involvedObject.resourceVersion",
        "involvedObject.fieldPath",
        "reason",
        "reportingComponent",
        "source",
        "type",
    },
}
_GENERIC_FIELDS = {"metadata.name", "metadata.namespace"}


def validate_field_selector(api_version: str, kind: str, selector: str) -> None:
    """Raise InvalidError (HTTP 400 shape) for selectors a real apiserver
    would reject on this resource."""
    if not selector:
        return
    allowed = _GENERIC_FIELDS | _SELECTABLE_FIELDS.get((api_version, kind), set())
    for part in selector.split(","):
        part = part.strip()
        if not part:
            continue
        key = part.split("!=" if "!=" in part else "=", 1)[0].strip()
        if key not in allowed:
            raise InvalidError(f"field label not supported: {key}")


def _key(namespace: str, name: str) -> tuple:
    return (namespace or "", name)


def _gvk(api_version: str, kind: str) -> tuple:
    return (api_version, kind)


class InMemoryAPIServer:
    """The store. Shared by any number of InMemoryClient instances (one per
    simulated actor) so tests exercise true multi-writer interleavings."""

    def __init__(self):
        self._store: dict = defaultdict(dict)  # gvk -> {(ns,name): obj}
        self._rv = itertools.count(1)
        self._lock = asyncio.Lock()
        self._watchers: list = []  # (gvk, queue)
        # bounded ring: a list re-slice per event past the cap is an
        # O(cap) copy per write and degrades long runs
        self._history: deque = deque(maxlen=_WATCH_HISTORY)  # (rv, gvk, event_type, obj)
        # test hooks: fn(verb, gvk, obj_or_name) -> Optional[APIError] raised if returned
        self.reactors: list = []
        # server-side CRD schema validation (real-apiserver behavior):
        # gvk -> fn(new, old_or_None) -> list[str]; violations → 422
        # (kube/crdschema.CRDValidator; the harness registers the chart's
        # NodeClaim/KaitoNodeClass CRDs here)
        self.validators: dict = {}
        # eviction hook: fn(pod) -> Optional[APIError]
        self.eviction_reactor: Optional[Callable] = None
        self.evictions: list = []  # recorded (namespace, name)

    # -- internals ----------------------------------------------------------

    def _next_rv(self) -> str:
        return str(next(self._rv))

    def _fire_reactors(self, verb: str, gvk: tuple, payload: Any) -> None:
        for r in list(self.reactors):
            err = r(verb, gvk, payload)
            if err is not None:
                raise err

    def _broadcast(self, gvk: tuple, event_type: str, obj: dict) -> None:
        # `obj` is an immutable revision object (see _apply_update): the
        # store, the history and every watcher share it without copying.
        # Watch consumers (informers) treat event objects as read-only —
        # the client-go cache contract — and the server never mutates a
        # stored revision in place (every write replaces it).
        rv = int(obj["metadata"]["resourceVersion"])
        self._history.append((rv, gvk, event_type, obj))
        for wgvk, queue in self._watchers:
            if wgvk == gvk:
                queue.put_nowait((event_type, obj))

    def _bump(self, obj: dict) -> None:
        obj["metadata"]["resourceVersion"] = self._next_rv()

    def _validate(self, gvk: tuple, new: dict, old: Optional[dict]) -> None:
        fn = self.validators.get(gvk)
        if fn is None:
            return
        errs = fn(new, old)
        if errs:
            raise InvalidError(
                f"{gvk[1]} {ko.name_of(new)} is invalid: " + "; ".join(errs[:5])
            )

    # -- verbs (all called under the lock by InMemoryClient) ---------------

    async def get(self, api_version: str, kind: str, name: str, namespace: str) -> dict:
        async with self._lock:
            gvk = _gvk(api_version, kind)
            self._fire_reactors("get", gvk, name)
            obj = self._store[gvk].get(_key(namespace, name))
            if obj is None:
                raise NotFoundError(f"{kind} {namespace}/{name} not found")
            return ko.deep_copy(obj)

    async def list(
        self,
        api_version: str,
        kind: str,
        namespace: str,
        label_selector: str,
        field_selector: str,
    ) -> tuple:
        async with self._lock:
            gvk = _gvk(api_version, kind)
            validate_field_selector(api_version, kind, field_selector)
            self._fire_reactors("list", gvk, None)
            sel = LabelSelector.parse(label_selector) if label_selector else None
            out = []
            for (ns, _), obj in self._store[gvk].items():
                if namespace and ns != namespace:
                    continue
                if sel and not sel.matches(ko.labels_of(obj)):
                    continue
                if not match_field_selector(obj, field_selector):
                    continue
                out.append(ko.deep_copy(obj))
            # list resourceVersion = high-water mark
            return out, str(self._peek_rv())

    def _peek_rv(self) -> int:
        # itertools.count has no peek; track via history / probe
        c = next(self._rv)
        self._rv = itertools.chain([c], self._rv)  # push back
        return c - 1

    async def create(self, obj: dict) -> dict:
        async with self._lock:
            api_version, kind = obj.get("apiVersion", ""), obj.get("kind", "")
            if not api_version or not kind or not ko.name_of(obj):
                raise InvalidError("apiVersion, kind and metadata.name are required")
            gvk = _gvk(api_version, kind)
            self._fire_reactors("create", gvk, obj)
            key = _key(ko.namespace_of(obj), ko.name_of(obj))
            if key in self._store[gvk]:
                raise AlreadyExistsError(f"{kind} {key[0]}/{key[1]} already exists")
            stored = ko.deep_copy(obj)
            m = ko.meta(stored)
            m["uid"] = m.get("uid") or str(uuid.uuid4())
            m["creationTimestamp"] = m.get("creationTimestamp") or ko.fmt_time(ko.now())
            m["generation"] = 1
            self._validate(gvk, stored, None)  # CRD schema + defaults (422)
            self._bump(stored)
            self._store[gvk][key] = stored
            # Events are capped like a real cluster's event TTL would bound
            # them — long soaks otherwise grow the store without limit
            if kind == "Event" and len(self._store[gvk]) > 20000:
                drop = next(iter(self._store[gvk]))
                del self._store[gvk][drop]
            self._broadcast(gvk, ADDED, stored)
            return ko.deep_copy(stored)

    async def update(self, obj: dict, subresource: str = "") -> dict:
        async with self._lock:
            api_version, kind = obj.get("apiVersion", ""), obj.get("kind", "")
            gvk = _gvk(api_version, kind)
            self._fire_reactors("update", gvk, obj)
            key = _key(ko.namespace_of(obj), ko.name_of(obj))
            cur = self._store[gvk].get(key)
            if cur is None:
                raise NotFoundError(f"{kind} {key[0]}/{key[1]} not found")
            rv = obj.get("metadata", {}).get("resourceVersion")
            if rv and rv != cur["metadata"]["resourceVersion"]:
                raise ConflictError(
                    f"{kind} {key[1]}: resourceVersion mismatch ({rv} != {cur['metadata']['resourceVersion']})"
                )
            return self._apply_update(gvk, key, cur, obj, subresource)

    def _apply_update(self, gvk: tuple, key: tuple, cur: dict, obj: dict, subresource: str) -> dict:
        # Revision objects are immutable: a new revision may SHARE unchanged
        # subtrees with the previous one (they are never mutated in place),
        # so a status write is a shallow re-root + one status copy, not a
        # full-object copy. Caller-provided data is still deep-copied — the
        # caller retains its reference and may mutate it afterwards.
        if subresource == "status":
            stored = {**cur, "metadata": {**cur["metadata"]}}
            stored["status"] = ko.deep_copy(obj.get("status", {}))
        else:
            # real-apiserver rule (registry/rest validation): once an object
            # is terminating, NEW finalizers may not be added — attempting
            # it is a 422 "no new finalizers can be added if the object is
            # being deleted"
            if ko.is_deleting(cur):
                added = set(ko.finalizers_of(obj)) - set(ko.finalizers_of(cur))
                if added:
                    raise InvalidError(
                        "metadata.finalizers: Forbidden: no new finalizers can "
                        f"be added if the object is being deleted (added: {sorted(added)})"
                    )
            new = ko.deep_copy(obj)
            # status is a subresource: ignore status changes on main-resource update
            new["status"] = cur.get("status", {})
            # immutable metadata
            nm = ko.meta(new)
            cm = cur["metadata"]
            for f in ("uid", "creationTimestamp", "generation", "deletionTimestamp"):
                if f in cm:
                    nm[f] = cm[f]
                else:
                    nm.pop(f, None)
            if new.get("spec") != cur.get("spec"):
                nm["generation"] = cm.get("generation", 1) + 1
            elif "spec" in new:
                # re-share the unchanged spec subtree with the previous
                # revision: deep_copy broke identity, and identity is what
                # lets the CRD validator's changed-only walk (and any other
                # revision differ) prune in O(1) instead of deep-comparing
                new["spec"] = cur.get("spec")
            stored = new
        self._validate(gvk, stored, cur)  # CRD schema + CEL immutability (422)
        self._bump(stored)
        # finalizer-aware deletion: removing last finalizer on a deleting object
        if ko.is_deleting(stored) and not ko.finalizers_of(stored) and subresource != "status":
            del self._store[gvk][key]
            self._broadcast(gvk, DELETED, stored)
        else:
            self._store[gvk][key] = stored
            self._broadcast(gvk, MODIFIED, stored)
        return ko.deep_copy(stored)

    async def patch(
        self,
        api_version: str,
        kind: str,
        name: str,
        patch: dict,
        namespace: str,
        subresource: str,
    ) -> dict:
        async with self._lock:
            gvk = _gvk(api_version, kind)
            self._fire_reactors("patch", gvk, name)
            key = _key(namespace, name)
            cur = self._store[gvk].get(key)
            if cur is None:
                raise NotFoundError(f"{kind} {namespace}/{name} not found")
            # a merge patch that includes metadata.resourceVersion is an
            # optimistic-lock precondition (real apiserver semantics);
            # without it, merge wins unconditionally
            req_rv = (patch.get("metadata") or {}).get("resourceVersion")
            if req_rv and req_rv != cur["metadata"]["resourceVersion"]:
                raise ConflictError(
                    f"{kind} {name}: resourceVersion mismatch "
                    f"({req_rv} != {cur['metadata']['resourceVersion']})"
                )
            if subresource == "status":
                merged_status = json_merge_patch(cur.get("status", {}), patch.get("status", patch))
                return self._apply_update(gvk, key, cur, {"status": merged_status}, "status")
            merged = json_merge_patch(cur, patch)
            merged.setdefault("metadata", {})["resourceVersion"] = cur["metadata"]["resourceVersion"]
            return self._apply_update(gvk, key, cur, merged, "")

    async def delete(
        self,
        api_version: str,
        kind: str,
        name: str,
        namespace: str,
        uid_precondition: str = "",
        grace_period_seconds: Optional[int] = None,
    ) -> None:
        async with self._lock:
            gvk = _gvk(api_version, kind)
            self._fire_reactors("delete", gvk, name)
            key = _key(namespace, name)
            cur = self._store[gvk].get(key)
            if cur is None:
                raise NotFoundError(f"{kind} {namespace}/{name} not found")
            if uid_precondition and ko.uid_of(cur) != uid_precondition:
                raise ConflictError(f"{kind} {name}: uid precondition failed")
            if ko.finalizers_of(cur):
                if not ko.is_deleting(cur):
                    stored = {**cur, "metadata": {**cur["metadata"]}}
                    ko.meta(stored)["deletionTimestamp"] = ko.fmt_time(ko.now())
                    self._bump(stored)
                    self._store[gvk][key] = stored
                    self._broadcast(gvk, MODIFIED, stored)
                return
            stored = {**cur, "metadata": {**cur["metadata"]}}
            ko.meta(stored)["deletionTimestamp"] = ko.fmt_time(ko.now())
            self._bump(stored)
            del self._store[gvk][key]
            self._broadcast(gvk, DELETED, stored)

    async def evict(self, pod: dict, grace_period_seconds: Optional[int]) -> None:
        async with self._lock:
            if self.eviction_reactor is not None:
                err = self.eviction_reactor(pod)
                if err is not None:
                    raise err
            gvk = _gvk("v1", "Pod")
            key = _key(ko.namespace_of(pod), ko.name_of(pod))
            cur = self._store[gvk].get(key)
            if cur is None:
                raise NotFoundError(f"Pod {key[0]}/{key[1]} not found")
            self.evictions.append(key)
            # eviction == graceful delete
            stored = ko.deep_copy(cur)
            ko.meta(stored)["deletionTimestamp"] = ko.fmt_time(ko.now())
            self._bump(stored)
            if ko.finalizers_of(cur):
                self._store[gvk][key] = stored
                self._broadcast(gvk, MODIFIED, stored)
            else:
                del self._store[gvk][key]
                self._broadcast(gvk, DELETED, stored)

    def subscribe(self, api_version: str, kind: str, resource_version: str) -> tuple:
        """Register a watcher queue; replays history after resource_version.
        Returns (queue, unsubscribe). Raises GoneError if rv expired."""
        gvk = _gvk(api_version, kind)
        queue: asyncio.Queue = asyncio.Queue()
        if resource_version:
            rv = int(resource_version)
            if self._history:
                oldest = self._history[0][0]
                if rv + 1 < oldest and any(True for _ in self._store[gvk]):
                    # can't prove continuity — force re-list
                    raise GoneError(f"resourceVersion {rv} too old")
            for erv, egvk, etype, eobj in self._history:
                if egvk == gvk and erv > rv:
                    queue.put_nowait((etype, ko.deep_copy(eobj)))
        entry = (gvk, queue)
        self._watchers.append(entry)

        def unsubscribe():
            try:
                self._watchers.remove(entry)
            except ValueError:
                pass

        return queue, unsubscribe

    def break_watches(self) -> int:
        """Chaos hook: simulate every open watch stream dropping (as a real
        apiserver does on timeout/netsplit). Watchers receive a stream error
        and must relist+rewatch. Returns the number of streams broken."""
        broken = 0
        for _, queue in list(self._watchers):
            queue.put_nowait((_WATCH_BROKEN, None))
            broken += 1
        return broken


class InMemoryClient(KubeClient):
    """KubeClient over an InMemoryAPIServer."""

    def __init__(self, server: InMemoryAPIServer):
        self.server = server

    async def get(self, api_version: str, kind: str, name: str, namespace: str = "") -> dict:
        return await self.server.get(api_version, kind, name, namespace)

    async def list(
        self,
        api_version: str,
        kind: str,
        namespace: str = "",
        label_selector: str = "",
        field_selector: str = "",
    ) -> list:
        items, _rv = await self.server.list(api_version, kind, namespace, label_selector, field_selector)
        return items

    async def list_with_rv(
        self,
        api_version: str,
        kind: str,
        namespace: str = "",
        label_selector: str = "",
        field_selector: str = "",
    ) -> tuple:
        return await self.server.list(api_version, kind, namespace, label_selector, field_selector)

    async def create(self, obj: dict) -> dict:
        return await self.server.create(obj)

    async def update(self, obj: dict) -> dict:
        return await self.server.update(obj)

    async def update_status(self, obj: dict) -> dict:
        return await self.server.update(obj, subresource="status")

    async def patch(
        self,
        api_version: str,
        kind: str,
        name: str,
        patch: dict,
        namespace: str = "",
        subresource: str = "",
    ) -> dict:
        return await self.server.patch(api_version, kind, name, patch, namespace, subresource)

    async def delete(
        self,
        api_version: str,
        kind: str,
        name: str,
        namespace: str = "",
        uid_precondition: str = "",
        grace_period_seconds: Optional[int] = None,
    ) -> None:
        await self.server.delete(
            api_version, kind, name, namespace, uid_precondition, grace_period_seconds
        )

    async def watch(
        self,
        api_version: str,
        kind: str,
        namespace: str = "",
        resource_version: str = "",
        label_selector: str = "",
    ) -> AsyncIterator[tuple]:
        queue, unsubscribe = self.server.subscribe(api_version, kind, resource_version)
        sel = LabelSelector.parse(label_selector) if label_selector else None
        try:
            while True:
                event_type, obj = await queue.get()
                if event_type == _WATCH_BROKEN:
                    raise APIError("watch stream broken")
                if namespace and ko.namespace_of(obj) != namespace:
                    continue
                if sel and not sel.matches(ko.labels_of(obj)):
                    continue
                yield event_type, obj
        finally:
            unsubscribe()

    async def evict(self, pod: dict, grace_period_seconds: Optional[int] = None) -> None:
        await self.server.evict(pod, grace_period_seconds)
