"""Unit tests for the kube core: quantities, selectors, objects, workqueue,
and the in-memory apiserver's API semantics (optimistic concurrency,
finalizers, watches)."""
import asyncio

import pytest

from gpu_provisioner_amd.fake.apiserver import InMemoryAPIServer, InMemoryClient
from gpu_provisioner_amd.kube import objects as ko
from gpu_provisioner_amd.kube.client import (
    ADDED,
    DELETED,
    AlreadyExistsError,
    ConflictError,
    LabelSelector,
    NotFoundError,
    json_merge_patch,
    match_field_selector,
)
from gpu_provisioner_amd.kube.workqueue import RateLimitingQueue, RateLimiter
from tests.conftest import run

# ---------------------------------------------------------------- quantities


def test_quantity_parsing_and_arithmetic():
    assert ko.qty("8").value == 8
    assert ko.qty("100m").value == pytest.approx(0.1)
    assert ko.qty("2Gi").value == 2 * 2**30
    assert ko.qty("288G").value == 288e9
    assert ko.qty("1.5Ti").value == 1.5 * 2**40
    assert ko.qty("2Gi") - ko.qty("1Gi") == ko.qty("1Gi")
    assert ko.qty("0").is_zero()
    assert not ko.qty("8").is_zero()
    with pytest.raises(ValueError):
        ko.qty("8Zi")


# ----------------------------------------------------------------- selectors


def test_label_selector_forms():
    sel = LabelSelector.parse("a=1,b!=2,c,!d,e in (x,y),f notin (z)")
    assert sel.matches({"a": "1", "c": "yes", "e": "x", "f": "w"})
    assert not sel.matches({"a": "2", "c": "yes", "e": "x"})  # a mismatched
    assert not sel.matches({"a": "1", "e": "x"})  # c missing
    assert not sel.matches({"a": "1", "c": "1", "d": "1", "e": "x"})  # d present
    assert not sel.matches({"a": "1", "c": "1", "e": "q"})  # e not in set
    assert not sel.matches({"a": "1", "c": "1", "e": "x", "f": "z"})  # f in notin


def test_field_selector():
    obj = {"spec": {"nodeName": "n1"}, "metadata": {"name": "p"}}
    assert match_field_selector(obj, "spec.nodeName=n1")
    assert not match_field_selector(obj, "spec.nodeName=n2")
    assert match_field_selector(obj, "spec.nodeName=n1,metadata.name=p")
    assert match_field_selector(obj, "spec.nodeName!=n2")


def test_json_merge_patch():
    target = {"a": {"b": 1, "c": 2}, "d": 3}
    assert json_merge_patch(target, {"a": {"b": 9}, "e": 4}) == {
        "a": {"b": 9, "c": 2},
        "d": 3,
        "e": 4,
    }
    assert json_merge_patch(target, {"d": None}) == {"a": {"b": 1, "c": 2}}


# ------------------------------------------------------------------- objects


def test_conditions_transition_semantics():
    obj = {"metadata": {"generation": 3}, "status": {}}
    assert ko.set_condition(obj, "Ready", ko.CONDITION_FALSE, "NotReady")
    c1 = ko.get_condition(obj, "Ready")
    t1 = c1["lastTransitionTime"]
    # same status: no transition-time change
    ko.set_condition(obj, "Ready", ko.CONDITION_FALSE, "StillNotReady")
    assert ko.get_condition(obj, "Ready")["lastTransitionTime"] == t1
    assert ko.get_condition(obj, "Ready")["reason"] == "StillNotReady"
    assert not ko.condition_is_true(obj, "Ready")
    ko.set_condition(obj, "Ready", ko.CONDITION_TRUE, "Ready")
    assert ko.condition_is_true(obj, "Ready")


def test_taint_merge():
    existing = [{"key": "a", "effect": "NoSchedule", "value": "1"}]
    desired = [
        {"key": "a", "effect": "NoSchedule", "value": "2"},  # same identity — kept as existing
        {"key": "b", "effect": "NoExecute"},
    ]
    merged = ko.merge_taints(existing, desired)
    assert len(merged) == 2
    assert merged[0]["value"] == "1"
    assert merged[1]["key"] == "b"


def test_finalizer_helpers():
    obj = {"metadata": {"name": "x"}}
    assert ko.add_finalizer(obj, "f1")
    assert not ko.add_finalizer(obj, "f1")
    assert ko.has_finalizer(obj, "f1")
    assert ko.remove_finalizer(obj, "f1")
    assert not ko.remove_finalizer(obj, "f1")


# ----------------------------------------------------------------- workqueue


def test_workqueue_dedup_and_requeue_during_processing():
    async def main():
        q = RateLimitingQueue()
        await q.add("a")
        await q.add("a")  # dedup
        item = await q.get()
        assert item == "a"
        # re-add while processing: must come back after done()
        await q.add("a")
        assert q.depth == 0
        await q.done("a")
        assert await asyncio.wait_for(q.get(), 1) == "a"
        await q.done("a")

    run(main())


def test_workqueue_delayed_add_and_backoff():
    async def main():
        q = RateLimitingQueue(RateLimiter(base=0.01, cap=0.05))
        await q.add_rate_limited("x")  # first failure: ~10ms
        t0 = asyncio.get_event_loop().time()
        assert await q.get() == "x"
        assert asyncio.get_event_loop().time() - t0 >= 0.009
        await q.done("x")
        assert q.num_requeues("x") == 1
        q.forget("x")
        assert q.num_requeues("x") == 0

    run(main())


# ----------------------------------------------------------- in-memory server


def _mk(name, labels=None, finalizers=None):
    obj = {"apiVersion": "karpenter.sh/v1", "kind": "NodeClaim", "metadata": {"name": name}}
    if labels:
        obj["metadata"]["labels"] = labels
    if finalizers:
        obj["metadata"]["finalizers"] = finalizers
    return obj


def test_apiserver_crud_and_conflicts():
    async def main():
        c = InMemoryClient(InMemoryAPIServer())
        created = await c.create(_mk("a", labels={"x": "1"}))
        assert ko.uid_of(created) and created["metadata"]["resourceVersion"]
        with pytest.raises(AlreadyExistsError):
            await c.create(_mk("a"))
        got = await c.get("karpenter.sh/v1", "NodeClaim", "a")
        assert ko.labels_of(got) == {"x": "1"}
        # optimistic concurrency
        stale = ko.deep_copy(got)
        got["metadata"]["labels"]["x"] = "2"
        await c.update(got)
        stale["metadata"]["labels"]["x"] = "3"
        with pytest.raises(ConflictError):
            await c.update(stale)
        # list with selector
        await c.create(_mk("b", labels={"x": "2"}))
        assert len(await c.list("karpenter.sh/v1", "NodeClaim")) == 2
        assert len(await c.list("karpenter.sh/v1", "NodeClaim", label_selector="x=2")) == 2
        with pytest.raises(NotFoundError):
            await c.get("karpenter.sh/v1", "NodeClaim", "zzz")

    run(main())


def test_apiserver_finalizer_deletion_flow():
    async def main():
        c = InMemoryClient(InMemoryAPIServer())
        await c.create(_mk("a", finalizers=["keep"]))
        await c.delete("karpenter.sh/v1", "NodeClaim", "a")
        # still present, but deleting
        obj = await c.get("karpenter.sh/v1", "NodeClaim", "a")
        assert ko.is_deleting(obj)
        # second delete is a no-op
        await c.delete("karpenter.sh/v1", "NodeClaim", "a")
        # removing the finalizer releases the object
        ko.remove_finalizer(obj, "keep")
        await c.update(obj)
        with pytest.raises(NotFoundError):
            await c.get("karpenter.sh/v1", "NodeClaim", "a")

    run(main())


def test_apiserver_status_subresource_isolation():
    async def main():
        c = InMemoryClient(InMemoryAPIServer())
        await c.create(_mk("a"))
        obj = await c.get("karpenter.sh/v1", "NodeClaim", "a")
        # main-resource update cannot change status
        obj["status"] = {"providerID": "sneaky"}
        await c.update(obj)
        assert (await c.get("karpenter.sh/v1", "NodeClaim", "a")).get("status", {}) == {}
        # status patch only changes status
        await c.patch(
            "karpenter.sh/v1", "NodeClaim", "a", {"status": {"providerID": "pid"}},
            subresource="status",
        )
        got = await c.get("karpenter.sh/v1", "NodeClaim", "a")
        assert got["status"]["providerID"] == "pid"

    run(main())


def test_apiserver_generation_bumps_on_spec_change_only():
    async def main():
        c = InMemoryClient(InMemoryAPIServer())
        await c.create({**_mk("a"), "spec": {"v": 1}})
        obj = await c.get("karpenter.sh/v1", "NodeClaim", "a")
        assert obj["metadata"]["generation"] == 1
        ko.set_label(obj, "l", "1")
        obj = await c.update(obj)
        assert obj["metadata"]["generation"] == 1  # label change: no bump
        obj["spec"]["v"] = 2
        obj = await c.update(obj)
        assert obj["metadata"]["generation"] == 2

    run(main())


def test_apiserver_watch_stream_and_resume():
    async def main():
        server = InMemoryAPIServer()
        c = InMemoryClient(server)
        created = await c.create(_mk("a"))
        events = []

        async def consume():
            async for et, obj in c.watch("karpenter.sh/v1", "NodeClaim", resource_version="0"):
                events.append((et, ko.name_of(obj)))
                if len(events) >= 3:
                    return

        task = asyncio.create_task(consume())
        await asyncio.sleep(0.01)
        await c.create(_mk("b"))
        await c.delete("karpenter.sh/v1", "NodeClaim", "a")
        await asyncio.wait_for(task, 2)
        assert events == [(ADDED, "a"), (ADDED, "b"), (DELETED, "a")]

    run(main())


def test_apiserver_uid_precondition_delete():
    async def main():
        c = InMemoryClient(InMemoryAPIServer())
        created = await c.create(_mk("a"))
        with pytest.raises(ConflictError):
            await c.delete("karpenter.sh/v1", "NodeClaim", "a", uid_precondition="wrong-uid")
        await c.delete("karpenter.sh/v1", "NodeClaim", "a", uid_precondition=ko.uid_of(created))
        with pytest.raises(NotFoundError):
            await c.get("karpenter.sh/v1", "NodeClaim", "a")

    run(main())


def test_patch_with_resource_version_precondition():
    """A merge patch carrying metadata.resourceVersion is an optimistic-lock
    precondition (real apiserver semantics); without it, merge wins."""
    from gpu_provisioner_amd.fake.apiserver import InMemoryAPIServer, InMemoryClient
    from gpu_provisioner_amd.kube.client import ConflictError

    async def main():
        kube = InMemoryClient(InMemoryAPIServer())
        await kube.create(
            {"apiVersion": "v1", "kind": "Node", "metadata": {"name": "n1"},
             "spec": {}, "status": {}}
        )
        n1 = await kube.get("v1", "Node", "n1")
        stale_rv = n1["metadata"]["resourceVersion"]
        await kube.patch("v1", "Node", "n1", {"metadata": {"labels": {"a": "1"}}})
        with pytest.raises(ConflictError):
            await kube.patch(
                "v1", "Node", "n1",
                {"metadata": {"resourceVersion": stale_rv, "labels": {"b": "2"}}},
            )
        # no precondition: merge wins over the stale read
        got = await kube.patch("v1", "Node", "n1", {"metadata": {"labels": {"c": "3"}}})
        assert got["metadata"]["labels"]["c"] == "3"

    run(main())


def test_lifecycle_status_patch_preserves_foreign_conditions():
    """A condition written by another controller (e.g. Drifted) between the
    lifecycle controller's read and its status patch must survive — the
    conflict-retry path re-grafts only the fields lifecycle owns."""
    from gpu_provisioner_amd.apis import v1 as karpv1
    from gpu_provisioner_amd.fake.harness import Harness

    async def main():
        h = Harness().add_all_controllers(gc_interval=60.0, with_drift=False)
        # don't start controllers: drive _patch_status by hand
        nc = h.make_nodeclaim("race1")
        await h.kube.create(nc)
        in_hand = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "race1")
        ko.set_condition(in_hand, karpv1.COND_LAUNCHED, ko.CONDITION_TRUE, "Launched")

        # a foreign writer lands between lifecycle's read and its patch
        foreign = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "race1")
        ko.set_condition(foreign, karpv1.COND_DRIFTED, ko.CONDITION_TRUE, "NodeImageDrift")
        await h.kube.update_status(foreign)

        await h.lifecycle._patch_status(in_hand)
        final = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "race1")
        assert ko.condition_is_true(final, karpv1.COND_LAUNCHED)
        assert ko.condition_is_true(final, karpv1.COND_DRIFTED), "foreign condition clobbered"

    run(main())
