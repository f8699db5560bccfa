"""Real-apiserver conformance tier for the in-memory fake.

Kubebuilder envtest binaries (etcd + kube-apiserver) are not obtainable in
this offline environment, so divergences between `fake/apiserver.py` and a
real kube-apiserver are instead ENCODED here as an executable semantics
matrix: each test states a documented behavior of the real apiserver (with
the upstream source or API-convention citation) and asserts the fake
implements it. When a live cluster is available, the same expectations run
against it through tests/e2e_env.py's live backend.

This tier exists because round 1 shipped a latent bug a real apiserver would
have caught on day one: a `spec.nodeName` field selector on VolumeAttachment
(unsupported → HTTP 400) silently skipped the volume-detach wait.
"""
import asyncio

import pytest

from gpu_provisioner_amd.apis import v1 as karpv1
from gpu_provisioner_amd.fake.apiserver import InMemoryAPIServer, InMemoryClient
from gpu_provisioner_amd.fake.harness import Harness
from gpu_provisioner_amd.kube import objects as ko
from gpu_provisioner_amd.kube.client import ConflictError, InvalidError
from gpu_provisioner_amd.kube.informer import object_key
from tests.conftest import run


def client():
    return InMemoryClient(InMemoryAPIServer())


# ---------------------------------------------------------------------------
# Field-selector support matrix
# (upstream pkg/registry/<group>/<resource>/strategy.go ToSelectableFields;
#  CRDs without spec.selectableFields accept only metadata.name/namespace)
# ---------------------------------------------------------------------------


def test_field_selector_pod_spec_nodename_supported():
    """Pod registers spec.nodeName as selectable (upstream
    pkg/registry/core/pod/strategy.go)."""

    async def main():
        kube = client()
        await kube.create(
            {"apiVersion": "v1", "kind": "Pod",
             "metadata": {"name": "p1", "namespace": "default"},
             "spec": {"nodeName": "n1"}}
        )
        await kube.create(
            {"apiVersion": "v1", "kind": "Pod",
             "metadata": {"name": "p2", "namespace": "default"},
             "spec": {"nodeName": "n2"}}
        )
        got = await kube.list("v1", "Pod", field_selector="spec.nodeName=n1")
        assert [ko.name_of(p) for p in got] == ["p1"]

    run(main())


def test_field_selector_volumeattachment_spec_nodename_rejected():
    """VolumeAttachment registers NO selectable fields beyond the generic
    metadata ones — a real apiserver answers `field label not supported:
    spec.nodeName` with HTTP 400 (upstream
    pkg/registry/storage/volumeattachment has no ToSelectableFields)."""

    async def main():
        kube = client()
        with pytest.raises(InvalidError):
            await kube.list(
                "storage.k8s.io/v1", "VolumeAttachment",
                field_selector="spec.nodeName=n1",
            )

    run(main())


def test_field_selector_crd_only_metadata():
    """CRDs (NodeClaim) without spec.selectableFields accept only
    metadata.name/metadata.namespace selectors (KEP-4358; selectableFields
    is opt-in since v1.31 and our CRD does not declare any)."""

    async def main():
        kube = client()
        await kube.create(karpv1.new_nodeclaim("c1", labels={}))
        got = await kube.list(
            karpv1.API_VERSION, karpv1.KIND_NODECLAIM,
            field_selector="metadata.name=c1",
        )
        assert [ko.name_of(o) for o in got] == ["c1"]
        with pytest.raises(InvalidError):
            await kube.list(
                karpv1.API_VERSION, karpv1.KIND_NODECLAIM,
                field_selector="status.providerID=azure:///x",
            )

    run(main())


def test_field_selector_node_unschedulable_supported():
    """Node registers spec.unschedulable (upstream
    pkg/registry/core/node/strategy.go)."""

    async def main():
        kube = client()
        await kube.create(
            {"apiVersion": "v1", "kind": "Node", "metadata": {"name": "n1"},
             "spec": {"unschedulable": True}}
        )
        got = await kube.list("v1", "Node", field_selector="spec.unschedulable=True")
        assert len(got) == 1
        with pytest.raises(InvalidError):
            await kube.list("v1", "Node", field_selector="spec.providerID=x")

    run(main())


# ---------------------------------------------------------------------------
# Write semantics the controllers rely on
# (Kubernetes API conventions: status is a subresource; JSON merge patch
#  RFC 7386 replaces lists wholesale; rv preconditions → 409)
# ---------------------------------------------------------------------------


def test_status_subresource_isolated_from_main_update():
    async def main():
        kube = client()
        nc = await kube.create(karpv1.new_nodeclaim("s1", labels={}))
        nc["status"] = {"providerID": "azure:///x"}
        await kube.update_status(nc)
        fresh = await kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "s1")
        # main-resource update cannot change status (API conventions:
        # status is only writable through the /status subresource)
        fresh["status"] = {"providerID": "azure:///CLOBBER"}
        fresh["metadata"]["labels"] = {"a": "b"}
        await kube.update(fresh)
        after = await kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "s1")
        assert after["status"]["providerID"] == "azure:///x"
        assert after["metadata"]["labels"] == {"a": "b"}

    run(main())


def test_merge_patch_replaces_lists_wholesale():
    """RFC 7386: arrays in a merge patch REPLACE the target array — patching
    conditions/finalizers from a stale read erases concurrent additions,
    which is why every conditions write in the controllers carries an rv
    precondition."""

    async def main():
        kube = client()
        nc = await kube.create(karpv1.new_nodeclaim("m1", labels={}))
        nc["status"] = {"conditions": [
            {"type": "A", "status": "True"}, {"type": "B", "status": "True"},
        ]}
        await kube.update_status(nc)
        await kube.patch(
            karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "m1",
            {"status": {"conditions": [{"type": "A", "status": "False"}]}},
            subresource="status",
        )
        after = await kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "m1")
        assert after["status"]["conditions"] == [{"type": "A", "status": "False"}]

    run(main())


def test_rv_preconditioned_patch_conflicts_on_stale_read():
    async def main():
        kube = client()
        nc = await kube.create(karpv1.new_nodeclaim("rv1", labels={}))
        stale_rv = nc["metadata"]["resourceVersion"]
        await kube.patch(
            karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "rv1",
            {"metadata": {"labels": {"x": "1"}}},
        )
        with pytest.raises(ConflictError):
            await kube.patch(
                karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "rv1",
                {"metadata": {"resourceVersion": stale_rv, "labels": {"y": "2"}}},
            )

    run(main())


# ---------------------------------------------------------------------------
# Shared-informer cache integrity (client-go contract: cached objects are
# read-only; any in-place mutation by a controller corrupts every reader)
# ---------------------------------------------------------------------------


def _install_cache_guards(h: Harness) -> dict:
    """Record a deep snapshot of every object at the moment the informer
    stores it; any later divergence between cache and snapshot proves an
    in-place mutation by a cache consumer."""
    guards = {}
    for inf in h.informers._informers.values():
        snaps: dict = {}
        orig = inf._store

        def make(orig_store, snaps_map):
            def store(obj, event):
                orig_store(obj, event)
                snaps_map[object_key(obj)] = ko.deep_copy(obj)
            return store

        inf._store = make(orig, snaps)
        guards[inf.kind] = (inf, snaps)
    return guards


def test_controllers_never_mutate_informer_cache_in_place():
    async def main():
        h = Harness(ready_latency=0.02, plugin_latency=0.02).add_all_controllers(
            gc_interval=0.3, adoption_age=0.1
        )
        guards = _install_cache_guards(h)
        await h.start()
        try:
            # full provision cycle incl. health-relevant node conditions
            await h.kube.create(h.make_nodeclaim("guard1"))
            await h.wait_initialized("guard1")
            # one GC + drift sweep pass over live objects
            await asyncio.sleep(0.5)
            for kind, (inf, snaps) in guards.items():
                for key, obj in inf._cache.items():
                    assert key in snaps, f"{kind} {key} stored without snapshot"
                    assert obj == snaps[key], (
                        f"{kind} {key}: informer cache object mutated in place"
                    )
        finally:
            await h.stop()

    run(main())


def test_no_new_finalizers_on_terminating_object():
    """Real apiservers reject adding finalizers to a deleting object with
    422 (registry/rest update validation: 'no new finalizers can be added
    if the object is being deleted'); removing finalizers stays allowed."""

    async def main():
        kube = client()
        nc = karpv1.new_nodeclaim("fin1", labels={})
        nc["metadata"]["finalizers"] = ["keep.example.com/one"]
        await kube.create(nc)
        await kube.delete(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "fin1")
        cur = await kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "fin1")
        assert ko.is_deleting(cur)
        # adding → 422
        cur["metadata"]["finalizers"] = ["keep.example.com/one", "new.example.com/two"]
        with pytest.raises(InvalidError, match="no new finalizers"):
            await kube.update(cur)
        # removing → allowed, and removing the last one completes deletion
        cur = await kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "fin1")
        cur["metadata"]["finalizers"] = []
        await kube.update(cur)
        with pytest.raises(Exception):
            await kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "fin1")

    run(main())


# ---------------------------------------------------------------------------
# Server-side CRD schema validation (real apiservers validate CR writes
# against the installed CRD's structural schema + CEL; the harness server
# carries the CHART's CRDs — fake/harness.install_chart_crd_validators)
# ---------------------------------------------------------------------------


def _valid_spec():
    return {
        "requirements": [
            {"key": karpv1.INSTANCE_TYPE_LABEL_KEY, "operator": "In",
             "values": ["Standard_ND128isr_MI355X_v6"]}
        ],
        "nodeClassRef": {"group": "kaito.sh", "kind": "KaitoNodeClass", "name": "d"},
    }


def test_crd_schema_enforced_on_create():
    async def main():
        h = Harness()
        # missing required spec.requirements/nodeClassRef → 422
        with pytest.raises(InvalidError, match="required"):
            await h.kube.create(karpv1.new_nodeclaim("bad1", labels={}))
        # invalid requirement operator enum → 422
        nc = karpv1.new_nodeclaim("bad2", labels={})
        nc["spec"] = _valid_spec()
        nc["spec"]["requirements"][0]["operator"] = "Matches"
        with pytest.raises(InvalidError, match="enum"):
            await h.kube.create(nc)
        # empty nodeClassRef.kind → CEL non-empty rule
        nc = karpv1.new_nodeclaim("bad3", labels={})
        nc["spec"] = _valid_spec()
        nc["spec"]["nodeClassRef"]["kind"] = ""
        with pytest.raises(InvalidError, match="empty"):
            await h.kube.create(nc)
        # a valid claim is admitted and gets the expireAfter DEFAULT applied
        good = karpv1.new_nodeclaim("good1", labels={})
        good["spec"] = _valid_spec()
        created = await h.kube.create(good)
        assert created["spec"]["expireAfter"] == "720h"

    run(main())


def test_crd_spec_immutability_cel_enforced():
    async def main():
        h = Harness()
        nc = karpv1.new_nodeclaim("imm1", labels={})
        nc["spec"] = _valid_spec()
        created = await h.kube.create(nc)
        created["spec"]["requirements"][0]["values"] = ["Standard_ND64is_MI355X_v6"]
        with pytest.raises(InvalidError, match="immutable"):
            await h.kube.update(created)
        # metadata/status writes remain allowed
        fresh = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "imm1")
        fresh["metadata"]["labels"] = {"a": "b"}
        await h.kube.update(fresh)
        fresh = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "imm1")
        fresh["status"] = {"providerID": "azure:///x"}
        await h.kube.update_status(fresh)

    run(main())


def test_crd_quantity_pattern_enforced_on_status():
    async def main():
        h = Harness()
        nc = karpv1.new_nodeclaim("qty1", labels={})
        nc["spec"] = _valid_spec()
        created = await h.kube.create(nc)
        created["status"] = {"capacity": {"amd.com/gpu": "eight"}}  # not a quantity
        with pytest.raises(InvalidError):
            await h.kube.update_status(created)
        created["status"] = {"capacity": {"amd.com/gpu": "8", "memory": "2048Gi"}}
        await h.kube.update_status(created)

    run(main())


def test_delete_uid_precondition_conflicts():
    """DeleteOptions.preconditions.uid mismatch → 409 Conflict (API
    conventions; the GC/drift/repair deletes all carry the observed uid so
    a name-reuse race can never delete the replacement object)."""

    async def main():
        kube = client()
        nc = await kube.create(karpv1.new_nodeclaim("uid1", labels={}))
        with pytest.raises(ConflictError):
            await kube.delete(
                karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "uid1",
                uid_precondition="some-other-uid",
            )
        # matching uid deletes
        await kube.delete(
            karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "uid1",
            uid_precondition=nc["metadata"]["uid"],
        )
        with pytest.raises(Exception):
            await kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "uid1")

    run(main())
