"""node.termination tests: drain ordering, PDB retries, grace clamping,
volume-detach wait, finalizer interplay with the NodeClaim lifecycle.
Behavioral spec: reference vendor/.../controllers/node/termination/."""
import asyncio


from gpu_provisioner_amd.apis import v1 as karpv1
from gpu_provisioner_amd.fake.harness import Harness
from gpu_provisioner_amd.kube import objects as ko
from gpu_provisioner_amd.kube.client import TooManyRequestsError
from gpu_provisioner_amd.utils import pod as podutils
from tests.conftest import run


def make_harness(**kw) -> Harness:
    return Harness(**kw).add_all_controllers(with_health=False)


def mk_pod(name, node, *, daemon=False, critical=False, phase="Running", ns="default"):
    pod = {
        "apiVersion": "v1",
        "kind": "Pod",
        "metadata": {"name": name, "namespace": ns},
        "spec": {"nodeName": node},
        "status": {"phase": phase},
    }
    if daemon:
        pod["metadata"]["ownerReferences"] = [
            {"apiVersion": "apps/v1", "kind": "DaemonSet", "name": "ds", "uid": "u1"}
        ]
    if critical:
        pod["spec"]["priorityClassName"] = "system-cluster-critical"
    return pod


# ---------------------------------------------------------------- pod utils


def test_eviction_grouping_order():
    pods = [
        mk_pod("a", "n"),                      # group 0
        mk_pod("b", "n", daemon=True),          # group 1
        mk_pod("c", "n", critical=True),        # group 2
        mk_pod("d", "n", daemon=True, critical=True),  # group 3
    ]
    group = podutils.group_for_eviction(pods)
    assert [ko.name_of(p) for p in group] == ["a"]
    group = podutils.group_for_eviction(pods[1:])
    assert [ko.name_of(p) for p in group] == ["b"]
    assert podutils.group_for_eviction([]) == []
    # terminal pods don't drain
    assert podutils.group_for_eviction([mk_pod("x", "n", phase="Succeeded")]) == []


def test_grace_clamp():
    pod = {"spec": {"terminationGracePeriodSeconds": 300}}
    assert podutils.clamp_grace_period(pod, None) is None
    assert podutils.clamp_grace_period(pod, 60.0) == 60
    assert podutils.clamp_grace_period(pod, 1000.0) == 300
    assert podutils.clamp_grace_period({}, 45.0) == 45


def test_parse_duration():
    from gpu_provisioner_amd.controllers.termination.controller import parse_duration

    assert parse_duration("30s").total_seconds() == 30
    assert parse_duration("5m").total_seconds() == 300
    assert parse_duration("1h30m").total_seconds() == 5400
    assert parse_duration("") is None


# ------------------------------------------------------------- node delete


def test_node_delete_drains_then_terminates():
    """kubectl delete node → taint, evict workload pods, delete NodeClaim +
    instance, node gone (reference e2e terminate-via-node-delete spec)."""

    async def main():
        h = make_harness()
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("t1"))
            done = await h.wait_initialized("t1")
            node_name = done["status"]["nodeName"]
            # workload + daemon pods on the node
            await h.kube.create(mk_pod("w1", node_name))
            await h.kube.create(mk_pod("w2", node_name))
            await h.kube.create(mk_pod("ds1", node_name, daemon=True))
            await h.kube.delete("v1", "Node", node_name)
            await h.wait_gone("v1", "Node", node_name)
            await h.wait_gone(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "t1")
            assert "t1" not in h.agent_pools.pools
            # both workload pods were evicted through the eviction subresource
            evicted = {n for (_, n) in h.server.evictions}
            assert {"w1", "w2"} <= evicted
        finally:
            await h.stop()

    run(main())


def test_node_delete_taints_before_drain():
    async def main():
        h = make_harness()
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("t2"))
            done = await h.wait_initialized("t2")
            node_name = done["status"]["nodeName"]
            # a pod with a finalizer holds the drain open so we can observe the taint
            pod = mk_pod("w1", node_name)
            pod["metadata"]["finalizers"] = ["hold"]
            await h.kube.create(pod)
            await h.kube.delete("v1", "Node", node_name)

            async def tainted():
                n = await h.kube.get("v1", "Node", node_name)
                return (
                    n
                    if any(
                        t.get("key") == karpv1.DISRUPTED_TAINT_KEY
                        for t in ko.node_taints(n)
                    )
                    else None
                )

            node = await h.wait_for(tainted)
            assert (
                ko.labels_of(node)[karpv1.EXCLUDE_FROM_LB_LABEL_KEY] == "karpenter"
            )
            # NodeClaim shows Drained=Unknown while pods remain
            nc = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "t2")
            # release the pod; teardown completes
            held = await h.kube.get("v1", "Pod", "w1", "default")
            held["metadata"]["finalizers"] = []
            await h.kube.update(held)
            await h.wait_gone("v1", "Node", node_name)
        finally:
            await h.stop()

    run(main())


def test_pdb_blocked_eviction_retries():
    """A PDB 429 on eviction retries with backoff until it succeeds
    (reference eviction.go:140-175)."""

    async def main():
        h = make_harness()
        attempts = {"n": 0}

        def pdb_reactor(pod):
            if ko.name_of(pod) == "w1" and attempts["n"] < 3:
                attempts["n"] += 1
                return TooManyRequestsError("pdb violation", 0.01)
            return None

        h.server.eviction_reactor = pdb_reactor
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("t3"))
            done = await h.wait_initialized("t3")
            node_name = done["status"]["nodeName"]
            await h.kube.create(mk_pod("w1", node_name))
            await h.kube.delete("v1", "Node", node_name)
            await h.wait_gone("v1", "Node", node_name, timeout=20)
            assert attempts["n"] == 3
        finally:
            await h.stop()

    run(main())


def test_volume_detachment_blocks_until_gone():
    async def main():
        h = make_harness()
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("t4"))
            done = await h.wait_initialized("t4")
            node_name = done["status"]["nodeName"]
            va = {
                "apiVersion": "storage.k8s.io/v1",
                "kind": "VolumeAttachment",
                "metadata": {"name": "va1"},
                "spec": {"nodeName": node_name},
            }
            await h.kube.create(va)
            await h.kube.delete("v1", "Node", node_name)
            # VolumesDetached goes Unknown while attachment remains
            await asyncio.sleep(0.4)
            node_still = await h.kube.get("v1", "Node", node_name)
            assert ko.is_deleting(node_still)
            nc = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "t4")
            cond = ko.get_condition(nc, karpv1.COND_VOLUMES_DETACHED)
            assert cond is not None and cond["status"] == ko.CONDITION_UNKNOWN
            await h.kube.delete("storage.k8s.io/v1", "VolumeAttachment", "va1")
            await h.wait_gone("v1", "Node", node_name)
        finally:
            await h.stop()

    run(main())


def test_termination_grace_period_cuts_volume_wait():
    """With spec.terminationGracePeriod elapsed, volume waits are skipped
    (reference awaitVolumeDetachment TGP override)."""

    async def main():
        h = make_harness()
        await h.start()
        try:
            nc = h.make_nodeclaim("t5")
            nc["spec"]["terminationGracePeriod"] = "1s"
            await h.kube.create(nc)
            done = await h.wait_initialized("t5")
            node_name = done["status"]["nodeName"]
            va = {
                "apiVersion": "storage.k8s.io/v1",
                "kind": "VolumeAttachment",
                "metadata": {"name": "va2"},
                "spec": {"nodeName": node_name},
            }
            await h.kube.create(va)
            # delete the NodeClaim: deletionTimestamp + TGP 1s ⇒ deadline in 1s
            await h.kube.delete(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "t5")
            # despite the stuck attachment, teardown completes once TGP elapses
            await h.wait_gone("v1", "Node", node_name, timeout=15)
            await h.wait_gone(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "t5", timeout=15)
        finally:
            await h.stop()

    run(main())
