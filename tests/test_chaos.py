"""Chaos convergence: random retryable ARM failures on every cloud verb
while a 24-claim fleet provisions and tears down. Level-triggered
controllers with rate-limited requeues must converge despite the noise —
the distributed-systems property the reference delegates to live-cluster
e2e, exercised here in-process with a seeded RNG."""
import asyncio
import os
import random

# campaign knob: CHAOS_SEED varies the fault interleavings (fixed defaults
# keep CI deterministic)
def _seed(default: int) -> int:
    return int(os.environ.get("CHAOS_SEED", default))

from gpu_provisioner_amd.apis import v1 as karpv1
from gpu_provisioner_amd.fake.harness import Harness
from gpu_provisioner_amd.kube import objects as ko
from gpu_provisioner_amd.providers.instance.armapi import ARMError
from tests.conftest import run

FLEET = 24
FAIL_P = 0.15


class ChaosError:
    """Drop-in for fake.agentpools.ScriptedError: raises a retryable ARM
    error with probability p (seeded — the test is deterministic)."""

    def __init__(self, rng: random.Random, p: float = FAIL_P):
        self.rng = rng
        self.p = p
        self.raised = 0

    def check(self) -> None:
        if self.rng.random() < self.p:
            self.raised += 1
            raise ARMError(503, "ServerBusy", "chaos: simulated ARM brownout")


def test_fleet_converges_under_arm_chaos():
    async def main():
        rng = random.Random(_seed(20260913))
        h = Harness(node_wait_interval=0.005).add_all_controllers(
            lifecycle_workers=64,
            termination_requeue=0.01,
            drain_requeue=0.01,
            instance_poll=0.01,
            gc_interval=1.0,
            adoption_age=0.5,
        )
        chaos = [ChaosError(rng) for _ in range(4)]
        (
            h.agent_pools.create_error,
            h.agent_pools.delete_error,
            h.agent_pools.get_error,
            h.agent_pools.list_error,
        ) = chaos
        await h.start()
        try:
            names = [f"chaos{i:02d}" for i in range(FLEET)]
            await asyncio.gather(
                *(h.kube.create(h.make_nodeclaim(n)) for n in names)
            )
            done = await asyncio.gather(
                *(h.wait_initialized(n, timeout=60) for n in names)
            )
            assert all(karpv1.is_initialized(nc) for nc in done)
            if "CHAOS_SEED" not in os.environ:  # guard only the CI seed
                assert sum(c.raised for c in chaos) > 0, "chaos never fired — test is vacuous"

            await asyncio.gather(
                *(
                    h.kube.delete(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, n)
                    for n in names
                )
            )
            await asyncio.gather(
                *(
                    h.wait_gone(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, n, timeout=60)
                    for n in names
                )
            )

            async def pools_empty():
                return not h.agent_pools.pools or None

            await h.wait_for(pools_empty, timeout=60)
            # no leaked nodes either
            assert await h.kube.list("v1", "Node") == []
        finally:
            await h.stop()

    run(main(), timeout=240)


def test_fleet_converges_under_kube_and_arm_chaos():
    """Both backends misbehave at once: every ARM verb AND every kube write
    verb (update/patch/delete) fails randomly with retryable errors. The
    stack must still converge a 16-claim fleet through provision+teardown."""
    from gpu_provisioner_amd.kube.client import ConflictError, TooManyRequestsError

    async def main():
        rng = random.Random(_seed(777))
        h = Harness(node_wait_interval=0.005).add_all_controllers(
            lifecycle_workers=64,
            termination_requeue=0.01,
            drain_requeue=0.01,
            instance_poll=0.01,
            gc_interval=1.0,
            adoption_age=0.5,
        )
        arm = [ChaosError(rng, p=0.10) for _ in range(4)]
        (
            h.agent_pools.create_error,
            h.agent_pools.delete_error,
            h.agent_pools.get_error,
            h.agent_pools.list_error,
        ) = arm
        kube_fired = {"n": 0}

        def kube_chaos(verb, gvk, payload):
            # eviction uses its own path; pods excluded so drain's PDB retry
            # semantics stay deterministic here
            if verb in ("update", "patch", "delete") and gvk[1] != "Pod":
                if rng.random() < 0.08:
                    kube_fired["n"] += 1
                    return (
                        ConflictError("chaos: conflict")
                        if rng.random() < 0.5
                        else TooManyRequestsError("chaos: throttled")
                    )
            return None

        h.server.reactors.append(kube_chaos)
        await h.start()
        try:
            names = [f"kchaos{i:02d}" for i in range(16)]
            await asyncio.gather(*(h.kube.create(h.make_nodeclaim(n)) for n in names))
            done = await asyncio.gather(
                *(h.wait_initialized(n, timeout=240) for n in names)
            )
            assert all(karpv1.is_initialized(nc) for nc in done)
            if "CHAOS_SEED" not in os.environ:  # guard only the CI seed
                assert kube_fired["n"] > 0 and sum(c.raised for c in arm) > 0

            async def chaos_tolerant_delete(n):
                # the test's own client retries like any well-behaved caller
                for _ in range(50):
                    try:
                        await h.kube.delete(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, n)
                        return
                    except Exception:
                        await asyncio.sleep(0.02)

            await asyncio.gather(*(chaos_tolerant_delete(n) for n in names))
            await asyncio.gather(
                *(
                    h.wait_gone(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, n, timeout=240)
                    for n in names
                )
            )

            async def pools_empty():
                return not h.agent_pools.pools or None

            await h.wait_for(pools_empty, timeout=240)
        finally:
            await h.stop()

    run(main(), timeout=600)


def test_fleet_converges_while_watch_streams_drop():
    """Every open watch stream drops repeatedly mid-provision (apiserver
    timeout / netsplit shape): informers must relist+rewatch and the fleet
    must still converge with no duplicate side effects."""

    async def main():
        h = Harness(node_wait_interval=0.005).add_all_controllers(
            lifecycle_workers=64,
            termination_requeue=0.01,
            drain_requeue=0.01,
            instance_poll=0.01,
            gc_interval=1.0,
            adoption_age=0.5,
        )
        await h.start()
        try:
            names = [f"wchaos{i:02d}" for i in range(12)]
            await asyncio.gather(*(h.kube.create(h.make_nodeclaim(n)) for n in names))
            broken = 0
            for _ in range(5):
                await asyncio.sleep(0.05)
                broken += h.server.break_watches()
            assert broken > 0
            done = await asyncio.gather(
                *(h.wait_initialized(n, timeout=60) for n in names)
            )
            assert all(karpv1.is_initialized(nc) for nc in done)
            # exactly one pool and one create per claim despite the relists
            assert sorted(h.agent_pools.pools) == sorted(names)
            assert h.agent_pools.create_calls == len(names)

            await asyncio.gather(
                *(
                    h.kube.delete(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, n)
                    for n in names
                )
            )
            h.server.break_watches()
            await asyncio.gather(
                *(
                    h.wait_gone(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, n, timeout=60)
                    for n in names
                )
            )
        finally:
            await h.stop()

    run(main(), timeout=240)


# Note on budgets: per-key failure backoff doubles to a 30s cap (client-go
# semantics), so a claim drawing a long streak of chaotic failures can sit
# out several backoff windows; the waits above cover worst-case stacking,
# not typical convergence (sub-second on the passing path).


def test_full_topology_chaos_with_repair_and_drift():
    """All controllers on (health repair with short tolerations, drift
    replacement, both GCs) while the environment misbehaves: GPU health
    conditions flap sick/healthy, pools mutate out-of-band (drift), and
    claims are deleted mid-life. The fleet must end with every surviving
    claim Initialized on a conforming pool, and teardown must leave
    nothing behind."""
    from gpu_provisioner_amd.nodeagent import (
        GPUReport,
        NodeReport,
        patch_node_condition,
    )

    def sick():
        return NodeReport(healthy=False, gpu_count=8,
                          gpus=[GPUReport(index=0, problems=["MFMA fail"])],
                          problems=["gpu0: MFMA fail"])

    def healthy():
        return NodeReport(healthy=True, gpu_count=8,
                          gpus=[GPUReport(index=0, healthy=True)])

    async def main():
        rng = random.Random(_seed(99))
        h = Harness(
            node_wait_interval=0.005, gpu_repair_toleration=0.4,
        ).add_all_controllers(
            gc_interval=0.5, adoption_age=0.3, drift_interval=0.2,
            drift_replace=True, termination_requeue=0.01, drain_requeue=0.01,
            instance_poll=0.01,
        )
        await h.start()
        try:
            names = [f"topo{i:02d}" for i in range(12)]
            await asyncio.gather(*(h.kube.create(h.make_nodeclaim(n)) for n in names))
            await asyncio.gather(*(h.wait_initialized(n, timeout=60) for n in names))

            # 2 seconds of mayhem
            for _ in range(10):
                await asyncio.sleep(0.2)
                n = rng.choice(names)
                action = rng.random()
                try:
                    if action < 0.4 and n in h.agent_pools.pools:
                        # out-of-band pool mutation → drift → replacement
                        h.agent_pools.pools[n]["properties"]["osSKU"] = "AzureLinux"
                    elif action < 0.7:
                        nc = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, n)
                        node_name = nc.get("status", {}).get("nodeName")
                        if node_name:
                            # GPU sickness flap: sick now, healthy shortly after
                            await patch_node_condition(h.kube, node_name, sick())
                            if rng.random() < 0.5:
                                await asyncio.sleep(0.1)
                                await patch_node_condition(h.kube, node_name, healthy())
                    else:
                        await h.kube.delete(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, n)
                except Exception:
                    pass  # races with replacement are part of the chaos

            # quiesce: every surviving claim must be Initialized on a
            # conforming (non-drifted) pool; every pool must belong to a claim
            async def settled():
                claims = [
                    nc for nc in await h.kube.list(karpv1.API_VERSION, karpv1.KIND_NODECLAIM)
                    if not ko.is_deleting(nc)
                ]
                if any(not karpv1.is_initialized(nc) for nc in claims):
                    return None
                live = {ko.name_of(nc) for nc in claims}
                if set(h.agent_pools.pools) != live:
                    return None
                for name in live:
                    if h.agent_pools.pools[name]["properties"].get("osSKU") == "AzureLinux":
                        return None  # drifted pool still awaiting replacement
                return claims

            claims = await h.wait_for(settled, timeout=120, interval=0.1)

            # full teardown
            await asyncio.gather(
                *(
                    h.kube.delete(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, ko.name_of(nc))
                    for nc in claims
                )
            )

            async def empty():
                pools = not h.agent_pools.pools
                claims_left = await h.kube.list(karpv1.API_VERSION, karpv1.KIND_NODECLAIM)
                nodes_left = await h.kube.list("v1", "Node")
                return (pools and not claims_left and not nodes_left) or None

            await h.wait_for(empty, timeout=120, interval=0.1)
        finally:
            await h.stop()

    run(main(), timeout=400)


def test_delete_recreate_same_name_while_watches_broken():
    """A claim is deleted AND recreated (same name, new uid) while every
    watch stream is down: the relist coalesces the transition into a
    MODIFIED for the same key. The recreated claim must converge — the
    old pool (same agent-pool name) is either adopted or GC'd and
    recreated, and the final state belongs to the NEW uid."""

    async def main():
        h = Harness(node_wait_interval=0.005).add_all_controllers(
            gc_interval=0.3, adoption_age=0.2, termination_requeue=0.01,
            drain_requeue=0.01, instance_poll=0.01,
        )
        await h.start()
        try:
            await h.kube.create(h.make_nodeclaim("phoenix"))
            first = await h.wait_initialized("phoenix")
            old_uid = first["metadata"]["uid"]

            h.server.break_watches()
            # while informers relist: tear down and recreate under the
            # same name (finalizer flow still runs server-side)
            await h.kube.delete(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "phoenix")

            async def old_gone():
                try:
                    nc = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "phoenix")
                    return None if nc["metadata"]["uid"] == old_uid else True
                except Exception:
                    return True

            await h.wait_for(old_gone, timeout=30)
            h.server.break_watches()
            await h.kube.create(h.make_nodeclaim("phoenix"))
            h.server.break_watches()

            async def new_ready():
                try:
                    nc = await h.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, "phoenix")
                except Exception:
                    return None
                if nc["metadata"]["uid"] == old_uid:
                    return None
                return nc if karpv1.is_initialized(nc) else None

            fresh = await h.wait_for(new_ready, timeout=60, interval=0.05)
            assert fresh["metadata"]["uid"] != old_uid
            assert "phoenix" in h.agent_pools.pools
            # exactly one claim and its pool; no zombie node from the old life
            claims = await h.kube.list(karpv1.API_VERSION, karpv1.KIND_NODECLAIM)
            assert len(claims) == 1
            nodes = [
                n for n in await h.kube.list("v1", "Node")
                if ko.labels_of(n).get("agentpool") == "phoenix"
            ]
            assert len(nodes) == 1, [ko.name_of(n) for n in nodes]
        finally:
            await h.stop()

    run(main(), timeout=240)
