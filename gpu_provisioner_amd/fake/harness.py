"""Full-stack in-process harness: apiserver + fake AKS + real controllers.

Wires the real instance provider, cloudprovider adapter (metrics-decorated),
and controllers against the in-memory apiserver and the AKS simulator —
BASELINE.json config #1 ("Single NodeClaim reconciled against fake
cloudprovider, CPU-only plumbing") and the substrate for bench.py. The
reference's analogue is its envtest + gomock seams (pkg/fake/), but here the
whole provision path runs e2e in one process.
"""
from __future__ import annotations

import asyncio
from typing import Optional

from ..apis import v1 as karpv1
from ..cloudprovider.azure import AzureCloudProvider
from ..cloudprovider.decorator import MetricsDecorator
from ..events.recorder import EventRecorder
from ..kube.informer import Informer, InformerFactory
from ..kube import objects as ko
from ..providers.instance.provider import InstanceProvider
from ..providers.instancetype.catalog import InstanceTypeProvider
from .agentpools import AKSSimulator, FakeAgentPools
from .apiserver import InMemoryAPIServer, InMemoryClient


def install_chart_crd_validators(server: InMemoryAPIServer) -> None:
    """Server-side CRD schema validation from the CHART's CRDs — the same
    schemas a real cluster installs — so controller writes a real
    apiserver would 422 fail here too."""
    import os

    from ..kube.crdschema import CRDValidator

    crds_dir = os.path.join(
        os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))),
        "charts",
        "gpu-provisioner-amd",
        "crds",
    )
    mapping = {
        ("karpenter.sh/v1", "NodeClaim"): "karpenter.sh_nodeclaims.yaml",
        ("kaito.sh/v1alpha1", "KaitoNodeClass"): "kaito.sh_kaitonodeclasses.yaml",
    }
    for gvk, fname in mapping.items():
        path = os.path.join(crds_dir, fname)
        if os.path.exists(path):
            server.validators[gvk] = CRDValidator.from_file(path)


class Harness:
    def __init__(
        self,
        *,
        create_latency: float = 0.0,
        delete_latency: float = 0.0,
        api_latency: float = 0.0,
        ready_latency: float = 0.0,
        plugin_latency: float = 0.0,
        node_wait_interval: float = 0.02,
        region: str = "eastus2",
        repair_toleration: Optional[float] = None,
        gpu_repair_toleration: Optional[float] = None,
        gc_pacer: bool = True,
    ):
        self._gc_pacer_enabled = gc_pacer
        self.server = InMemoryAPIServer()
        install_chart_crd_validators(self.server)
        self.kube = InMemoryClient(self.server)
        self.catalog = InstanceTypeProvider(region)
        self.agent_pools = FakeAgentPools(
            create_latency=create_latency,
            delete_latency=delete_latency,
            api_latency=api_latency,
        )
        self.aks = AKSSimulator(
            self.kube,
            self.agent_pools,
            ready_latency=ready_latency,
            plugin_latency=plugin_latency,
            gpu_count_for=self.catalog.gpu_count,
        )
        self.instances = InstanceProvider(
            self.agent_pools,
            self.kube,
            self.catalog,
            resource_group="rg",
            cluster_name="cluster",
            node_wait_interval=node_wait_interval,
        )
        cloud_kwargs = {}
        if repair_toleration is not None:
            cloud_kwargs["repair_toleration"] = repair_toleration
        if gpu_repair_toleration is not None:
            cloud_kwargs["gpu_repair_toleration"] = gpu_repair_toleration
        self.cloud = MetricsDecorator(
            AzureCloudProvider(self.instances, self.catalog, **cloud_kwargs)
        )
        self.recorder = EventRecorder(self.kube)
        self.informers = InformerFactory(self.kube)
        self.nodeclaims: Informer = self.informers.informer(karpv1.API_VERSION, karpv1.KIND_NODECLAIM)
        self.nodes: Informer = self.informers.informer("v1", "Node")
        self.instances.set_nodes_informer(self.nodes)
        self.pods: Informer = self.informers.informer("v1", "Pod")
        self.pods.add_index("nodeName", lambda o: o.get("spec", {}).get("nodeName") or None)
        self.volumeattachments: Informer = self.informers.informer(
            "storage.k8s.io/v1", "VolumeAttachment"
        )
        self.controllers: list = []
        self._started = False

    def add_all_controllers(
        self,
        *,
        lifecycle_workers: int = 64,
        termination_workers: int = 32,
        termination_requeue: float = 0.05,
        drain_requeue: float = 0.05,
        instance_poll: float = 0.05,
        gc_interval: float = 0.5,
        adoption_age: float = 0.2,
        with_health: bool = True,
        with_drift: bool = True,
        drift_interval: Optional[float] = None,
        drift_replace: bool = False,
    ) -> "Harness":
        """Wire the full controller set (the production main() topology) with
        test-friendly cadences."""
        from ..controllers.drift.controller import DriftController
        from ..controllers.garbagecollection.controller import (
            InstanceGCController,
            NodeClaimGCController,
        )
        from ..controllers.health.controller import HealthController
        from ..controllers.lifecycle.controller import LifecycleController
        from ..controllers.termination.controller import TerminationController
        from ..controllers.termination.eviction import EvictionQueue

        self.eviction_queue = EvictionQueue(self.kube, self.recorder, workers=8)
        self.lifecycle = LifecycleController(
            self.kube, self.cloud, self.recorder, self.nodeclaims, self.nodes,
            workers=lifecycle_workers, termination_requeue=termination_requeue,
        )
        self.termination = TerminationController(
            self.kube, self.cloud, self.recorder, self.nodes, self.nodeclaims,
            self.pods, self.volumeattachments, self.eviction_queue,
            workers=termination_workers, drain_requeue=drain_requeue, instance_poll=instance_poll,
        )
        self.instance_gc = InstanceGCController(
            self.kube, self.cloud, self.recorder,
            interval=gc_interval, adoption_age=adoption_age, nodes=self.nodes,
        )
        self.nodeclaim_gc = NodeClaimGCController(
            self.kube, self.cloud, self.recorder, interval=gc_interval,
            nodes=self.nodes,
        )
        self.controllers += [
            self.eviction_queue, self.lifecycle, self.termination,
            self.instance_gc, self.nodeclaim_gc,
        ]
        if with_health:
            self.health = HealthController(
                self.kube, self.cloud, self.recorder, self.nodes, self.nodeclaims
            )
            self.controllers.append(self.health)
        if with_drift:
            self.drift = DriftController(
                self.kube, self.cloud, self.recorder,
                interval=drift_interval if drift_interval is not None else gc_interval,
                replace=drift_replace,
            )
            self.controllers.append(self.drift)
        return self

    async def crash_restart_controllers(self, **controller_kwargs) -> None:
        """Simulate a controller-manager crash+restart: stop every
        controller, discard ALL in-memory controller state (workqueues,
        launch idempotency caches, eviction dedup) and start fresh
        instances against the same apiserver/cloud state. Informers re-sync
        from the server as a restarted manager's caches would."""
        for c in self.controllers:
            await c.controller.stop()
        self.controllers = []
        # drop the dead controllers' event handlers (their queues are shut)
        for inf in self.informers._informers.values():
            inf._handlers.clear()
        self.add_all_controllers(**controller_kwargs)
        for c in self.controllers:
            c.controller.start()

    # -- lifecycle -----------------------------------------------------------

    async def start(self) -> None:
        self.informers.start_all()
        await self.informers.wait_for_sync()
        for c in self.controllers:
            c.controller.start()
        if self._gc_pacer_enabled:
            # production Manager topology: paced cyclic GC once steady
            from ..operator.gcpacer import GCPacer

            self.gc_pacer = GCPacer()
            self.gc_pacer.engage()
        else:
            self.gc_pacer = None
        self._started = True

    async def stop(self) -> None:
        if getattr(self, "gc_pacer", None) is not None:
            await self.gc_pacer.disengage()
        for c in self.controllers:
            await c.controller.stop()
        await self.informers.stop_all()

    # -- helpers -------------------------------------------------------------

    def make_nodeclaim(
        self,
        name: str,
        vm_size: str = "Standard_ND128isr_MI355X_v6",
        labels: Optional[dict] = None,
        **spec_overrides,
    ) -> dict:
        nc = karpv1.new_nodeclaim(
            name,
            labels=labels if labels is not None else {karpv1.KAITO_WORKSPACE_LABEL_KEY: "ws"},
        )
        nc["spec"] = {
            "requirements": [
                {
                    "key": karpv1.INSTANCE_TYPE_LABEL_KEY,
                    "operator": "In",
                    "values": [vm_size],
                }
            ],
            "resources": {"requests": {karpv1.AMD_GPU_RESOURCE: str(self.catalog.gpu_count(vm_size) or 8)}},
            "nodeClassRef": {"group": "kaito.sh", "kind": "KaitoNodeClass", "name": "default"},
            **spec_overrides,
        }
        return nc

    async def wait_for(self, predicate, timeout: float = 10.0, interval: float = 0.01):
        """Poll an async predicate until truthy; returns its value."""
        deadline = asyncio.get_event_loop().time() + timeout
        while True:
            val = await predicate()
            if val:
                return val
            if asyncio.get_event_loop().time() > deadline:
                raise TimeoutError("condition not met within timeout")
            await asyncio.sleep(interval)

    def _informer_for(self, api_version: str, kind: str):
        for inf in (self.nodeclaims, self.nodes, self.pods, self.volumeattachments):
            if inf.api_version == api_version and inf.kind == kind:
                return inf if inf.has_synced else None
        return None

    async def wait_initialized(
        self, name: str, timeout: float = 10.0, interval: float = 0.01
    ) -> dict:
        """Wait through the watch surface (the same view a real client like
        KAITO observes): event-driven on the NodeClaim informer (handler
        replay covers already-initialized claims), polling fallback before
        sync. A polling wait at N-hundred concurrent waiters floods the
        loop with timers."""
        inf = self._informer_for(karpv1.API_VERSION, karpv1.KIND_NODECLAIM)
        if inf is not None:
            def pred(event_type: str, obj):
                if event_type in ("DELETED", "ABSENT") or obj is None:
                    return None
                return obj if karpv1.is_initialized(obj) else None

            try:
                found = await inf.wait_until(pred, name=name, timeout=timeout)
            except asyncio.TimeoutError:
                raise TimeoutError(f"NodeClaim {name} not Initialized within {timeout}s")
            # private copy: callers may mutate (cache objects are shared)
            return ko.deep_copy(found)

        async def check():
            try:
                nc = await self.kube.get(karpv1.API_VERSION, karpv1.KIND_NODECLAIM, name)
            except Exception:
                return None
            return nc if nc and karpv1.is_initialized(nc) else None

        return ko.deep_copy(await self.wait_for(check, timeout, interval))

    async def wait_gone(
        self, api_version: str, kind: str, name: str, timeout: float = 10.0,
        interval: float = 0.01,
    ):
        inf = self._informer_for(api_version, kind)
        if inf is not None:
            def pred(event_type: str, obj):
                return True if event_type in ("DELETED", "ABSENT") else None

            try:
                return await inf.wait_until(pred, name=name, timeout=timeout)
            except asyncio.TimeoutError:
                raise TimeoutError(f"{kind} {name} still present after {timeout}s")

        async def check():
            try:
                await self.kube.get(api_version, kind, name)
                return None
            except Exception:
                return True

        return await self.wait_for(check, timeout, interval)
