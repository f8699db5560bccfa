"""Entrypoint: wire the operator, decorate the cloud provider, register
controllers, run the manager.

Spec: reference cmd/controller/main.go:34-59 — NewOperator → metrics-
decorated CloudProvider → karpenter controllers (lifecycle, node
termination+eviction, nodeclaim GC, node health when NodeRepair is gated on)
+ the first-party instance GC → manager Start.

Run in-cluster: ``python -m gpu_provisioner_amd``.
"""
from __future__ import annotations

import asyncio
import logging

from .apis import v1 as karpv1
from .cloudprovider.decorator import MetricsDecorator
from .controllers.drift.controller import DriftController
from .controllers.garbagecollection.controller import (
    InstanceGCController,
    NodeClaimGCController,
)
from .controllers.health.controller import HealthController
from .controllers.lifecycle.controller import LifecycleController
from .controllers.termination.controller import TerminationController
from .controllers.termination.eviction import EvictionQueue
from .events.recorder import EventRecorder
from .kube.http import HTTPClient
from .kube.informer import InformerFactory
from .operator.logging import setup_logging
from .operator.manager import Manager
from .operator.operator import Operator
from .operator.options import Options

log = logging.getLogger("gpu-provisioner-amd")


def build_manager(kube, options: Options, cloud_provider, version: str = "0.1.0") -> Manager:
    """Assemble informers + the full controller set on any KubeClient (the
    production HTTPClient or the in-memory fake in tests)."""
    cloud = MetricsDecorator(cloud_provider)
    informers = InformerFactory(kube)
    nodeclaims = informers.informer(karpv1.API_VERSION, karpv1.KIND_NODECLAIM)
    nodes = informers.informer("v1", "Node")
    # cache-backed node lookups for the instance provider's node wait
    inner = getattr(cloud_provider, "instances", None)
    if inner is not None and hasattr(inner, "set_nodes_informer"):
        inner.set_nodes_informer(nodes)
    pods = informers.informer("v1", "Pod")
    volumeattachments = informers.informer("storage.k8s.io/v1", "VolumeAttachment")
    recorder = EventRecorder(kube)

    from .controllers.lifecycle.controller import REGISTRATION_TTL_SECONDS

    eviction_queue = EvictionQueue(kube, recorder)
    controllers = [
        eviction_queue,
        LifecycleController(
            kube, cloud, recorder, nodeclaims, nodes,
            registration_ttl=(
                REGISTRATION_TTL_SECONDS
                if options.feature_gates.registration_liveness
                else None
            ),
        ),
        TerminationController(
            kube, cloud, recorder, nodes, nodeclaims, pods, volumeattachments, eviction_queue
        ),
        InstanceGCController(kube, cloud, recorder, nodes=nodes),
        NodeClaimGCController(kube, cloud, recorder, nodes=nodes),
    ]
    # node.health is gated on RepairPolicies + the NodeRepair feature gate
    # (reference vendor/.../controllers/controllers.go:109-111)
    if options.feature_gates.node_repair and cloud.repair_policies():
        controllers.append(HealthController(kube, cloud, recorder, nodes, nodeclaims))
    # drift detection (net-new: the reference stubs IsDrifted); replacement
    # only when DriftReplace is explicitly gated on
    if options.feature_gates.drift:
        controllers.append(
            DriftController(kube, cloud, recorder, replace=options.feature_gates.drift_replace)
        )

    manager = Manager(
        kube,
        options,
        informers,
        required_crds=((karpv1.API_VERSION, karpv1.KIND_NODECLAIM),),
        version=version,
    )
    manager.register(*controllers)
    return manager


async def amain() -> None:
    options = Options.from_env_and_args()
    setup_logging(options.log_level)
    kube = HTTPClient.from_service_account(
        qps=options.kube_client_qps, burst=options.kube_client_burst
    )
    operator = Operator(kube)
    manager = build_manager(kube, options, operator.cloud_provider)
    log.info("starting gpu-provisioner-amd (MI355X)")
    await manager.run_forever()


def main() -> None:
    asyncio.run(amain())


if __name__ == "__main__":
    main()
