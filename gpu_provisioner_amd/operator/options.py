"""Operator options: flags with env-var fallbacks + feature gates.

Spec: reference vendor/sigs.k8s.io/karpenter/pkg/operator/options/
options.go:110-161 — the same surface (METRICS_PORT, HEALTH_PROBE_PORT,
KUBE_CLIENT_QPS/BURST, ENABLE_PROFILING, LEADER_ELECT, FEATURE_GATES with
NodeRepair=true default, DISABLE_WEBHOOK, BATCH_* unused here) so the
reference's Helm values drive this binary unchanged.
"""
from __future__ import annotations

import argparse
import os
from dataclasses import dataclass, field


@dataclass
class FeatureGates:
    node_repair: bool = True
    # Drift marks NodeClaims whose agent pool no longer matches their spec
    # (Drifted condition); DriftReplace additionally deletes them so the
    # owner (KAITO) re-creates a conforming node. Detection defaults on,
    # replacement defaults off (parity with the reference, which has no
    # active disruption controllers).
    drift: bool = True
    drift_replace: bool = False
    # delete NodeClaims that never reach Registered within the 30-min
    # liveness window; off by default (the reference ships liveness
    # disabled — AKS agent-pool creates can legitimately run long)
    registration_liveness: bool = False

    @classmethod
    def parse(cls, s: str) -> "FeatureGates":
        gates = cls()
        for part in (s or "").split(","):
            part = part.strip()
            if not part or "=" not in part:
                continue
            name, val = part.split("=", 1)
            enabled = val.strip().lower() == "true"
            name = name.strip()
            if name == "NodeRepair":
                gates.node_repair = enabled
            elif name == "Drift":
                gates.drift = enabled
            elif name == "DriftReplace":
                gates.drift_replace = enabled
            elif name == "RegistrationLiveness":
                gates.registration_liveness = enabled
        return gates


@dataclass
class Options:
    service_name: str = ""
    metrics_port: int = 8080
    health_probe_port: int = 8081
    kube_client_qps: float = 200.0
    kube_client_burst: int = 300
    enable_profiling: bool = False
    leader_elect: bool = False
    leader_election_name: str = "gpu-provisioner-amd-leader"
    leader_election_namespace: str = "kube-system"
    disable_webhook: bool = True
    log_level: str = "info"
    # paced cyclic GC (operator/gcpacer.py) — on by default; GC_PACER=false
    # restores CPython's automatic full collections
    gc_pacer: bool = True
    feature_gates: FeatureGates = field(default_factory=FeatureGates)

    @classmethod
    def from_env_and_args(cls, argv=None, environ=None) -> "Options":
        env = environ if environ is not None else os.environ

        def envv(key, default):
            return env.get(key, default)

        p = argparse.ArgumentParser("gpu-provisioner-amd")
        from .. import __version__

        p.add_argument("--version", action="version", version=f"gpu-provisioner-amd {__version__}")
        p.add_argument("--karpenter-service", default=envv("KARPENTER_SERVICE", ""))
        p.add_argument("--metrics-port", type=int, default=int(envv("METRICS_PORT", "8080")))
        p.add_argument(
            "--health-probe-port", type=int, default=int(envv("HEALTH_PROBE_PORT", "8081"))
        )
        p.add_argument(
            "--kube-client-qps", type=float, default=float(envv("KUBE_CLIENT_QPS", "200"))
        )
        p.add_argument(
            "--kube-client-burst", type=int, default=int(envv("KUBE_CLIENT_BURST", "300"))
        )
        p.add_argument(
            "--enable-profiling",
            action="store_true",
            default=envv("ENABLE_PROFILING", "false").lower() == "true",
        )
        p.add_argument(
            "--leader-elect",
            action="store_true",
            default=envv("LEADER_ELECT", "false").lower() == "true",
        )
        p.add_argument(
            "--leader-election-namespace",
            default=envv("LEADER_ELECTION_NAMESPACE", env.get("SYSTEM_NAMESPACE", "kube-system")),
        )
        p.add_argument(
            "--disable-webhook",
            action="store_true",
            default=envv("DISABLE_WEBHOOK", "true").lower() == "true",
        )
        p.add_argument("--log-level", default=envv("LOG_LEVEL", "info"))
        p.add_argument(
            "--gc-pacer",
            default=envv("GC_PACER", "true").lower() == "true",
            action="store_true",
        )
        p.add_argument(
            "--feature-gates", default=envv("FEATURE_GATES", "NodeRepair=true")
        )
        args = p.parse_args(argv if argv is not None else [])
        return cls(
            service_name=args.karpenter_service,
            metrics_port=args.metrics_port,
            health_probe_port=args.health_probe_port,
            kube_client_qps=args.kube_client_qps,
            kube_client_burst=args.kube_client_burst,
            enable_profiling=args.enable_profiling,
            leader_elect=args.leader_elect,
            leader_election_namespace=args.leader_election_namespace,
            disable_webhook=args.disable_webhook,
            log_level=args.log_level,
            gc_pacer=args.gc_pacer,
            feature_gates=FeatureGates.parse(args.feature_gates),
        )
